"""Per-actor-process session (reference session.py:8-81).

Inside a training actor, user callbacks can query their rank and push
arbitrary values to the driver through the queue (surfacing as
``additional_results["callback_returns"]``, reference main.py:902-922).
"""

import threading
from typing import Any, Optional


class _Session:
    def __init__(self, rank: int, world_size: int, queue=None):
        self.rank = rank
        self.world_size = world_size
        self.queue = queue


_session: Optional[_Session] = None
_lock = threading.Lock()


def init_session(rank: int, world_size: int, queue=None):
    global _session
    with _lock:
        _session = _Session(rank, world_size, queue)


def set_session_queue(queue):
    global _session
    with _lock:
        if _session is None:
            _session = _Session(0, 1, queue)
        else:
            _session.queue = queue


def shutdown_session():
    global _session
    with _lock:
        _session = None


def get_session() -> _Session:
    if _session is None:
        raise ValueError(
            "Session not initialized - this function must be called inside "
            "a training actor."
        )
    return _session


def get_actor_rank() -> int:
    return get_session().rank


def get_world_size() -> int:
    return get_session().world_size


def get_rabit_rank() -> int:
    """Collective-communicator rank == actor rank (reference session.py:68-76,
    where it reads the Rabit/xgb.collective rank)."""
    return get_session().rank


def put_queue(item: Any):
    """Push a value to the driver (keyed by this actor's rank)."""
    sess = get_session()
    if sess.queue is None:
        raise ValueError("No queue attached to this session")
    sess.queue.put((sess.rank, item))
