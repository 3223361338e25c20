"""Driver-side utilities (reference util.py:1-110).

The reference's Ray-actor-based Event/Queue/MultiActorTask primitives map
to multiprocessing primitives on the single 8xMI355X node: events and
queues are kernel objects shared with actor processes at spawn.
"""

import socket
import threading
import time
from typing import List, Optional


def find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class Future:
    """Result of an async actor call (Ray ObjectRef equivalent)."""

    def __init__(self, actor=None, method: str = ""):
        self._event = threading.Event()
        self._result = None
        self._error: Optional[BaseException] = None
        self.actor = actor
        self.method = method

    def set_result(self, result):
        self._result = result
        self._event.set()

    def set_error(self, err: BaseException):
        self._error = err
        self._event.set()

    def done(self) -> bool:
        return self._event.is_set()

    def result(self, timeout: Optional[float] = None):
        if not self._event.wait(timeout):
            raise TimeoutError(
                f"Actor call {self.method} did not complete in {timeout}s"
            )
        if self._error is not None:
            raise self._error
        return self._result


def wait_futures(
    futures: List[Future], timeout: Optional[float] = None
) -> (list, list):
    """ray.wait equivalent: returns (ready, not_ready)."""
    deadline = None if timeout is None else time.monotonic() + timeout
    while True:
        ready = [f for f in futures if f.done()]
        if ready or (deadline is not None and time.monotonic() >= deadline):
            return ready, [f for f in futures if not f.done()]
        time.sleep(0.01)


def get_all(futures: List[Future], timeout: Optional[float] = None):
    return [f.result(timeout) for f in futures]


class MultiActorTask:
    """Tracks a set of futures until all complete (reference util.py:52-77)."""

    def __init__(self, futures: Optional[List[Future]] = None):
        self._futures = list(futures or [])

    def is_ready(self) -> bool:
        return all(f.done() for f in self._futures)


def force_on_current_node(obj=None):
    """Single-node deployment: placement is trivial (reference util.py:82-108)."""
    return obj
