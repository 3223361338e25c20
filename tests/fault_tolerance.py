"""Fault-injection harness (reference tests/fault_tolerance.py:14-109).

``DieCallback`` kills the current actor process with SIGKILL at a chosen
boosting round; a lock file makes sure each (rank, round) kill fires only
once across restarts (reference tests/utils.py:111-142 lock files).
"""

import os
import signal


class SlowCallback:
    """Slows every boosting round down so elastic staging can win its
    race against training completion (reference DelayedLoadingCallback
    analogue, tests/fault_tolerance.py:68-80)."""

    def __init__(self, delay_s: float):
        self.delay_s = delay_s

    def after_iteration(self, booster, iteration, evals_log) -> bool:
        import time

        time.sleep(self.delay_s)
        return False


class DieCallback:
    def __init__(self, die_round: int, die_rank: int = 1, lock_dir: str = "/tmp"):
        self.die_round = die_round
        self.die_rank = die_rank
        self.lock_dir = lock_dir

    def after_iteration(self, booster, iteration, evals_log) -> bool:
        from xgboost_ray_amd.session import get_actor_rank

        rank = get_actor_rank()
        if rank != self.die_rank:
            return False
        if iteration != self.die_round:
            return False
        lock = os.path.join(
            self.lock_dir, f"die_{self.die_rank}_{self.die_round}.lock"
        )
        if os.path.exists(lock):
            return False
        with open(lock, "w") as f:
            f.write("died")
        os.kill(os.getpid(), signal.SIGKILL)
        return False  # unreachable
