"""Fault-tolerance benchmark grid (reference tests/release/benchmark_ft.py).

Conditions (reference benchmark_ft.py:289-347):
  fewer_workers        - train with N - affected workers from the start
  non_elastic          - kill `affected` workers at 50% of rounds; restart
                         policy recreates them (tries += 1)
  elastic_no_comeback  - elastic training, killed workers never return
  elastic_comeback     - elastic training, workers return at 75% of rounds

Prints per-condition wall time and final error.
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import argparse
import os
import shutil
import tempfile
import time

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train


class _DieCallback:
    def __init__(self, die_round, ranks, lock_dir):
        self.die_round = die_round
        self.ranks = ranks
        self.lock_dir = lock_dir

    def after_iteration(self, booster, iteration, evals_log):
        import os
        import signal

        from xgboost_ray_amd.session import get_actor_rank

        rank = get_actor_rank()
        if rank not in self.ranks or iteration != self.die_round:
            return False
        lock = os.path.join(self.lock_dir, f"die_{rank}.lock")
        if os.path.exists(lock):
            return False
        with open(lock, "w") as f:
            f.write("x")
        os.kill(os.getpid(), signal.SIGKILL)
        return False


def run_condition(condition, X, y, num_workers, num_rounds, affected, gpu):
    lock_dir = tempfile.mkdtemp(prefix="rxgb_ft_")
    params = {
        "objective": "binary:logistic",
        "tree_method": "gpu_hist" if gpu else "hist",
        "max_depth": 6,
        "eval_metric": ["error"],
    }
    callbacks = None
    if condition == "fewer_workers":
        rp = RayParams(num_actors=num_workers - affected, max_actor_restarts=2)
    elif condition == "non_elastic":
        rp = RayParams(num_actors=num_workers, max_actor_restarts=2)
        callbacks = [
            _DieCallback(num_rounds // 2, set(range(1, 1 + affected)), lock_dir)
        ]
    elif condition == "elastic_no_comeback":
        os.environ["RXGB_ELASTIC_RESTART_DISABLED"] = "1"
        rp = RayParams(
            num_actors=num_workers, elastic_training=True,
            max_failed_actors=affected, max_actor_restarts=2,
        )
        callbacks = [
            _DieCallback(num_rounds // 2, set(range(1, 1 + affected)), lock_dir)
        ]
    elif condition == "elastic_comeback":
        os.environ.pop("RXGB_ELASTIC_RESTART_DISABLED", None)
        os.environ["RXGB_ELASTIC_RESTART_RESOURCE_CHECK_S"] = "2"
        os.environ["RXGB_ELASTIC_RESTART_GRACE_PERIOD_S"] = "2"
        rp = RayParams(
            num_actors=num_workers, elastic_training=True,
            max_failed_actors=affected, max_actor_restarts=2,
        )
        callbacks = [
            _DieCallback(num_rounds // 2, set(range(1, 1 + affected)), lock_dir)
        ]
    else:
        raise ValueError(condition)

    dtrain = RayDMatrix(X, label=y)
    res, add = {}, {}
    t0 = time.time()
    bst = train(
        params, dtrain, num_rounds, evals=[(dtrain, "train")],
        evals_result=res, additional_results=add, ray_params=rp,
        callbacks=callbacks,
    )
    elapsed = time.time() - t0
    shutil.rmtree(lock_dir, ignore_errors=True)
    return {
        "condition": condition,
        "affected": affected,
        "time_s": elapsed,
        "rounds": bst.num_boosted_rounds(),
        "final_error": res["train"]["error"][-1],
        "total_n": add.get("total_n"),
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--workers", type=int, default=4)
    p.add_argument("--rounds", type=int, default=20)
    p.add_argument("--rows", type=int, default=200_000)
    p.add_argument("--cols", type=int, default=20)
    p.add_argument("--affected", type=int, default=1)
    p.add_argument("--gpu", action="store_true")
    p.add_argument("--smoke-test", action="store_true")
    args = p.parse_args()
    if args.smoke_test:
        args.rows, args.rounds, args.workers = 10_000, 8, 2

    rng = np.random.RandomState(0)
    X = rng.randn(args.rows, args.cols).astype(np.float32)
    w = np.linspace(-1, 1, args.cols)
    y = ((X @ w + 0.5 * rng.randn(args.rows)) > 0).astype(np.float32)

    conditions = ["fewer_workers", "non_elastic"]
    if args.affected > 0:
        conditions += ["elastic_no_comeback", "elastic_comeback"]
    else:
        print("affected=0: skipping elastic conditions "
              "(max_failed_actors must be > 0)")
    for condition in conditions:
        out = run_condition(
            condition, X, y, args.workers, args.rounds, args.affected,
            args.gpu,
        )
        print(out, flush=True)


if __name__ == "__main__":
    main()
