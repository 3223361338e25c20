"""Control-plane realism soak: 200 rounds of the PUBLIC train() API on
the actor layer (shm-store central loading of an 11M x 28 matrix, queue
checkpoints every 10 rounds, per-round eval) on one GPU."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import time

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train


def main():
    rng = np.random.RandomState(7)
    n, F = 11_000_000, 28
    X = rng.randn(n, F).astype(np.float32)
    w = rng.randn(F).astype(np.float32) * 0.5
    y = ((X @ w + rng.randn(n).astype(np.float32)) > 0).astype(np.float32)

    dm = RayDMatrix(X, label=y)
    res = {}
    t0 = time.perf_counter()
    bst = train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 8, "eta": 0.1, "max_bin": 256,
         "eval_metric": ["logloss", "auc"]},
        dm, 200,
        evals=[(dm, "train")],
        evals_result=res,
        ray_params=RayParams(num_actors=1, gpus_per_actor=1,
                             checkpoint_frequency=10),
    )
    dt = time.perf_counter() - t0
    print("train() control-plane soak: %.2f s total (%.2f ms/round "
          "incl. load+actors), final auc=%.4f, %d rounds"
          % (dt, dt * 1000 / 200, res["train"]["auc"][-1],
             bst.num_boosted_rounds()))


if __name__ == "__main__":
    main()
