"""HIGGS-style training example (reference examples/higgs.py:41-61).

The real HIGGS CSV (11M x 28) is not downloadable in this offline
environment; pass --csv if you have it locally, otherwise a synthetic
matrix of the same shape is used.
"""

import argparse
import time

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train

colnames = ["label"] + ["feature-%02d" % i for i in range(1, 29)]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--csv", default=None, help="path to HIGGS.csv")
    p.add_argument("--rows", type=int, default=1_000_000)
    p.add_argument("--actors", type=int, default=4)
    p.add_argument("--gpu", action="store_true")
    args = p.parse_args()

    if args.csv:
        dtrain = RayDMatrix(
            args.csv, label="label", names=colnames
        )
    else:
        rng = np.random.RandomState(2)
        X = rng.randn(args.rows, 28).astype(np.float32)
        y = ((X[:, 0] + 0.3 * X[:, 1] ** 2) > 0.5).astype(np.float32)
        dtrain = RayDMatrix(X, label=y)

    config = {
        "tree_method": "gpu_hist" if args.gpu else "hist",
        "eval_metric": ["logloss", "error"],
        "max_depth": 8,
    }

    evals_result = {}
    start = time.time()
    bst = train(
        config,
        dtrain,
        evals_result=evals_result,
        num_boost_round=100,
        evals=[(dtrain, "train")],
        ray_params=RayParams(num_actors=args.actors),
    )
    taken = time.time() - start
    print(f"TRAIN TIME TAKEN: {taken:.2f} seconds")

    bst.save_model("higgs.json")
    print(
        "Final training error: {:.4f}".format(
            evals_result["train"]["error"][-1]
        )
    )


if __name__ == "__main__":
    main()
