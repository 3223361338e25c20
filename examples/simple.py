"""Minimal distributed training example (reference examples/simple.py)."""

import argparse

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train


def main(cpus_per_actor, num_actors):
    rng = np.random.RandomState(1234)
    n, f = 100_000, 16
    X = rng.randn(n, f).astype(np.float32)
    y = ((X[:, 0] + X[:, 3] * 0.5) > 0).astype(np.float32)

    train_set = RayDMatrix(X, label=y)
    evals_result = {}
    bst = train(
        {
            "objective": "binary:logistic",
            "eval_metric": ["logloss", "error"],
        },
        train_set,
        num_boost_round=30,
        evals_result=evals_result,
        evals=[(train_set, "train")],
        verbose_eval=False,
        ray_params=RayParams(
            num_actors=num_actors, cpus_per_actor=cpus_per_actor
        ),
    )
    bst.save_model("simple.json")
    print(
        "Final training error: {:.4f}".format(
            evals_result["train"]["error"][-1]
        )
    )


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--num-actors", type=int, default=2)
    parser.add_argument("--cpus-per-actor", type=int, default=1)
    args = parser.parse_args()
    main(args.cpus_per_actor, args.num_actors)
