"""Distributed per-actor parquet loading (reference higgs_parquet.py /
train_with_ml_dataset.py analogue)."""

import os
import tempfile

import numpy as np
import pandas as pd

from xgboost_ray_amd import RayDMatrix, RayParams, train


def main():
    tmp = tempfile.mkdtemp(prefix="rxgb_parquet_")
    files = []
    rng = np.random.RandomState(0)
    for i in range(8):
        X = rng.randn(20_000, 12).astype(np.float32)
        y = ((X[:, 0] - 0.5 * X[:, 4]) > 0).astype(np.float32)
        df = pd.DataFrame(X, columns=[f"f{j}" for j in range(12)])
        df["label"] = y
        path = os.path.join(tmp, f"part-{i}.parquet")
        df.to_parquet(path)
        files.append(path)

    # each actor loads only its own files (FIXED sharding)
    dtrain = RayDMatrix(files, label="label")
    res = {}
    train(
        {"objective": "binary:logistic", "eval_metric": ["logloss"]},
        dtrain, 20, evals=[(dtrain, "train")], evals_result=res,
        ray_params=RayParams(num_actors=4),
    )
    print("Final logloss:", res["train"]["logloss"][-1])


if __name__ == "__main__":
    main()
