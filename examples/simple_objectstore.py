"""Object-store partitions example (reference examples/simple_objectstore.py):
put shards into the shared-memory store, train from the refs."""

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train
from xgboost_ray_amd import shm_store


def main():
    rng = np.random.RandomState(7)
    parts = []
    for i in range(4):
        X = rng.randn(10_000, 8).astype(np.float32)
        y = ((X[:, 0] + X[:, 1]) > 0).astype(np.float32)
        import pandas as pd

        df = pd.DataFrame(X, columns=[f"f{j}" for j in range(8)])
        df["label"] = y
        parts.append(shm_store.put(df))

    dtrain = RayDMatrix(parts, label="label")
    res = {}
    train(
        {"objective": "binary:logistic", "eval_metric": ["error"]},
        dtrain, 20, evals=[(dtrain, "train")], evals_result=res,
        ray_params=RayParams(num_actors=2),
    )
    print("Final training error:", res["train"]["error"][-1])
    shm_store.get_store().shutdown()


if __name__ == "__main__":
    main()
