"""Train on a `__partitioned__`-protocol object
(reference examples/simple_partitioned.py)."""

import numpy as np
import pandas as pd

from xgboost_ray_amd import RayDMatrix, RayParams, train


class PartitionedTable:
    """Any object exposing the __partitioned__ dict protocol works."""

    def __init__(self, frames):
        self.__partitioned__ = {
            "get": lambda objs: objs,
            "shape": (sum(len(f) for f in frames), frames[0].shape[1]),
            "partition_tiling": (len(frames), 1),
            "partitions": {
                (i, 0): {
                    "start": (sum(len(f) for f in frames[:i]), 0),
                    "shape": f.shape,
                    "data": f,
                    "location": ["127.0.0.1"],
                }
                for i, f in enumerate(frames)
            },
        }


def main():
    rng = np.random.RandomState(7)
    frames = []
    for _ in range(4):
        df = pd.DataFrame(
            rng.randn(25_000, 8).astype(np.float32),
            columns=[f"f{i}" for i in range(8)],
        )
        df["label"] = (df["f0"] + 0.5 * df["f1"] > 0).astype(np.float32)
        frames.append(df)

    dm = RayDMatrix(PartitionedTable(frames), label="label")
    res = {}
    bst = train(
        {"objective": "binary:logistic", "eval_metric": ["error"]},
        dm, 20, evals=[(dm, "train")], evals_result=res,
        verbose_eval=False, ray_params=RayParams(num_actors=2),
    )
    print(f"Final training error: {res['train']['error'][-1]:.4f}")
    bst.save_model("partitioned.ubj")
    print("Model saved: partitioned.ubj")


if __name__ == "__main__":
    main()
