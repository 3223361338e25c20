"""Train on a dask DataFrame (reference examples/simple_dask.py).

Requires `dask` (not shipped in this image): exits gracefully when the
import fails. With a dask.distributed cluster running, partition
locality is probed via `who_has` and shards stay node-local.
"""

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train


def main():
    try:
        import dask.dataframe as dd
        import pandas as pd
    except ImportError:
        print("dask is not installed - skipping (install `dask[dataframe]`)")
        return

    rng = np.random.RandomState(3)
    pdf = pd.DataFrame(
        rng.randn(100_000, 8).astype(np.float32),
        columns=[f"f{i}" for i in range(8)],
    )
    pdf["label"] = (pdf["f0"] > 0).astype(np.float32)
    ddf = dd.from_pandas(pdf, npartitions=4)

    dm = RayDMatrix(ddf, label="label")
    res = {}
    train(
        {"objective": "binary:logistic", "eval_metric": ["error"]},
        dm, 20, evals=[(dm, "train")], evals_result=res,
        verbose_eval=False, ray_params=RayParams(num_actors=2),
    )
    print(f"Final training error: {res['train']['error'][-1]:.4f}")


if __name__ == "__main__":
    main()
