"""Train on a ray.data.Dataset (reference examples/simple_ray_dataset.py).

Requires `ray[data]` (not shipped in this image): exits gracefully when
the import fails. The dataset is split into one shard per actor.
"""

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train


def main():
    try:
        import pandas as pd
        import ray
    except ImportError:
        print("ray is not installed - skipping (install `ray[data]`)")
        return

    rng = np.random.RandomState(3)
    pdf = pd.DataFrame(
        rng.randn(100_000, 6).astype(np.float32),
        columns=[f"f{i}" for i in range(6)],
    )
    pdf["label"] = (pdf["f0"] > 0).astype(np.float32)
    ds = ray.data.from_pandas(pdf)

    dm = RayDMatrix(ds, label="label")
    res = {}
    train(
        {"objective": "binary:logistic", "eval_metric": ["error"]},
        dm, 20, evals=[(dm, "train")], evals_result=res,
        verbose_eval=False, ray_params=RayParams(num_actors=2),
    )
    print(f"Final training error: {res['train']['error'][-1]:.4f}")


if __name__ == "__main__":
    main()
