"""Train on a Modin DataFrame (reference examples/simple_modin.py).

Requires `modin` (not shipped in this image): exits gracefully when the
import fails. Partition->node IPs come from
modin.distributed.dataframe.pandas.unwrap_partitions(get_ip=True).
"""

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train


def main():
    try:
        import modin.pandas as mpd
    except ImportError:
        print("modin is not installed - skipping (install `modin`)")
        return

    rng = np.random.RandomState(3)
    mdf = mpd.DataFrame(
        rng.randn(100_000, 9).astype(np.float32),
        columns=[f"f{i}" for i in range(9)],
    )
    mdf["label"] = (mdf["f0"] > 0).astype(np.float32)

    dm = RayDMatrix(mdf, label="label")
    res = {}
    train(
        {"objective": "binary:logistic", "eval_metric": ["error"]},
        dm, 20, evals=[(dm, "train")], evals_result=res,
        verbose_eval=False, ray_params=RayParams(num_actors=2),
    )
    print(f"Final training error: {res['train']['error'][-1]:.4f}")


if __name__ == "__main__":
    main()
