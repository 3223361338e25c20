"""Tune-style tuning loop with per-trial checkpoints
(reference examples/simple_tune.py).

With Ray installed, TuneReportCheckpointCallback reports through
ray.train and attaches real Tune checkpoints. Without it (this image),
the same callback writes `checkpoint_NNNNNN/<filename>` into
`results_dir` and this script runs the sweep inline.
"""

import os
import shutil
import tempfile

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, train
from xgboost_ray_amd.tune import TuneReportCheckpointCallback, load_model


def train_one(config, data, results_dir):
    X, y = data
    dm = RayDMatrix(X, label=y)
    res = {}
    train(
        {
            "objective": "binary:logistic",
            "eval_metric": ["logloss"],
            **config,
        },
        dm, 12,
        evals=[(dm, "train")],
        evals_result=res,
        verbose_eval=False,
        ray_params=RayParams(num_actors=2),
        callbacks=[TuneReportCheckpointCallback(
            filename="model.ubj", frequency=4, results_dir=results_dir
        )],
    )
    return res["train"]["logloss"][-1]


def main():
    rng = np.random.RandomState(42)
    X = rng.randn(50_000, 10).astype(np.float32)
    y = ((X[:, 0] + 0.3 * X[:, 1]) > 0).astype(np.float32)

    space = [{"eta": e, "max_depth": d}
             for e in (0.1, 0.3) for d in (4, 6)]
    base = tempfile.mkdtemp(prefix="rxgb_tune_")
    best = None
    try:
        for i, cfg in enumerate(space):
            trial_dir = os.path.join(base, f"trial_{i}")
            os.makedirs(trial_dir, exist_ok=True)
            score = train_one(cfg, (X, y), trial_dir)
            print(f"trial {i} {cfg}: logloss={score:.5f}")
            if best is None or score < best[0]:
                best = (score, cfg, trial_dir)
        score, cfg, trial_dir = best
        ckpts = sorted(os.listdir(trial_dir))
        print(f"Best config: {cfg} (logloss={score:.5f}); "
              f"checkpoints: {ckpts}")
        bst = load_model(os.path.join(trial_dir, ckpts[-1], "model.ubj"))
        print(f"Reloaded best model: {bst.num_boosted_rounds()} rounds")
    finally:
        shutil.rmtree(base, ignore_errors=True)


if __name__ == "__main__":
    main()
