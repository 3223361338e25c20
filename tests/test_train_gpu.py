"""End-to-end train()/predict() through the actor layer on a real GPU."""

import numpy as np
import pytest

from tests.utils import create_data
from xgboost_ray_amd import RayDMatrix, RayParams, predict, train

pytestmark = pytest.mark.gpu


def test_train_single_gpu_actor():
    X, y = create_data(200_000, 12)
    res, add = {}, {}
    bst = train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 6, "eta": 0.3, "eval_metric": ["logloss", "auc"]},
        RayDMatrix(X, label=y),
        10,
        evals_result=res,
        additional_results=add,
        ray_params=RayParams(num_actors=1, gpus_per_actor=1),
        evals=[(RayDMatrix(X, label=y), "train")],
    )
    assert bst.num_boosted_rounds() == 10
    assert res["train"]["auc"][-1] > 0.9
    pred = predict(
        bst, RayDMatrix(X), ray_params=RayParams(num_actors=1, gpus_per_actor=1)
    )
    acc = ((pred > 0.5) == (y > 0.5)).mean()
    assert acc > 0.85


def test_train_gpu_fault_restart(tmp_path):
    """SIGKILL mid-training + checkpoint restart on the GPU path."""
    from tests.fault_tolerance import DieCallback

    X, y = create_data(100_000, 8)
    bst = train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 5},
        RayDMatrix(X, label=y),
        12,
        ray_params=RayParams(num_actors=1, gpus_per_actor=1,
                             max_actor_restarts=1, checkpoint_frequency=3),
        callbacks=[DieCallback(die_round=6, die_rank=0,
                               lock_dir=str(tmp_path))],
    )
    assert bst.num_boosted_rounds() == 12


def test_streaming_train_gpu(tmp_path):
    """Out-of-core path (RayDeviceQuantileDMatrix) through GPU actors."""
    import pandas as pd

    from xgboost_ray_amd import RayDeviceQuantileDMatrix

    rng = np.random.RandomState(0)
    files = []
    for i in range(4):
        Xc = rng.randn(100_000, 10).astype(np.float32)
        yc = ((Xc[:, 0] + Xc[:, 1]) > 0).astype(np.float32)
        df = pd.DataFrame(Xc, columns=[f"f{j}" for j in range(10)])
        df["label"] = yc
        path = str(tmp_path / f"s{i}.parquet")
        df.to_parquet(path)
        files.append(path)
    res = {}
    dm = RayDeviceQuantileDMatrix(files, label="label")
    bst = train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 6, "eval_metric": ["auc"]},
        dm,
        8,
        evals=[(dm, "train")],
        evals_result=res,
        ray_params=RayParams(num_actors=1, gpus_per_actor=1),
    )
    assert bst.num_boosted_rounds() == 8
    assert res["train"]["auc"][-1] > 0.9


def test_deep_tree_gpu():
    """depth 12 / 4096-node frontiers exercise the chunked machinery."""
    X, y = create_data(300_000, 10)
    res = {}
    train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 12, "eta": 0.3, "eval_metric": ["logloss"]},
        RayDMatrix(X, label=y), 4,
        evals_result=res,
        evals=[(RayDMatrix(X, label=y), "train")],
        ray_params=RayParams(num_actors=1, gpus_per_actor=1),
    )
    assert res["train"]["logloss"][-1] < 0.3
