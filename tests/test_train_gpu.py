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
