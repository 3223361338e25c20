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


def test_tune_callback_payload_is_picklable():
    """The queue is a multiprocessing.Queue: its feeder thread pickles
    items, and closures are silently dropped. The report/checkpoint
    payload must survive a real pickle round-trip."""
    import pickle

    from xgboost_ray_amd.tune import _DriverReportAndCheckpoint

    item = _DriverReportAndCheckpoint(
        {"train-logloss": 0.5}, b"rawmodel", 4, "ckpt", "/tmp/x"
    )
    clone = pickle.loads(pickle.dumps(item))
    assert clone.metrics == {"train-logloss": 0.5}
    assert clone.raw_model == b"rawmodel"


def test_tune_checkpoint_files_written(tmp_path):
    """filename/frequency honored: a loadable checkpoint file lands in
    the trial dir every `frequency` rounds plus once at end
    (reference tune.py:26-48 upstream checkpoint semantics)."""
    from xgboost_ray_amd.tune import load_model

    X, y = create_data(600, 4)
    train(
        {"objective": "binary:logistic", "eval_metric": ["logloss"]},
        RayDMatrix(X, label=y), 7,
        evals=[(RayDMatrix(X, label=y), "valid")],
        ray_params=RayParams(num_actors=2),
        callbacks=[TuneReportCheckpointCallback(
            filename="model.ubj", frequency=3, results_dir=str(tmp_path)
        )],
    )
    # iterations 0..6: due after iters 2 and 5 ((it+1)%3==0), final at 6
    dirs = sorted(p.name for p in tmp_path.iterdir())
    assert dirs == ["checkpoint_000002", "checkpoint_000005",
                    "checkpoint_000006"]
    for d in dirs:
        path = tmp_path / d / "model.ubj"
        assert path.exists()
        bst = load_model(str(path))
        preds = bst.predict(X)
        assert preds.shape == (600,)
    # checkpoint at iter 2 has 3 rounds, the final one 7
    assert load_model(
        str(tmp_path / "checkpoint_000002" / "model.ubj")
    ).num_boosted_rounds() == 3
    assert load_model(
        str(tmp_path / "checkpoint_000006" / "model.ubj")
    ).num_boosted_rounds() == 7


def test_tune_frequency_zero_reports_only(tmp_path):
    X, y = create_data(400, 4)
    train(
        {"objective": "binary:logistic", "eval_metric": ["logloss"]},
        RayDMatrix(X, label=y), 3,
        evals=[(RayDMatrix(X, label=y), "valid")],
        ray_params=RayParams(num_actors=1),
        callbacks=[TuneReportCheckpointCallback(
            frequency=0, results_dir=str(tmp_path)
        )],
    )
    # final after_training checkpoint still lands (upstream semantics)
    dirs = sorted(p.name for p in tmp_path.iterdir())
    assert dirs == ["checkpoint_000002"]
