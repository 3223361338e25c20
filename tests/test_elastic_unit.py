"""Elastic scheduler unit tests with fake actors - no processes involved
(reference test_fault_tolerance.py:451-585 uses _FakeTask/MagicMock)."""

import time

import pytest

from xgboost_ray_amd import elastic
from xgboost_ray_amd.main import (
    RayParams,
    RayXGBoostActorAvailable,
    _Checkpoint,
    _TrainingState,
)
from xgboost_ray_amd.util import Future


class _FakeActor:
    def __init__(self, rank):
        self.rank = rank
        self.killed = False

    def kill(self):
        self.killed = True


def _state(num_actors=4, alive=(0, 2)):
    actors = [None] * num_actors
    for r in alive:
        actors[r] = _FakeActor(r)
    return _TrainingState(
        actors=actors,
        queue=None,
        stop_event=None,
        checkpoint=_Checkpoint(),
        additional_results={},
        failed_actor_ranks=set(),
    )


class TestUpdateScheduledStates:
    def test_promotes_ready_actor_and_schedules_restart(self, monkeypatch):
        monkeypatch.setenv("RXGB_ELASTIC_RESTART_GRACE_PERIOD_S", "3600")
        state = _state()
        actor = _FakeActor(1)
        fut = Future()
        fut.set_result({"uid": 100})
        state.pending_actors[1] = (actor, [fut])
        elastic._update_scheduled_actor_states(state)
        assert state.actors[1] is actor
        assert 1 not in state.pending_actors
        assert state.restart_training_at is not None

    def test_raises_available_after_grace(self, monkeypatch):
        monkeypatch.setenv("RXGB_ELASTIC_RESTART_GRACE_PERIOD_S", "0")
        state = _state()
        actor = _FakeActor(1)
        fut = Future()
        fut.set_result({})
        state.pending_actors[1] = (actor, [fut])
        with pytest.raises(RayXGBoostActorAvailable):
            elastic._update_scheduled_actor_states(state)
            # second call passes the (zero) grace period
            elastic._update_scheduled_actor_states(state)

    def test_failed_staging_dropped(self):
        state = _state()
        actor = _FakeActor(3)
        fut = Future()
        fut.set_error(RuntimeError("staging failed"))
        state.pending_actors[3] = (actor, [fut])
        elastic._update_scheduled_actor_states(state)
        assert 3 not in state.pending_actors
        assert state.actors[3] is None
        assert actor.killed

    def test_pending_not_ready_untouched(self):
        state = _state()
        actor = _FakeActor(1)
        fut = Future()  # not done
        state.pending_actors[1] = (actor, [fut])
        elastic._update_scheduled_actor_states(state)
        assert 1 in state.pending_actors
        assert state.actors[1] is None


class TestMaybeSchedule:
    def test_respects_resource_check_interval(self, monkeypatch):
        monkeypatch.setenv("RXGB_ELASTIC_RESTART_RESOURCE_CHECK_S", "3600")
        state = _state()
        state.last_resource_check_at = time.time()
        scheduled = elastic._maybe_schedule_new_actors(
            state, RayParams(num_actors=4, elastic_training=True,
                             max_failed_actors=2),
            use_gpu=False, load_data=[],
        )
        assert scheduled is False
        assert not state.pending_actors
