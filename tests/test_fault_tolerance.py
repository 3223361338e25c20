"""Fault tolerance (mirrors reference test_fault_tolerance.py).

Key assertions: tree count survives failure+restart, total_n reflects the
participating world, and - the strongest - checkpoint-resume after a
SIGKILL yields the IDENTICAL model a failure-free run produces
(reference testSameResultWithAndWithoutError, :401-449)."""

import numpy as np
import pytest

from tests.fault_tolerance import DieCallback
from tests.utils import create_data
from xgboost_ray_amd import RayDMatrix, RayParams, train
from xgboost_ray_amd.env import ENV


PARAMS = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3}


def _data():
    return create_data(2400, 5)


class TestNonElastic:
    def test_restart_completes_all_rounds(self, tmp_path):
        X, y = _data()
        dtrain = RayDMatrix(X, label=y)
        add = {}
        bst = train(
            PARAMS, dtrain, 20,
            ray_params=RayParams(
                num_actors=2, max_actor_restarts=1, checkpoint_frequency=5
            ),
            additional_results=add,
            callbacks=[DieCallback(die_round=9, die_rank=1,
                                   lock_dir=str(tmp_path))],
        )
        assert bst.num_boosted_rounds() == 20
        assert add["total_n"] == 2400  # full world after restart

    def test_same_result_with_and_without_error(self, tmp_path):
        """Determinism: fail+checkpoint-resume == no failure."""
        X, y = _data()
        bst_noerr = train(
            PARAMS, RayDMatrix(X, label=y), 20,
            ray_params=RayParams(num_actors=2, max_actor_restarts=0,
                                 checkpoint_frequency=5),
        )
        bst_err = train(
            PARAMS, RayDMatrix(X, label=y), 20,
            ray_params=RayParams(num_actors=2, max_actor_restarts=1,
                                 checkpoint_frequency=5),
            callbacks=[DieCallback(die_round=11, die_rank=1,
                                   lock_dir=str(tmp_path))],
        )
        assert bst_err.num_boosted_rounds() == 20
        np.testing.assert_array_equal(
            bst_noerr.predict(X, output_margin=True),
            bst_err.predict(X, output_margin=True),
        )

    def test_no_restarts_raises(self, tmp_path):
        X, y = _data()
        with pytest.raises(RuntimeError, match="max_actor_restarts"):
            train(
                PARAMS, RayDMatrix(X, label=y), 20,
                ray_params=RayParams(num_actors=2, max_actor_restarts=0),
                callbacks=[DieCallback(die_round=5, die_rank=1,
                                       lock_dir=str(tmp_path))],
            )


class TestElastic:
    def test_continue_with_fewer_actors(self, tmp_path, monkeypatch):
        monkeypatch.setenv("RXGB_ELASTIC_RESTART_DISABLED", "1")
        X, y = _data()
        add = {}
        bst = train(
            PARAMS, RayDMatrix(X, label=y), 20,
            ray_params=RayParams(
                num_actors=2, elastic_training=True, max_failed_actors=1,
                max_actor_restarts=1, checkpoint_frequency=5,
            ),
            additional_results=add,
            callbacks=[DieCallback(die_round=9, die_rank=1,
                                   lock_dir=str(tmp_path))],
        )
        assert bst.num_boosted_rounds() == 20
        # after the failure only one actor (half the rows) continued
        assert add["total_n"] == 1200

    def test_elastic_comeback(self, tmp_path, monkeypatch):
        """Dead actor is restored mid-training and training restarts
        with the full world (reference elastic_comeback condition)."""
        monkeypatch.setenv("RXGB_ELASTIC_RESTART_RESOURCE_CHECK_S", "1")
        monkeypatch.setenv("RXGB_ELASTIC_RESTART_GRACE_PERIOD_S", "1")
        from tests.fault_tolerance import SlowCallback

        X, y = _data()
        add = {}
        bst = train(
            {**PARAMS, "eta": 0.1}, RayDMatrix(X, label=y), 150,
            ray_params=RayParams(
                num_actors=2, elastic_training=True, max_failed_actors=1,
                max_actor_restarts=1, checkpoint_frequency=2,
            ),
            additional_results=add,
            callbacks=[DieCallback(die_round=10, die_rank=1,
                                   lock_dir=str(tmp_path)),
                       SlowCallback(0.1)],
        )
        assert bst.num_boosted_rounds() == 150
        # the comeback restored the full world before the end
        assert add["total_n"] == 2400

    def test_elastic_needs_max_failed_actors(self):
        X, y = _data()
        with pytest.raises(ValueError, match="max_failed_actors"):
            train(
                PARAMS, RayDMatrix(X, label=y), 20,
                ray_params=RayParams(
                    num_actors=2, elastic_training=True, max_failed_actors=0,
                    max_actor_restarts=1,
                ),
            )


class TestRestartDeterminismVariants:
    """fail+checkpoint-resume == no-failure for the round-2 feature
    combinations whose state must survive the checkpoint (dart weight
    ledger, per-node column sampling, row sampling margin cache)."""

    import pytest as _pytest

    @_pytest.mark.parametrize("extra", [
        {"booster": "dart", "rate_drop": 0.3, "sample_type": "weighted"},
        {"colsample_bynode": 0.6},
        {"subsample": 0.7},
    ])
    def test_variant(self, tmp_path, extra):
        X, y = _data()
        params = dict(PARAMS, **extra, seed=9)
        kw = dict(checkpoint_frequency=5)
        bst_noerr = train(
            params, RayDMatrix(X, label=y), 16,
            ray_params=RayParams(num_actors=2, max_actor_restarts=0, **kw),
        )
        bst_err = train(
            params, RayDMatrix(X, label=y), 16,
            ray_params=RayParams(num_actors=2, max_actor_restarts=1, **kw),
            callbacks=[DieCallback(die_round=9, die_rank=1,
                                   lock_dir=str(tmp_path))],
        )
        assert bst_err.num_boosted_rounds() == 16
        a = bst_noerr.predict(X, output_margin=True)
        b = bst_err.predict(X, output_margin=True)
        if extra.get("booster") == "dart":
            np.testing.assert_allclose(a, b, atol=2e-6, rtol=1e-6)
        else:
            np.testing.assert_array_equal(a, b)
