"""Distributed end-to-end correctness (mirrors reference test_end_to_end.py).

The centerpiece is the joint-training test: a dataset whose BATCH-sharded
halves are individually unlearnable, so 100% accuracy proves the
cross-actor histogram allreduce is correct (reference
test_end_to_end.py:56-211)."""

import numpy as np
import pytest

from tests.utils import create_data, one_hot_impossible_halves
from xgboost_ray_amd import (
    RayDMatrix,
    RayParams,
    RayShardingMode,
    predict,
    train,
)
from xgboost_ray_amd.actor import TrainingError
from xgboost_ray_amd.callback import DistributedCallback


class PutCallback:
    def after_iteration(self, booster, iteration, evals_log):
        from xgboost_ray_amd.session import get_actor_rank, put_queue

        put_queue(("it", iteration, get_actor_rank()))
        return False


class BoomCallback:
    def after_iteration(self, booster, iteration, evals_log):
        raise RuntimeError("boom-from-actor")


class TestJointTraining:
    def test_disjoint_halves_learn_whole(self):
        X, y = one_hot_impossible_halves(repeat=32)
        dtrain = RayDMatrix(
            X, label=y, sharding=RayShardingMode.BATCH
        )
        bst = train(
            {"objective": "multi:softmax", "num_class": 4, "max_depth": 3,
             "eta": 1.0, "reg_lambda": 0.0, "min_child_weight": 0.0},
            dtrain, 10, ray_params=RayParams(num_actors=2),
        )
        pred = bst.predict(X)
        acc = (pred.astype(np.int64) == y.astype(np.int64)).mean()
        assert acc == 1.0, f"allreduce broken: accuracy {acc} < 1.0"

    def test_distributed_equals_single(self):
        """2-actor training must equal 1-actor training bitwise (int64
        histograms are world-size invariant)."""
        X, y = create_data(4096, 6)
        params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3}
        preds = []
        for num_actors in (1, 2):
            dtrain = RayDMatrix(X, label=y)
            bst = train(
                params, dtrain, 8, ray_params=RayParams(num_actors=num_actors)
            )
            preds.append(bst.predict(X, output_margin=True))
        np.testing.assert_allclose(preds[0], preds[1], rtol=1e-5, atol=1e-6)

    def test_lossguide_distributed_equals_single(self):
        """Leaf-wise growth is heap-ordered by globally identical gains,
        so 2-actor == 1-actor bitwise."""
        X, y = create_data(4096, 6)
        params = {"objective": "binary:logistic", "grow_policy": "lossguide",
                  "max_leaves": 24, "max_depth": 0, "eta": 0.3}
        preds = []
        for num_actors in (1, 2):
            bst = train(
                params, RayDMatrix(X, label=y), 5,
                ray_params=RayParams(num_actors=num_actors),
            )
            preds.append(bst.predict(X, output_margin=True))
        np.testing.assert_allclose(preds[0], preds[1], rtol=1e-6, atol=1e-7)

    def test_overlapped_allreduce_path(self, monkeypatch):
        """The chunked build + async-AllReduce pipeline (the multi-GPU
        hot path) must match the plain path exactly - forced on under
        gloo via RXGB_FORCE_OVERLAP_ALLREDUCE."""
        monkeypatch.setenv("RXGB_FORCE_OVERLAP_ALLREDUCE", "1")
        X, y = create_data(3000, 40)
        params = {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3}
        dtrain = RayDMatrix(X, label=y)
        bst_overlap = train(
            params, dtrain, 6, ray_params=RayParams(num_actors=2)
        )
        monkeypatch.delenv("RXGB_FORCE_OVERLAP_ALLREDUCE")
        bst_plain = train(
            params, RayDMatrix(X, label=y), 6,
            ray_params=RayParams(num_actors=2),
        )
        np.testing.assert_array_equal(
            bst_overlap.predict(X, output_margin=True),
            bst_plain.predict(X, output_margin=True),
        )


class TestPredictions:
    def test_softprob_2d_combine(self):
        X, y = create_data(1200, 6, kind="multi")
        dtrain = RayDMatrix(X, label=y)
        bst = train(
            {"objective": "multi:softprob", "num_class": 4, "max_depth": 4,
             "eta": 0.5},
            dtrain, 8, ray_params=RayParams(num_actors=2),
        )
        pred = predict(bst, RayDMatrix(X), ray_params=RayParams(num_actors=2))
        assert pred.shape == (1200, 4)
        np.testing.assert_allclose(pred.sum(axis=1), 1.0, rtol=1e-4)
        local = bst.predict(X)
        np.testing.assert_allclose(pred, local, rtol=1e-5, atol=1e-6)

    def test_distributed_predict_matches_local(self):
        X, y = create_data(2000, 5)
        dtrain = RayDMatrix(X, label=y)
        bst = train(
            {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3},
            dtrain, 5, ray_params=RayParams(num_actors=2),
        )
        dist_pred = predict(
            bst, RayDMatrix(X), ray_params=RayParams(num_actors=2)
        )
        np.testing.assert_allclose(
            dist_pred, bst.predict(X), rtol=1e-5, atol=1e-6
        )


class TestEvalsAndResults:
    def test_evals_and_additional_results(self):
        X, y = create_data(2000, 5)
        Xv, yv = create_data(500, 5, seed=5)
        dtrain = RayDMatrix(X, label=y)
        dval = RayDMatrix(Xv, label=yv)
        res, add = {}, {}
        train(
            {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
             "eval_metric": ["logloss", "auc"]},
            dtrain, 6, evals=[(dtrain, "train"), (dval, "valid")],
            evals_result=res, additional_results=add,
            ray_params=RayParams(num_actors=2),
        )
        assert set(res.keys()) == {"train", "valid"}
        assert len(res["valid"]["logloss"]) == 6
        assert add["total_n"] == 2000
        assert add["training_time_s"] > 0
        assert add["total_time_s"] >= add["training_time_s"]

    def test_callback_returns_via_put_queue(self):
        X, y = create_data(600, 4)
        dtrain = RayDMatrix(X, label=y)
        add = {}
        train(
            {"objective": "binary:logistic", "max_depth": 3},
            dtrain, 4, ray_params=RayParams(num_actors=2),
            additional_results=add, callbacks=[PutCallback()],
        )
        cr = add["callback_returns"]
        assert sorted(cr.keys()) == [0, 1]
        assert len(cr[0]) == 4
        assert cr[0][0][0] == "it"


class TestValidation:
    def test_wrong_dtrain_type(self):
        with pytest.raises(ValueError, match="RayDMatrix"):
            train({}, np.zeros((10, 2)), ray_params=RayParams(num_actors=2))

    def test_invalid_ray_params(self):
        X, y = create_data(100, 3)
        with pytest.raises(ValueError, match="num_actors"):
            train({}, RayDMatrix(X, label=y), ray_params=RayParams())

    def test_exact_tree_method_rejected(self):
        X, y = create_data(100, 3)
        with pytest.raises(ValueError, match="tree_method"):
            train(
                {"tree_method": "exact"}, RayDMatrix(X, label=y),
                ray_params=RayParams(num_actors=2),
            )

    def test_error_propagates_from_actor(self):
        """Exceptions inside actors surface on the driver
        (reference test_end_to_end.py:321-353)."""
        X, y = create_data(200, 3)

        with pytest.raises(RuntimeError):
            train(
                {"objective": "binary:logistic"},
                RayDMatrix(X, label=y), 3,
                ray_params=RayParams(num_actors=2, max_actor_restarts=0),
                callbacks=[BoomCallback()],
            )


class LogCallback(DistributedCallback):
    def __init__(self, log_dir):
        self.log_dir = log_dir

    def _log(self, actor, event):
        with open(f"{self.log_dir}/rank_{actor.rank}.log", "a") as f:
            f.write(event + "\n")

    def on_init(self, actor, *a, **kw):
        self._log(actor, "init")

    def before_data_loading(self, actor, data, *a, **kw):
        self._log(actor, "before_load")

    def after_data_loading(self, actor, data, *a, **kw):
        self._log(actor, "after_load")

    def before_train(self, actor, *a, **kw):
        self._log(actor, "before_train")

    def after_train(self, actor, result_dict, *a, **kw):
        self._log(actor, "after_train")


class TestDistributedCallbacks:
    def test_callback_order(self, tmp_path):
        """Lifecycle hooks fire in order on every actor
        (reference test_end_to_end.py:279-305)."""
        X, y = create_data(400, 4)
        train(
            {"objective": "binary:logistic"},
            RayDMatrix(X, label=y), 2,
            ray_params=RayParams(
                num_actors=2,
                distributed_callbacks=[LogCallback(str(tmp_path))],
            ),
        )
        for rank in range(2):
            with open(f"{tmp_path}/rank_{rank}.log") as f:
                events = f.read().splitlines()
            assert events == [
                "init", "before_load", "before_load", "after_load",
                "after_load", "before_train", "after_train",
            ] or events == [
                "init", "before_load", "after_load",
                "before_train", "after_train",
            ]


def test_predict_modes_distributed():
    """pred_leaf / pred_contribs through the 2-actor predict path."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, predict, train

    rng = np.random.RandomState(0)
    X = rng.rand(2000, 5).astype(np.float32)
    y = (X[:, 0] + X[:, 1] > 1).astype(np.float32)
    bst = train(
        {"objective": "binary:logistic", "max_depth": 3},
        RayDMatrix(X, y),
        num_boost_round=3,
        ray_params=RayParams(num_actors=2),
    )
    leaves = predict(
        bst, RayDMatrix(X), ray_params=RayParams(num_actors=2),
        pred_leaf=True,
    )
    assert leaves.shape == (2000, 3)
    for ti, t in enumerate(bst.trees):
        assert (t.feat[leaves[:, ti].astype(int)] < 0).all()
    contribs = predict(
        bst, RayDMatrix(X), ray_params=RayParams(num_actors=2),
        pred_contribs=True,
    )
    margin = predict(
        bst, RayDMatrix(X), ray_params=RayParams(num_actors=2),
        output_margin=True,
    )
    assert np.abs(contribs.sum(1) - margin).max() < 1e-4


def test_boost_from_prediction():
    """base_margin continuation == continued training (reference
    test_xgboost_api.py test_boost_from_prediction_hist semantics)."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, predict, train

    rng = np.random.RandomState(0)
    X = rng.rand(4000, 6).astype(np.float32)
    y = (X[:, 0] + X[:, 1] * 2 > 1.2).astype(np.float32)
    rp = RayParams(num_actors=2)
    params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3}

    bst_a = train(params, RayDMatrix(X, y), num_boost_round=4, ray_params=rp)
    margin_a = predict(
        bst_a, RayDMatrix(X), ray_params=rp, output_margin=True
    )
    bst_b = train(
        params, RayDMatrix(X, y, base_margin=margin_a),
        num_boost_round=4, ray_params=rp,
    )
    margin_b = predict(
        bst_b, RayDMatrix(X, base_margin=margin_a), ray_params=rp,
        output_margin=True,
    )

    bst_full = train(
        params, RayDMatrix(X, y), num_boost_round=8, ray_params=rp
    )
    margin_full = predict(
        bst_full, RayDMatrix(X), ray_params=rp, output_margin=True
    )
    # not bitwise (per-tree quantization scales differ) but numerically
    # the same boosting trajectory
    assert np.allclose(margin_b, margin_full, atol=1e-3)


def test_eval_set_weights():
    """Weighted eval metrics flow through the distributed eval path."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, train

    rng = np.random.RandomState(1)
    X = rng.rand(3000, 5).astype(np.float32)
    y = (X[:, 0] > 0.5).astype(np.float32)
    w = np.where(y > 0, 2.0, 1.0).astype(np.float32)
    res = {}
    train(
        {"objective": "binary:logistic", "max_depth": 3,
         "eval_metric": ["logloss"]},
        RayDMatrix(X, y),
        num_boost_round=3,
        ray_params=RayParams(num_actors=2),
        evals=[(RayDMatrix(X, y, weight=w), "weighted")],
        evals_result=res,
    )
    assert "weighted" in res and "logloss" in res["weighted"]
    vals = res["weighted"]["logloss"]
    assert len(vals) == 3 and vals[-1] < vals[0]


def test_gblinear_distributed():
    """booster=gblinear through 2 actors == single-actor training."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, predict, train

    rng = np.random.RandomState(0)
    X = rng.randn(6000, 5).astype(np.float32)
    w_true = np.array([1.0, -2.0, 0.5, 0.0, 3.0], np.float32)
    y = (X @ w_true + 0.3).astype(np.float32)
    params = {"objective": "reg:squarederror", "booster": "gblinear",
              "eta": 0.5, "lambda": 0.0, "base_score": 0.0}
    bst2 = train(params, RayDMatrix(X, y), num_boost_round=30,
                 ray_params=RayParams(num_actors=2))
    w2 = bst2.linear_weights[:, 0]
    assert np.abs(w2[:5] - w_true).max() < 0.05
    bst1 = train(params, RayDMatrix(X, y), num_boost_round=30,
                 ray_params=RayParams(num_actors=1))
    assert np.allclose(bst1.linear_weights, bst2.linear_weights, atol=1e-4)
    pred = predict(bst2, RayDMatrix(X), ray_params=RayParams(num_actors=2))
    assert np.abs(pred - (X @ w2[:5] + w2[5])).max() < 1e-3


def test_dart_distributed_equals_single():
    """booster=dart: dropout RNG is (seed, iteration)-keyed, so the
    distributed model equals the single-actor model bitwise."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, train

    rng = np.random.RandomState(0)
    X = rng.rand(6000, 5).astype(np.float32)
    y = (X[:, 0] + X[:, 1] + 0.1 * rng.randn(6000)).astype(np.float32)
    params = {"objective": "reg:squarederror", "booster": "dart",
              "rate_drop": 0.3, "max_depth": 4, "eta": 0.3, "seed": 7}
    b1 = train(params, RayDMatrix(X, y), num_boost_round=10,
               ray_params=RayParams(num_actors=1))
    b2 = train(params, RayDMatrix(X, y), num_boost_round=10,
               ray_params=RayParams(num_actors=2))
    for ta, tb in zip(b1.trees, b2.trees):
        assert np.array_equal(ta.thr, tb.thr)
        assert np.array_equal(ta.value, tb.value)


def _feval_neg_err(margin, dmat):
    import numpy as np

    label = np.asarray(dmat.get_label())
    pred = (1.0 / (1.0 + np.exp(-np.asarray(margin)))) > 0.5
    return "negerr", -float((pred != (label > 0.5)).mean())


def test_custom_feval_maximize_early_stopping():
    """Custom feval + explicit maximize drive early stopping
    (xgboost.train(maximize=...) parity)."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, train

    rng = np.random.RandomState(0)
    X = rng.rand(3000, 5).astype(np.float32)
    y = (X[:, 0] + X[:, 1] > 1).astype(np.float32)
    res = {}
    bst = train(
        {"objective": "binary:logistic", "max_depth": 3,
         "disable_default_eval_metric": True},
        RayDMatrix(X, y),
        num_boost_round=30,
        ray_params=RayParams(num_actors=2),
        evals=[(RayDMatrix(X, y), "train")],
        feval=_feval_neg_err,
        maximize=True,
        early_stopping_rounds=5,
        evals_result=res,
    )
    assert "negerr" in res["train"]
    assert bst.best_iteration is not None


@pytest.mark.parametrize(
    "extra",
    [
        {"colsample_bytree": 0.7, "colsample_bylevel": 0.8, "seed": 11},
        {"grow_policy": "lossguide", "max_leaves": 24, "max_depth": 0},
        {"booster": "dart", "rate_drop": 0.25, "seed": 4},
        {"num_parallel_tree": 3, "eta": 0.2},
        {"monotone_constraints": "(1,0,-1,0,0)", "seed": 2},
        {"interaction_constraints": [[0, 1], [2, 3, 4]], "max_depth": 5},
        {"booster": "dart", "rate_drop": 0.3, "colsample_bytree": 0.7,
         "seed": 9},
        {"_actors": 3, "colsample_bylevel": 0.6, "max_depth": 5,
         "seed": 13},
        {"colsample_bynode": 0.6, "max_depth": 5, "seed": 17},
        {"booster": "dart", "rate_drop": 0.3, "sample_type": "weighted",
         "seed": 21},
    ],
)
def test_distributed_equals_single_matrix(extra):
    """Randomized differential: 2-actor training grows bitwise-identical
    trees to 1-actor training across parameter variants (the invariant
    every rank-divergence bug breaks). Row subsampling is excluded: each
    rank samples its own shard (reference/xgboost behavior), so the
    realized row sets - and therefore the trees - legitimately differ
    between world sizes while remaining per-configuration deterministic
    (see test_subsample_deterministic_per_world_size)."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, train

    rng = np.random.RandomState(0)
    X = rng.rand(4000, 5).astype(np.float32)
    y = (X[:, 0] + X[:, 2] + 0.3 * rng.randn(4000) > 1).astype(np.float32)
    params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3}
    params.update(extra)
    n_actors = params.pop("_actors", 2)
    b1 = train(dict(params), RayDMatrix(X, y), num_boost_round=6,
               ray_params=RayParams(num_actors=1))
    b2 = train(dict(params), RayDMatrix(X, y), num_boost_round=6,
               ray_params=RayParams(num_actors=n_actors))
    assert len(b1.trees) == len(b2.trees)
    for ta, tb in zip(b1.trees, b2.trees):
        assert np.array_equal(ta.feat, tb.feat)
        assert np.array_equal(ta.thr, tb.thr)
        assert np.array_equal(ta.value, tb.value)


def test_subsample_deterministic_per_world_size():
    """Subsampled training is reproducible run-to-run at a fixed world
    size (the checkpoint-resume contract), even though different world
    sizes realize different row samples."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, train

    rng = np.random.RandomState(0)
    X = rng.rand(4000, 5).astype(np.float32)
    y = (X[:, 0] + X[:, 2] + 0.3 * rng.randn(4000) > 1).astype(np.float32)
    params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
              "subsample": 0.8, "seed": 11}
    runs = [
        train(dict(params), RayDMatrix(X, y), num_boost_round=5,
              ray_params=RayParams(num_actors=2))
        for _ in range(2)
    ]
    for ta, tb in zip(runs[0].trees, runs[1].trees):
        assert np.array_equal(ta.feat, tb.feat)
        assert np.array_equal(ta.thr, tb.thr)
        assert np.array_equal(ta.value, tb.value)


def test_ranking_distributed_equals_single():
    """Group-aware qid sharding: whole query groups per actor makes
    distributed ranking bitwise-equal to single-actor training (the
    reference fragments groups across row shards), and qid-matrix
    predictions recombine through the recorded group indices."""
    import numpy as np

    from xgboost_ray_amd import RayDMatrix, RayParams, predict, train

    rng = np.random.RandomState(0)
    n = 4000
    qid = np.repeat(np.arange(n // 20), 20).astype(np.int64)
    X = rng.rand(n, 5).astype(np.float32)
    rel = np.clip(
        (X[:, 0] * 3 + 0.3 * rng.randn(n)).round(), 0, 3
    ).astype(np.float32)
    params = {"objective": "rank:ndcg", "max_depth": 4, "eta": 0.3}
    b1 = train(dict(params), RayDMatrix(X, rel, qid=qid),
               num_boost_round=4, ray_params=RayParams(num_actors=1))
    b2 = train(dict(params), RayDMatrix(X, rel, qid=qid),
               num_boost_round=4, ray_params=RayParams(num_actors=2))
    for a, b in zip(b1.trees, b2.trees):
        assert np.array_equal(a.feat, b.feat)
        assert np.array_equal(a.thr, b.thr)
        assert np.array_equal(a.value, b.value)
    # prediction on a qid matrix reassembles in original row order
    p2 = predict(b2, RayDMatrix(X, qid=qid),
                 ray_params=RayParams(num_actors=2))
    p1 = predict(b1, RayDMatrix(X), ray_params=RayParams(num_actors=1))
    assert np.array_equal(p1, p2)


def test_staged_work_writes_back():
    """_StagedWork.wait() must complete the host reduce then write the
    result into the original tensor (gloo transport w/ GPU compute)."""
    import torch

    from xgboost_ray_amd.engine.collective import _StagedWork

    dest = torch.zeros(4)
    host = torch.arange(4, dtype=torch.float32)
    w = _StagedWork(None, host, dest)
    assert w.wait() is True
    assert torch.equal(dest, host)


class _BurstQueueCallback:
    """Floods the queue, including a burst on the FINAL round right
    before the actor returns - the race window where the driver's train
    futures can resolve before the mp.Queue feeder delivers."""

    def __init__(self, rounds, per_round=20):
        self.rounds = rounds
        self.per_round = per_round

    def after_iteration(self, booster, iteration, evals_log):
        from xgboost_ray_amd.session import get_actor_rank, put_queue

        r = get_actor_rank()
        for i in range(self.per_round):
            put_queue(("item", r, iteration, i))
        return False


def test_queue_drain_no_lost_items():
    """Every queue item must reach callback_returns, including the final
    round's burst (settled post-success drain; VERDICT race-guard item)."""
    from tests.utils import create_data
    from xgboost_ray_amd import RayDMatrix, RayParams, train

    X, y = create_data(2000, 4)
    rounds, per_round, actors = 5, 20, 2
    add = {}
    train(
        {"objective": "binary:logistic", "max_depth": 3},
        RayDMatrix(X, label=y), rounds,
        ray_params=RayParams(num_actors=actors),
        additional_results=add,
        callbacks=[_BurstQueueCallback(rounds, per_round)],
    )
    returns = add["callback_returns"]
    for r in range(actors):
        items = [it for it in returns.get(r, []) if it[0] == "item"]
        assert len(items) == rounds * per_round, (r, len(items))
        # per-rank FIFO order preserved
        assert items == sorted(items, key=lambda t: (t[2], t[3]))


def test_invalid_train_kwarg_raises():
    """Unknown train() kwargs must fail fast with a TypeError naming the
    argument (reference testKwargsValidation, test_end_to_end.py:355) -
    not be silently dropped inside the actors."""
    from tests.utils import create_data
    from xgboost_ray_amd import RayDMatrix, RayParams, train

    X, y = create_data(200, 4)
    with pytest.raises(TypeError, match="totally_invalid_kwarg"):
        train(
            {"objective": "binary:logistic"},
            RayDMatrix(X, label=y), 2,
            ray_params=RayParams(num_actors=1),
            totally_invalid_kwarg="",
        )


def test_invalid_predict_kwarg_raises():
    from tests.utils import create_data
    from xgboost_ray_amd import RayDMatrix, RayParams, predict, train

    X, y = create_data(300, 4)
    bst = train({"objective": "binary:logistic"}, RayDMatrix(X, label=y),
                2, ray_params=RayParams(num_actors=1))
    with pytest.raises(TypeError, match="bogus_kwarg"):
        predict(bst, RayDMatrix(X), ray_params=RayParams(num_actors=1),
                bogus_kwarg=1)


def test_tune_resources_requires_ray():
    from xgboost_ray_amd.main import RayParams

    with pytest.raises(RuntimeError, match="ray"):
        RayParams(num_actors=2).get_tune_resources()
