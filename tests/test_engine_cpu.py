"""Single-rank engine correctness on CPU (numerics oracle for the GPU)."""

import numpy as np
import pytest
import torch

from tests.utils import create_data
from xgboost_ray_amd import ops
from xgboost_ray_amd.engine.quantile import BinnedMatrix, build_cuts
from xgboost_ray_amd.engine.trainer import (
    BoostingEngine,
    EvalPack,
    run_training,
)


def _binned(n=4000, f=6, kind="binary", seed=0, max_bin=64):
    X, y = create_data(n, f, seed, kind)
    return (
        BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=max_bin
        ),
        X,
        y,
    )


class TestBinning:
    def test_bins_reconstruct_order(self):
        X, _ = create_data(500, 3)
        Xt = torch.from_numpy(X)
        cuts = build_cuts(Xt, 32)
        bins = ops.bin_matrix(Xt, cuts.cuts_flat, cuts.cut_ptr)
        # binning must be monotone in the value
        for f in range(3):
            order = np.argsort(X[:, f], kind="stable")
            b = bins[:, f].numpy()[order]
            assert (np.diff(b.astype(np.int32)) >= 0).all()

    def test_missing_values(self):
        X = np.array([[0.0], [np.nan], [1.0]], dtype=np.float32)
        Xt = torch.from_numpy(X)
        cuts = build_cuts(Xt, 8)
        bins = ops.bin_matrix(Xt, cuts.cuts_flat, cuts.cut_ptr)
        assert bins[1, 0] == ops.MISSING_BIN
        assert bins[0, 0] != ops.MISSING_BIN

    def test_few_distinct_values_one_bin_each(self):
        X = np.array([[0.0], [1.0], [2.0], [1.0], [0.0]], dtype=np.float32)
        Xt = torch.from_numpy(X)
        cuts = build_cuts(Xt, 256)
        bins = ops.bin_matrix(Xt, cuts.cuts_flat, cuts.cut_ptr).numpy().flatten()
        assert bins.tolist() == [0, 1, 2, 1, 0]


class TestHistogram:
    def test_histogram_totals_match_sums(self):
        dm, X, y = _binned()
        n = dm.n_rows
        gq = ops.quantize_gpair(
            torch.stack(
                [torch.from_numpy(y) - 0.5, torch.ones(n)], dim=1
            ),
            2.0**20,
            2.0**20,
        )
        ridx = torch.arange(n, dtype=torch.int32)
        hist = ops.build_histogram(
            dm.bins, gq, ridx, torch.tensor([0]), torch.tensor([n]), dm.cuts.max_bins
        )
        # per feature: sum over bins == total (no missing values here)
        tot_g = int(gq[:, 0].sum())
        for f in range(dm.n_features):
            assert int(hist[0, f, :, 0].sum()) == tot_g

    def test_histogram_subtraction_consistency(self):
        dm, X, y = _binned()
        n = dm.n_rows
        gq = ops.quantize_gpair(
            torch.stack([torch.from_numpy(y) - 0.5, torch.ones(n)], dim=1),
            2.0**20,
            2.0**20,
        )
        ridx = torch.arange(n, dtype=torch.int32)
        whole = ops.build_histogram(
            dm.bins, gq, ridx, torch.tensor([0]), torch.tensor([n]), dm.cuts.max_bins
        )
        half = ops.build_histogram(
            dm.bins,
            gq,
            ridx,
            torch.tensor([0, n // 2]),
            torch.tensor([n // 2, n - n // 2]),
            dm.cuts.max_bins,
        )
        torch.testing.assert_close(whole[0], half[0] + half[1])


class TestTraining:
    def test_loss_decreases_binary(self):
        dm, X, y = _binned()
        res = {}
        run_training(
            {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
             "eval_metric": ["logloss"]},
            dm, 15, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        ll = res["train"]["logloss"]
        assert ll[-1] < ll[0] * 0.6

    def test_regression(self):
        dm, X, y = _binned(kind="reg")
        res = {}
        run_training(
            {"objective": "reg:squarederror", "max_depth": 4, "eta": 0.3,
             "eval_metric": ["rmse"]},
            dm, 15, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        assert res["train"]["rmse"][-1] < res["train"]["rmse"][0] * 0.5

    def test_multiclass_softprob(self):
        dm, X, y = _binned(kind="multi")
        res = {}
        bst = run_training(
            {"objective": "multi:softprob", "num_class": 4, "max_depth": 4,
             "eta": 0.5, "eval_metric": ["mlogloss", "merror"]},
            dm, 10, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        assert res["train"]["merror"][-1] < 0.05
        pred = bst.predict(X[:100])
        assert pred.shape == (100, 4)
        np.testing.assert_allclose(pred.sum(axis=1), 1.0, rtol=1e-4)

    def test_predict_matches_train_margins(self):
        """The float tree-walk predictor must agree with the binned
        training-time margins (threshold <-> bin equivalence)."""
        dm, X, y = _binned(max_bin=64)
        engine = BoostingEngine(
            {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3}, dm
        )
        for _ in range(5):
            engine.update()
        pred_margin = engine.booster.predict(X, output_margin=True)
        np.testing.assert_allclose(
            pred_margin, engine.margin.numpy(), rtol=1e-5, atol=1e-5
        )

    def test_determinism_two_runs(self):
        dm, X, y = _binned()
        outs = []
        for _ in range(2):
            bst = run_training(
                {"objective": "binary:logistic", "max_depth": 6, "eta": 0.3},
                dm, 8,
            )
            outs.append(bst.predict(X, output_margin=True))
        np.testing.assert_array_equal(outs[0], outs[1])

    def test_resume_equals_continuous(self):
        """xgb_model resume must produce the identical model
        (determinism contract, reference test_fault_tolerance.py:401-449)."""
        dm, X, y = _binned()
        bst_full = run_training(
            {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3},
            dm, 12,
        )
        bst_half = run_training(
            {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3},
            dm, 6,
        )
        bst_resumed = run_training(
            {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3},
            dm, 6, xgb_model=bst_half,
        )
        np.testing.assert_array_equal(
            bst_full.predict(X, output_margin=True),
            bst_resumed.predict(X, output_margin=True),
        )

    def test_subsample_colsample(self):
        dm, X, y = _binned()
        res = {}
        run_training(
            {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
             "subsample": 0.7, "colsample_bytree": 0.8,
             "eval_metric": ["logloss"]},
            dm, 10, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        assert res["train"]["logloss"][-1] < res["train"]["logloss"][0]

    def test_colsample_bylevel_and_feature_weights(self):
        dm, X, y = _binned()
        dm.feature_weights = torch.tensor([5.0, 5.0, 1.0, 1.0, 1.0, 1.0])
        res = {}
        run_training(
            {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
             "colsample_bylevel": 0.5, "colsample_bytree": 0.8,
             "eval_metric": ["logloss"]},
            dm, 10, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        assert res["train"]["logloss"][-1] < res["train"]["logloss"][0]

    def test_scale_pos_weight(self):
        dm, X, y = _binned()
        res1, res2 = {}, {}
        for spw, res in ((1.0, res1), (5.0, res2)):
            run_training(
                {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
                 "scale_pos_weight": spw, "eval_metric": ["auc"]},
                dm, 5, evals=[EvalPack(name="train", X=None)],
                evals_result=res,
            )
        # different weighting must change the model
        assert res1["train"]["auc"] != res2["train"]["auc"]

    def test_lossguide_growth(self):
        """grow_policy=lossguide: exact leaf budget, deterministic,
        loss-decreasing, predict-consistent."""
        dm, X, y = _binned()
        res = {}
        bst = run_training(
            {"objective": "binary:logistic", "grow_policy": "lossguide",
             "max_leaves": 15, "max_depth": 0, "eta": 0.3,
             "eval_metric": ["logloss"]},
            dm, 10, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        t = bst.trees[0]
        assert int((t.feat < 0).sum()) == 15
        assert res["train"]["logloss"][-1] < res["train"]["logloss"][0] * 0.6
        bst2 = run_training(
            {"objective": "binary:logistic", "grow_policy": "lossguide",
             "max_leaves": 15, "max_depth": 0, "eta": 0.3},
            dm, 10,
        )
        np.testing.assert_array_equal(
            bst.predict(X, output_margin=True),
            bst2.predict(X, output_margin=True),
        )

    def test_early_stopping(self):
        X, y = create_data(2000, 6, 0, "binary")
        Xv, yv = create_data(500, 6, 7, "binary")
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        ev = EvalPack(
            name="valid",
            X=torch.from_numpy(Xv),
            label=torch.from_numpy(yv),
        )
        res = {}
        bst = run_training(
            {"objective": "binary:logistic", "max_depth": 6, "eta": 0.9,
             "eval_metric": ["logloss"]},
            dm, 100, evals=[ev], evals_result=res,
            early_stopping_rounds=5,
        )
        assert bst.best_iteration is not None
        assert len(res["valid"]["logloss"]) < 100

    def test_eval_on_holdout(self):
        X, y = create_data(3000, 6, 0, "binary")
        Xv, yv = create_data(800, 6, 3, "binary")
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        ev = EvalPack(
            name="valid", X=torch.from_numpy(Xv), label=torch.from_numpy(yv)
        )
        res = {}
        run_training(
            {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
             "eval_metric": ["auc"]},
            dm, 10, evals=[ev], evals_result=res,
        )
        assert res["valid"]["auc"][-1] > 0.85

    def test_custom_objective_and_feval(self):
        dm, X, y = _binned()

        def squared_log(preds, dtrain):
            y = dtrain.get_label()
            grad = preds - y
            hess = np.ones_like(preds)
            return grad, hess

        def rmse_feval(preds, dtrain):
            y = dtrain.get_label()
            return "custom-rmse", float(np.sqrt(np.mean((preds - y) ** 2)))

        res = {}
        run_training(
            {"max_depth": 4, "eta": 0.3, "disable_default_eval_metric": 1},
            dm, 8, evals=[EvalPack(name="train", X=None)],
            obj=squared_log, feval=rmse_feval, evals_result=res,
        )
        vals = res["train"]["custom-rmse"]
        assert vals[-1] < vals[0]


class TestRanking:
    def test_rank_ndcg_improves(self):
        from tests.utils import create_labeled_sorted_rank_data

        X, y, qid = create_labeled_sorted_rank_data()
        dm = BinnedMatrix.build(
            torch.from_numpy(X),
            label=torch.from_numpy(y),
            qid=torch.from_numpy(qid),
            max_bin=64,
        )
        res = {}
        run_training(
            {"objective": "rank:ndcg", "max_depth": 4, "eta": 0.3,
             "eval_metric": ["ndcg"]},
            dm, 15, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        ndcg = res["train"]["ndcg"]
        assert ndcg[-1] > ndcg[0]
        assert ndcg[-1] > 0.9

    def test_rank_pairwise_runs(self):
        from tests.utils import create_labeled_sorted_rank_data

        X, y, qid = create_labeled_sorted_rank_data(n_groups=10)
        dm = BinnedMatrix.build(
            torch.from_numpy(X),
            label=torch.from_numpy(y),
            qid=torch.from_numpy(qid),
            max_bin=64,
        )
        res = {}
        run_training(
            {"objective": "rank:pairwise", "max_depth": 3, "eta": 0.3,
             "eval_metric": ["map"]},
            dm, 8, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        assert res["train"]["map"][-1] >= res["train"]["map"][0]


class TestMetrics:
    def test_auc_known_value(self):
        from xgboost_ray_amd.engine.metrics import get_metric

        # perfect separation -> AUC 1
        margin = torch.tensor([-2.0, -1.0, 1.0, 2.0])
        label = torch.tensor([0.0, 0.0, 1.0, 1.0])
        m = get_metric("auc")
        assert m.finalize(m.local_stats(margin, label, None, None, None)) == 1.0
        # random-ish
        margin2 = torch.tensor([1.0, -1.0, 1.0, -1.0])
        v = m.finalize(m.local_stats(margin2, label, None, None, None))
        assert abs(v - 0.5) < 1e-6

    def test_rmse_weighted(self):
        from xgboost_ray_amd.engine.metrics import get_metric

        m = get_metric("rmse")
        margin = torch.tensor([1.0, 3.0])
        label = torch.tensor([0.0, 0.0])
        w = torch.tensor([3.0, 1.0])
        v = m.finalize(m.local_stats(margin, label, w, None, None))
        assert abs(v - np.sqrt((3 * 1 + 9) / 4)) < 1e-9


class TestMonotone:
    def test_monotone_increasing_constraint(self):
        rng = np.random.RandomState(0)
        n = 6000
        X = rng.rand(n, 3).astype(np.float32)
        # f0 has a NON-monotone true effect; the constraint must force a
        # monotone-increasing model response anyway
        y = (np.sin(X[:, 0] * 6) + X[:, 1] + 0.1 * rng.randn(n)).astype(
            np.float32
        )
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        bst = run_training(
            {"objective": "reg:squarederror", "max_depth": 5, "eta": 0.3,
             "monotone_constraints": "(1,0,0)"},
            dm, 20,
        )
        # sweep f0 with the other features fixed
        grid = np.zeros((50, 3), dtype=np.float32)
        grid[:, 0] = np.linspace(0, 1, 50)
        grid[:, 1] = 0.5
        grid[:, 2] = 0.5
        pred = bst.predict(grid, output_margin=True)
        assert (np.diff(pred) >= -1e-6).all(), "response must be monotone"
        # and without the constraint it must NOT be monotone (sanity)
        bst2 = run_training(
            {"objective": "reg:squarederror", "max_depth": 5, "eta": 0.3},
            dm, 20,
        )
        pred2 = bst2.predict(grid, output_margin=True)
        assert not (np.diff(pred2) >= -1e-6).all()

    def test_monotone_decreasing(self):
        rng = np.random.RandomState(1)
        n = 4000
        X = rng.rand(n, 2).astype(np.float32)
        y = (-X[:, 0] + 0.5 * X[:, 1] + 0.05 * rng.randn(n)).astype(np.float32)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        bst = run_training(
            {"objective": "reg:squarederror", "max_depth": 4, "eta": 0.3,
             "monotone_constraints": [-1, 0]},
            dm, 15,
        )
        grid = np.zeros((40, 2), dtype=np.float32)
        grid[:, 0] = np.linspace(0, 1, 40)
        grid[:, 1] = 0.5
        pred = bst.predict(grid, output_margin=True)
        assert (np.diff(pred) <= 1e-6).all()


def _paths_respect_constraints(bst, sets):
    """Every root->node path must satisfy: each split feature shares a
    constraint set with ALL features used above it on the path."""
    sets = [frozenset(s) for s in sets]
    for t in bst.trees:
        stack = [(0, frozenset())]
        while stack:
            nid, path = stack.pop()
            f = int(t.feat[nid])
            if f < 0:
                continue
            if path:
                ok = any(path | {f} <= cs for cs in sets)
                if not ok:
                    return False
            lid = int(t.left[nid])
            stack.append((lid, path | {f}))
            stack.append((lid + 1, path | {f}))
    return True


class TestInteractionConstraints:
    def _data(self, seed=0, n=8000):
        rng = np.random.RandomState(seed)
        X = rng.rand(n, 6).astype(np.float32)
        # strong pairwise interactions across the constraint boundary so
        # an unconstrained tree would freely mix all features
        y = (
            X[:, 0] * X[:, 3]
            + X[:, 1] * X[:, 4]
            + X[:, 2] * X[:, 5]
            + 0.05 * rng.randn(n)
        ).astype(np.float32)
        return X, y

    def test_paths_confined_to_sets(self):
        X, y = self._data()
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        sets = [[0, 1, 2], [3, 4, 5]]
        bst = run_training(
            {"objective": "reg:squarederror", "max_depth": 5, "eta": 0.3,
             "interaction_constraints": sets},
            dm, 10,
        )
        assert _paths_respect_constraints(bst, sets)
        # sanity: without constraints the same data mixes features
        bst2 = run_training(
            {"objective": "reg:squarederror", "max_depth": 5, "eta": 0.3},
            dm, 10,
        )
        assert not _paths_respect_constraints(bst2, sets)

    def test_string_spec_and_lossguide(self):
        X, y = self._data(seed=1)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        sets = [[0, 1], [2, 3], [4, 5]]
        bst = run_training(
            {"objective": "reg:squarederror", "eta": 0.3,
             "grow_policy": "lossguide", "max_leaves": 24, "max_depth": 6,
             "interaction_constraints": "[[0, 1], [2, 3], [4, 5]]"},
            dm, 8,
        )
        assert _paths_respect_constraints(bst, sets)


class TestExtraMetricsAndPredict:
    def test_aucpr_metric(self):
        from xgboost_ray_amd.engine.metrics import get_metric

        rng = np.random.RandomState(0)
        n = 20000
        score = torch.from_numpy(rng.randn(n).astype(np.float32))
        label = torch.from_numpy(
            (score.numpy() + rng.randn(n) > 0).astype(np.float32)
        )
        m = get_metric("aucpr")
        v = m.finalize(m.local_stats(score, label, None, None, None))
        from sklearn.metrics import average_precision_score
        import scipy.special as sp

        ref = average_precision_score(label.numpy(), sp.expit(score.numpy()))
        assert abs(v - ref) < 0.01

    def test_aucpr_in_training(self):
        from xgboost_ray_amd.engine.trainer import EvalPack

        dm, _, _ = _binned(kind="binary")
        res = {}
        bst = run_training(
            {"objective": "binary:logistic", "max_depth": 3,
             "eval_metric": ["aucpr"]},
            dm, 5, evals=[EvalPack(name="train", X=None)],
            evals_result=res,
        )
        assert "aucpr" in res.get("train", {})
        vals = res["train"]["aucpr"]
        assert len(vals) == 5 and 0.0 <= vals[-1] <= 1.0
        assert vals[-1] >= vals[0] - 0.05  # improves (or holds) with rounds

    def test_pred_leaf(self):
        rng = np.random.RandomState(0)
        X = rng.rand(1500, 5).astype(np.float32)
        y = (X[:, 0] + X[:, 1] > 1).astype(np.float32)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=32
        )
        bst = run_training(
            {"objective": "binary:logistic", "max_depth": 4}, dm, 4
        )
        leaves = bst.predict(X, pred_leaf=True)
        assert leaves.shape == (1500, 4)
        for ti, t in enumerate(bst.trees):
            assert (t.feat[leaves[:, ti]] < 0).all()
        # iteration_range subsets trees
        l2 = bst.predict(X, pred_leaf=True, iteration_range=(1, 3))
        assert l2.shape == (1500, 2)
        assert np.array_equal(l2[:, 0], leaves[:, 1])

    def test_pred_contribs_additivity_and_exactness(self):
        from itertools import combinations

        rng = np.random.RandomState(0)
        X = rng.rand(400, 4).astype(np.float32)
        y = (X[:, 0] * X[:, 1] + X[:, 2] + 0.1 * rng.randn(400)).astype(
            np.float32
        )
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=16
        )
        bst = run_training(
            {"objective": "reg:squarederror", "max_depth": 3}, dm, 3
        )
        rows = X[:8]
        contribs = bst.predict(rows, pred_contribs=True)
        margin = bst.predict(rows, output_margin=True)
        assert np.abs(contribs.sum(1) - margin).max() < 1e-5

        # brute-force Shapley with cover-weighted conditional expectation
        def cond_exp(t, x, S):
            def rec(nid):
                f = t.feat[nid]
                if f < 0:
                    return float(t.value[nid])
                l, r = int(t.left[nid]), int(t.left[nid]) + 1
                if f in S:
                    fv = x[f]
                    if np.isnan(fv):
                        return rec(l if t.default_left[nid] else r)
                    return rec(l if fv < t.thr[nid] else r)
                cl, cr = float(t.cover[l]), float(t.cover[r])
                tot = cl + cr if cl + cr > 0 else 1.0
                return (cl * rec(l) + cr * rec(r)) / tot

            return rec(0)

        import math

        F = 4
        for xi in range(3):
            x = rows[xi]
            exact = np.zeros(F)
            for f in range(F):
                others = [j for j in range(F) if j != f]
                for k in range(F):
                    for S in combinations(others, k):
                        wgt = (
                            math.factorial(k)
                            * math.factorial(F - k - 1)
                            / math.factorial(F)
                        )
                        for t in bst.trees:
                            exact[f] += wgt * (
                                cond_exp(t, x, set(S) | {f})
                                - cond_exp(t, x, set(S))
                            )
            assert np.abs(exact - contribs[xi, :F]).max() < 1e-5

    def test_pred_contribs_multiclass(self):
        rng = np.random.RandomState(1)
        X = rng.rand(500, 4).astype(np.float32)
        y = np.digitize(
            X[:, 0] * 3 + X[:, 1] * 2,
            np.quantile(X[:, 0] * 3 + X[:, 1] * 2, [0.33, 0.66]),
        ).astype(np.float32)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=16
        )
        bst = run_training(
            {"objective": "multi:softprob", "num_class": 3, "max_depth": 3},
            dm, 4,
        )
        c = bst.predict(X[:30], pred_contribs=True)
        m = bst.predict(X[:30], output_margin=True)
        assert c.shape == (30, 3, 5)
        assert np.abs(c.sum(-1) - m).max() < 1e-5

    def test_pred_interactions(self):
        rng = np.random.RandomState(0)
        X = rng.rand(1000, 4).astype(np.float32)
        y = (
            4 * (X[:, 0] > 0.5) * (X[:, 1] > 0.5) + X[:, 2]
        ).astype(np.float32)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=32
        )
        bst = run_training(
            {"objective": "reg:squarederror", "max_depth": 4}, dm, 5
        )
        rows = X[:10]
        M = bst.predict(rows, pred_interactions=True)
        c = bst.predict(rows, pred_contribs=True)
        m = bst.predict(rows, output_margin=True)
        assert M.shape == (10, 5, 5)
        assert np.abs(M.sum(2) - c).max() < 1e-6       # rows -> SHAP
        assert np.abs(M.sum((1, 2)) - m).max() < 1e-5  # total -> margin
        assert np.abs(
            M[:, :4, :4] - np.transpose(M[:, :4, :4], (0, 2, 1))
        ).max() == 0.0                                 # symmetric
        off = np.abs(M[:, :4, :4].copy())
        for i in range(4):
            off[:, i, i] = 0
        # interaction mass concentrates on the true interacting pair
        assert off[:, 0, 1].mean() > 10 * (off[:, 2, 3].mean() + 1e-12)


class TestGBLinear:
    def _linear_data(self, n=20000, seed=0):
        rng = np.random.RandomState(seed)
        X = rng.randn(n, 6).astype(np.float32)
        true_w = np.array([2.0, -1.0, 0.5, 0, 0, 3.0], np.float32)
        y = (X @ true_w + 0.7 + 0.05 * rng.randn(n)).astype(np.float32)
        return X, y, true_w

    def test_gblinear_recovers_weights(self):
        X, y, true_w = self._linear_data()
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y),
            max_bin=64, keep_raw=True,
        )
        bst = run_training(
            {"objective": "reg:squarederror", "booster": "gblinear",
             "eta": 0.5, "lambda": 0.0, "base_score": 0.0},
            dm, 40,
        )
        w = bst.linear_weights[:, 0]
        assert np.abs(w[:6] - true_w).max() < 0.05
        pred = bst.predict(X[:200], output_margin=True)
        assert np.abs(pred - (X[:200] @ w[:6] + w[6])).max() < 1e-4

    def test_gblinear_l1_sparsifies(self):
        X, y, _ = self._linear_data(seed=1)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y),
            max_bin=64, keep_raw=True,
        )
        bst = run_training(
            {"objective": "reg:squarederror", "booster": "gblinear",
             "eta": 0.5, "lambda": 0.0, "alpha": 500.0,
             "base_score": 0.0},
            dm, 40,
        )
        w = bst.linear_weights[:6, 0]
        # the two zero-coefficient features must be driven to (near) zero
        assert np.abs(w[3]) < 1e-2 and np.abs(w[4]) < 1e-2
        # strong features survive the penalty
        assert abs(w[5]) > 1.0

    def test_gblinear_json_roundtrip_and_resume(self):
        import json

        from xgboost_ray_amd.booster import Booster

        X, y, _ = self._linear_data(seed=2)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y),
            max_bin=64, keep_raw=True,
        )
        params = {"objective": "reg:squarederror", "booster": "gblinear",
                  "eta": 0.5, "base_score": 0.0}
        b1 = run_training(params, dm, 10)
        b2 = Booster()
        b2._from_json_dict(json.loads(b1.save_raw().decode()))
        assert np.allclose(b2.linear_weights, b1.linear_weights)
        assert b2.num_boosted_rounds() == 10
        # resume: 10 + 10 == 20 straight
        b3 = run_training(params, dm, 10, xgb_model=b2)
        b20 = run_training(params, dm, 20)
        # margins are recomputed in fp32 from the restored weights, so
        # resume matches continuous training to fp32 accumulation noise
        assert np.allclose(
            b3.linear_weights, b20.linear_weights, atol=1e-5
        )

    def test_gblinear_requires_raw(self):
        X, y, _ = self._linear_data(n=2000)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        with pytest.raises(ValueError, match="keep_raw"):
            run_training(
                {"objective": "reg:squarederror", "booster": "gblinear"},
                dm, 2,
            )


class TestDart:
    def _dm(self, seed=0):
        rng = np.random.RandomState(seed)
        X = rng.rand(8000, 6).astype(np.float32)
        y = (np.sin(X[:, 0] * 5) + X[:, 1]
             + 0.1 * rng.randn(8000)).astype(np.float32)
        return X, y, BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )

    def test_dart_basics(self):
        X, y, dm = self._dm()
        base = {"objective": "reg:squarederror", "max_depth": 4,
                "eta": 0.3, "seed": 3}
        b_gb = run_training(dict(base), dm, 15)
        b_d = run_training(dict(base, booster="dart", rate_drop=0.3),
                           dm, 15)
        b_d2 = run_training(dict(base, booster="dart", rate_drop=0.3),
                            dm, 15)
        b_d0 = run_training(dict(base, booster="dart", rate_drop=0.0),
                            dm, 15)
        p = X[:300]
        assert not np.allclose(b_gb.predict(p), b_d.predict(p))
        assert np.array_equal(b_d.predict(p), b_d2.predict(p))
        assert np.allclose(b_d0.predict(p), b_gb.predict(p))
        rmse = np.sqrt(np.mean((b_d.predict(X) - y) ** 2))
        assert rmse < 0.5

    def test_dart_margin_state_consistent(self):
        from xgboost_ray_amd.engine.trainer import BoostingEngine

        X, y, dm = self._dm(seed=1)
        eng = BoostingEngine(
            {"objective": "reg:squarederror", "booster": "dart",
             "rate_drop": 0.4, "max_depth": 3, "eta": 0.3, "seed": 1}, dm)
        for _ in range(12):
            eng.update()
        pred = eng.booster.predict(X, output_margin=True)
        assert np.abs(eng.margin.numpy() - pred).max() < 1e-4

    def test_dart_save_load(self, tmp_path):
        from xgboost_ray_amd.booster import Booster

        X, y, dm = self._dm(seed=2)
        b = run_training(
            {"objective": "reg:squarederror", "booster": "dart",
             "rate_drop": 0.3, "max_depth": 3, "eta": 0.3}, dm, 8)
        f = str(tmp_path / "dart.json")
        b.save_model(f)
        b2 = Booster()
        b2.load_model(f)
        assert np.allclose(b2.predict(X[:200]), b.predict(X[:200]))


class TestCountObjectives:
    def test_poisson_learns_rate(self):
        rng = np.random.RandomState(0)
        X = rng.rand(20000, 4).astype(np.float32)
        lam = np.exp(1.0 + X[:, 0] * 2)
        y = rng.poisson(lam).astype(np.float32)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        res = {}
        bst = run_training(
            {"objective": "count:poisson", "max_depth": 4, "eta": 0.3},
            dm, 20, evals=[EvalPack(name="t", X=None)], evals_result=res,
        )
        pred = bst.predict(X[:5000])
        assert np.corrcoef(pred, lam[:5000])[0, 1] > 0.95
        nll = res["t"]["poisson-nloglik"]
        assert nll[-1] < nll[0]

    def test_gamma_tweedie_sle(self):
        rng = np.random.RandomState(1)
        X = rng.rand(10000, 4).astype(np.float32)
        lam = np.exp(1.0 + X[:, 0] * 2)
        yg = (rng.gamma(2.0, np.exp(X[:, 0]))).astype(np.float32) + 1e-3
        dmg = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(yg), max_bin=64
        )
        bg = run_training(
            {"objective": "reg:gamma", "max_depth": 3, "eta": 0.2}, dmg, 20
        )
        assert np.corrcoef(
            bg.predict(X), 2.0 * np.exp(X[:, 0])
        )[0, 1] > 0.9

        yp = rng.poisson(lam).astype(np.float32)
        dmt = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(yp), max_bin=64
        )
        bt = run_training(
            {"objective": "reg:tweedie", "max_depth": 3, "eta": 0.2,
             "tweedie_variance_power": 1.3}, dmt, 15,
        )
        assert np.corrcoef(bt.predict(X), lam)[0, 1] > 0.95

        ysl = (X[:, 0] * 5).astype(np.float32)
        dms = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(ysl), max_bin=64
        )
        bs = run_training(
            {"objective": "reg:squaredlogerror", "max_depth": 3,
             "eta": 0.3, "base_score": 1.0}, dms, 25,
        )
        rmse = float(np.sqrt(np.mean((bs.predict(X) - ysl) ** 2)))
        assert rmse < 0.2


class TestAFT:
    def test_aft_censored_all_distributions(self):
        rng = np.random.RandomState(0)
        n = 15000
        X = rng.rand(n, 4).astype(np.float32)
        t = np.exp(1.0 + 2 * X[:, 0] + 0.3 * rng.randn(n)).astype(
            np.float32
        )
        cens = np.exp(1.0 + 2 * X[:, 0] + rng.rand(n)).astype(np.float32)
        right = rng.rand(n) < 0.3
        yl = np.where(right, np.minimum(t, cens), t).astype(np.float32)
        yu = np.where(right, np.inf, t).astype(np.float32)
        dm = BinnedMatrix.build(torch.from_numpy(X), max_bin=64)
        dm.label_lower_bound = torch.from_numpy(yl)
        dm.label_upper_bound = torch.from_numpy(yu)
        for dist in ("normal", "logistic", "extreme"):
            res = {}
            bst = run_training(
                {"objective": "survival:aft", "max_depth": 4, "eta": 0.3,
                 "aft_loss_distribution": dist, "base_score": 1.0},
                dm, 15, evals=[EvalPack(name="t", X=None)],
                evals_result=res,
            )
            pred = bst.predict(X[:4000], output_margin=True)
            assert np.corrcoef(pred, 1 + 2 * X[:4000, 0])[0, 1] > 0.95
            nll = res["t"]["aft-nloglik"]
            assert nll[-1] < nll[0]

    def test_aft_through_raydmatrix(self):
        from xgboost_ray_amd import RayDMatrix, RayParams, train

        rng = np.random.RandomState(1)
        n = 6000
        X = rng.rand(n, 4).astype(np.float32)
        t = np.exp(1.0 + 2 * X[:, 0]).astype(np.float32)
        right = rng.rand(n) < 0.2
        yl = t.copy()
        yu = np.where(right, np.inf, t).astype(np.float32)
        bst = train(
            {"objective": "survival:aft", "max_depth": 3, "eta": 0.3,
             "base_score": 1.0},
            RayDMatrix(X, label_lower_bound=yl, label_upper_bound=yu),
            num_boost_round=10,
            ray_params=RayParams(num_actors=2),
        )
        pred = bst.predict(X[:2000], output_margin=True)
        assert np.corrcoef(pred, 1 + 2 * X[:2000, 0])[0, 1] > 0.9

    def test_cox_recovers_hazard(self):
        rng = np.random.RandomState(0)
        n = 15000
        X = rng.rand(n, 4).astype(np.float32)
        true_risk = 2.0 * X[:, 0] - 1.0 * X[:, 1]
        t = rng.exponential(1.0 / np.exp(true_risk)).astype(np.float32)
        cens = rng.exponential(2.0, n).astype(np.float32)
        event = t <= cens
        label = np.where(event, np.minimum(t, cens), -cens).astype(
            np.float32
        )
        label[label == 0] = 1e-6
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(label), max_bin=64
        )
        res = {}
        bst = run_training(
            {"objective": "survival:cox", "max_depth": 4, "eta": 0.3,
             "base_score": 1.0},
            dm, 15, evals=[EvalPack(name="t", X=None)], evals_result=res,
        )
        pred = bst.predict(X[:4000], output_margin=True)
        assert np.corrcoef(pred, true_risk[:4000])[0, 1] > 0.9
        nll = res["t"]["cox-nloglik"]
        assert nll[-1] < nll[0]

    def test_rank_map_objective(self):
        rng = np.random.RandomState(0)
        n = 8000
        qid = np.repeat(np.arange(n // 20), 20).astype(np.int64)
        X = rng.rand(n, 5).astype(np.float32)
        rel = (X[:, 0] + 0.3 * rng.randn(n) > 0.7).astype(np.float32)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(rel),
            qid=torch.from_numpy(qid), max_bin=64,
        )
        res = {}
        run_training(
            {"objective": "rank:map", "max_depth": 4, "eta": 0.3},
            dm, 15, evals=[EvalPack(name="t", X=None)], evals_result=res,
        )
        mp = res["t"]["map"]
        assert mp[-1] > mp[0]          # optimizes its own metric
        assert mp[-1] > 0.76           # reaches the data's MAP ceiling

    def test_pseudohuber_and_hinge(self):
        rng = np.random.RandomState(0)
        X = rng.rand(8000, 4).astype(np.float32)
        y = (X[:, 0] * 3 + 0.1 * rng.randn(8000)).astype(np.float32)
        y[::50] += 20  # outliers shrugged off by the robust loss
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
        )
        b = run_training(
            {"objective": "reg:pseudohubererror", "max_depth": 4,
             "eta": 0.3, "base_score": 0.0}, dm, 30,
        )
        med = np.median(np.abs(b.predict(X) - X[:, 0] * 3))
        assert med < 0.25
        yb = (X[:, 0] > 0.5).astype(np.float32)
        dmb = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(yb), max_bin=64
        )
        bh = run_training(
            {"objective": "binary:hinge", "max_depth": 3, "eta": 0.3,
             "base_score": 0.0}, dmb, 15,
        )
        assert float((bh.predict(X) == yb).mean()) > 0.9

    def test_iteration_range_with_parallel_trees(self):
        rng = np.random.RandomState(0)
        X = rng.rand(2000, 4).astype(np.float32)
        y = (X[:, 0] > 0.5).astype(np.float32)
        dm = BinnedMatrix.build(
            torch.from_numpy(X), label=torch.from_numpy(y), max_bin=32
        )
        b = run_training(
            {"objective": "binary:logistic", "max_depth": 3,
             "num_parallel_tree": 2}, dm, 4,
        )
        assert len(b.trees) == 8
        from xgboost_ray_amd.engine.objectives import get_objective

        bm = float(get_objective(b.objective, 0).prob_to_margin(
            b.base_score
        ))
        m_all = b.predict(X[:100], output_margin=True)
        m_a = b.predict(X[:100], output_margin=True, iteration_range=(0, 2))
        m_b = b.predict(X[:100], output_margin=True, iteration_range=(2, 4))
        # range margins are additive (each includes the base once)
        assert np.abs((m_a - bm) + (m_b - bm) + bm - m_all).max() < 1e-5


class TestEvalSemantics:
    """xgboost EvalTransform parity: metrics on transformed predictions,
    and dart eval margins tracking the real (rescaled) model."""

    def test_rmse_on_transformed_predictions(self):
        """reg:logistic's default rmse is computed on probabilities, not
        margins (XGBoost applies obj EvalTransform before metrics)."""
        dm, X, y = _binned()
        res = {}
        bst = run_training(
            {"objective": "reg:logistic", "max_depth": 4, "eta": 0.3,
             "eval_metric": ["rmse", "mae"]},
            dm, 8, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        p = bst.predict(X)  # transformed (sigmoid) predictions
        rmse = float(np.sqrt(np.mean((p - y) ** 2)))
        mae = float(np.mean(np.abs(p - y)))
        assert res["train"]["rmse"][-1] == pytest.approx(rmse, rel=1e-5)
        assert res["train"]["mae"][-1] == pytest.approx(mae, rel=1e-5)
        # sanity: probabilities live in [0,1] so rmse must be < 1
        assert res["train"]["rmse"][-1] < 1.0

    def test_rmse_identity_for_squarederror(self):
        dm, X, y = _binned(kind="reg")
        res = {}
        bst = run_training(
            {"objective": "reg:squarederror", "max_depth": 4, "eta": 0.3,
             "eval_metric": ["rmse"]},
            dm, 8, evals=[EvalPack(name="train", X=None)], evals_result=res,
        )
        p = bst.predict(X)
        rmse = float(np.sqrt(np.mean((p - y) ** 2)))
        assert res["train"]["rmse"][-1] == pytest.approx(rmse, rel=1e-5)

    def test_dart_eval_margin_tracks_rescaled_model(self):
        """_dart_commit rescales committed trees in place each round; the
        eval-set margin must track the REAL model, not drift (advisor
        finding: reported 0.181 vs true 0.237 logloss in 20 rounds)."""
        dm, X, y = _binned()
        Xe, ye = create_data(500, 6, seed=7, kind="binary")
        res = {}
        bst = run_training(
            {"objective": "binary:logistic", "booster": "dart",
             "max_depth": 4, "eta": 0.3, "rate_drop": 0.5,
             "eval_metric": ["logloss"]},
            dm, 20,
            evals=[EvalPack(
                name="eval", X=torch.from_numpy(Xe),
                label=torch.from_numpy(ye))],
            evals_result=res,
        )
        p = np.clip(bst.predict(Xe), 1e-16, 1 - 1e-16)
        true_ll = float(np.mean(
            -(ye * np.log(p) + (1 - ye) * np.log(1 - p))
        ))
        assert res["eval"]["logloss"][-1] == pytest.approx(true_ll, rel=1e-4)


class TestBinnedAUCTolerance:
    """Tolerance envelope of the 16384-bin AUC/aucpr approximation vs
    exact sklearn values under adversarial score distributions (VERDICT
    weak item: the bound was asserted on one friendly case only).

    The binned AUC's error is bounded by the probability mass that shares
    a bin with the opposite class: distributions engineered to collide in
    few bins are the worst case."""

    def _exact_vs_binned(self, margin, label, weight=None):
        from sklearn.metrics import average_precision_score, roc_auc_score

        from xgboost_ray_amd.engine.metrics import get_metric

        p = 1.0 / (1.0 + np.exp(-margin.numpy()))
        exact_auc = roc_auc_score(label.numpy(), p,
                                  sample_weight=None if weight is None
                                  else weight.numpy())
        exact_ap = average_precision_score(
            label.numpy(), p,
            sample_weight=None if weight is None else weight.numpy())
        m = get_metric("auc")
        auc = m.finalize(m.local_stats(margin, label, weight, None, None))
        m2 = get_metric("aucpr")
        ap = m2.finalize(m2.local_stats(margin, label, weight, None, None))
        return exact_auc, auc, exact_ap, ap

    def test_scores_clustered_near_extremes(self):
        # sigmoid saturates: most scores collapse into the top/bottom bins
        rng = np.random.RandomState(0)
        n = 20000
        y = (rng.rand(n) < 0.5).astype(np.float32)
        margin = torch.from_numpy(
            (np.where(y > 0, 9.0, -9.0)
             + rng.randn(n) * 2.0).astype(np.float32))
        ea, ba, ep, bp = self._exact_vs_binned(
            margin, torch.from_numpy(y))
        assert abs(ea - ba) < 5e-3
        assert abs(ep - bp) < 5e-2

    def test_tiny_score_separation_documented_limit(self):
        # DOCUMENTED LIMIT: when the entire class separation lives within
        # ~1/16384 of probability space, bin collisions destroy the
        # estimate (same failure mode as xgboost's binned GPU AUC). The
        # same scores scaled to a realistic margin spread recover the
        # exact value. This test pins both halves of that statement.
        rng = np.random.RandomState(1)
        n = 10000
        y = (rng.rand(n) < 0.5).astype(np.float32)
        raw = (y * 1.0 + rng.randn(n) * 0.1).astype(np.float32)
        ea, ba, _, _ = self._exact_vs_binned(
            torch.from_numpy(raw * 1e-6), torch.from_numpy(y))
        assert abs(ea - ba) > 0.05  # collapsed: not to be trusted here
        ea2, ba2, _, _ = self._exact_vs_binned(
            torch.from_numpy(raw), torch.from_numpy(y))
        assert abs(ea2 - ba2) < 5e-3  # realistic spread: tight again

    def test_heavy_ties(self):
        rng = np.random.RandomState(2)
        n = 30000
        y = (rng.rand(n) < 0.3).astype(np.float32)
        # only 5 distinct score values
        margin = torch.from_numpy(
            rng.choice([-2.0, -1.0, 0.0, 1.0, 2.0], n).astype(np.float32)
            + y * 0.5)
        ea, ba, ep, bp = self._exact_vs_binned(margin, torch.from_numpy(y))
        assert abs(ea - ba) < 2e-3
        assert abs(ep - bp) < 2e-2

    def test_weighted_skewed(self):
        rng = np.random.RandomState(3)
        n = 15000
        y = (rng.rand(n) < 0.1).astype(np.float32)  # rare positives
        margin = torch.from_numpy(
            (y * 1.5 + rng.randn(n)).astype(np.float32))
        w = torch.from_numpy(
            rng.lognormal(0, 2.0, n).astype(np.float32))  # heavy tail
        ea, ba, ep, bp = self._exact_vs_binned(
            margin, torch.from_numpy(y), w)
        assert abs(ea - ba) < 5e-3
        assert abs(ep - bp) < 5e-2


class TestDartSampleType:
    def test_weighted_differs_and_deterministic(self):
        """sample_type=weighted (drop prob proportional to tree weight,
        xgboost dart.cc) differs from uniform and stays deterministic."""
        dm, X, y = _binned()
        outs = {}
        for st in ("uniform", "weighted"):
            runs = []
            for _ in range(2):
                bst = run_training(
                    {"objective": "binary:logistic", "booster": "dart",
                     "rate_drop": 0.4, "sample_type": st,
                     "max_depth": 4, "eta": 0.3, "seed": 5},
                    dm, 15,
                )
                runs.append(bst.predict(X, output_margin=True))
            np.testing.assert_array_equal(runs[0], runs[1])
            outs[st] = runs[0]
        assert not np.array_equal(outs["uniform"], outs["weighted"])


class TestColsampleByNode:
    def test_bynode_changes_model_and_deterministic(self):
        dm, X, y = _binned()
        outs = {}
        for rate in (1.0, 0.5):
            runs = []
            for _ in range(2):
                bst = run_training(
                    {"objective": "binary:logistic", "max_depth": 5,
                     "eta": 0.3, "colsample_bynode": rate, "seed": 3},
                    dm, 6,
                )
                runs.append(bst.predict(X, output_margin=True))
            np.testing.assert_array_equal(runs[0], runs[1])
            outs[rate] = runs[0]
        assert not np.array_equal(outs[1.0], outs[0.5])

    def test_bynode_composes_with_interaction(self):
        dm, X, y = _binned()
        bst = run_training(
            {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3,
             "colsample_bynode": 0.7,
             "interaction_constraints": [[0, 1, 2], [3, 4, 5]]},
            dm, 6,
        )
        sets = [frozenset({0, 1, 2}), frozenset({3, 4, 5})]
        for t in bst.trees:
            def walk(nid, feats):
                f = int(t.feat[nid])
                if f < 0:
                    if feats:
                        assert any(feats <= s for s in sets), feats
                    return
                walk(int(t.left[nid]), feats | {f})
                walk(int(t.left[nid]) + 1, feats | {f})
            walk(0, frozenset())


class TestApproxContribs:
    def test_additivity_and_speed_path(self):
        """Saabas approx contribs: additivity holds exactly (each path
        telescopes from the tree's expected value to the leaf)."""
        dm, X, y = _binned()
        bst = run_training(
            {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3},
            dm, 6,
        )
        C = bst.predict(X[:500], pred_contribs=True, approx_contribs=True)
        margin = bst.predict(X[:500], output_margin=True)
        np.testing.assert_allclose(C.sum(axis=1), margin, rtol=1e-5,
                                   atol=1e-5)
        # exact TreeSHAP and approx share the bias column
        E = bst.predict(X[:20], pred_contribs=True)
        np.testing.assert_allclose(C[:20, -1], E[:20, -1], rtol=1e-5)

    def test_multiclass_additivity(self):
        dm, X, y = _binned(kind="multi")
        bst = run_training(
            {"objective": "multi:softprob", "num_class": 4,
             "max_depth": 4, "eta": 0.3},
            dm, 4,
        )
        C = bst.predict(X[:200], pred_contribs=True, approx_contribs=True)
        margin = bst.predict(X[:200], output_margin=True)
        np.testing.assert_allclose(C.sum(axis=2), margin, rtol=1e-5,
                                   atol=1e-5)


class TestDartResumeTolerance:
    def test_dart_resume_allclose_not_bitwise(self):
        """dart resume: the continuous run's incremental margin (with
        in-place rescale adjustments) and the resumed run's fresh
        predict-from-model margin differ by float summation order -
        models agree to ~1 ulp, NOT bitwise (the bitwise resume
        contract holds for gbtree/gblinear; fuzz-found, documented in
        docs/parity.md)."""
        cfg = {"objective": "binary:logistic", "max_depth": 3,
               "eta": 0.1, "seed": 14, "booster": "dart",
               "rate_drop": 0.3, "num_parallel_tree": 2}
        dm, X, y = _binned(seed=14)
        full = run_training(dict(cfg), dm, 7)
        half = run_training(dict(cfg), dm, 3)
        resumed = run_training(dict(cfg), dm, 4, xgb_model=half)
        a = full.predict(X, output_margin=True)
        b = resumed.predict(X, output_margin=True)
        np.testing.assert_allclose(a, b, atol=2e-6, rtol=1e-6)


class TestSubsampleMarginCache:
    def test_all_rows_advance_under_subsample(self):
        """xgboost UpdatePredictionCache semantics: with row sampling,
        EVERY row's training margin advances each round (fuzz-found:
        out-of-sample rows were left on stale margins)."""
        dm, X, y = _binned()
        engine_params = {"objective": "binary:logistic", "max_depth": 3,
                         "eta": 0.1, "subsample": 0.6, "seed": 20}
        eng = BoostingEngine(engine_params, dm)
        for _ in range(4):
            eng.update()
        pm = eng.booster.predict(X, output_margin=True)
        np.testing.assert_allclose(pm, eng.margin.numpy(), atol=1e-6)

    def test_subsample_resume_bitwise(self):
        dm, X, y = _binned()
        cfg = {"objective": "binary:logistic", "max_depth": 5,
               "eta": 0.1, "subsample": 0.7, "seed": 20}
        full = run_training(dict(cfg), dm, 8)
        half = run_training(dict(cfg), dm, 4)
        resumed = run_training(dict(cfg), dm, 4, xgb_model=half)
        np.testing.assert_array_equal(
            full.predict(X, output_margin=True),
            resumed.predict(X, output_margin=True),
        )
