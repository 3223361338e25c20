"""sklearn API tests (mirrors reference test_sklearn.py highlights)."""

import os
import pickle

import numpy as np
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

RP = RayParams(num_actors=2)


class TestClassifier:
    def test_binary(self):
        X, y = create_data(2000, 6)
        clf = RayXGBClassifier(n_estimators=10, max_depth=4)
        clf.fit(X, y, ray_params=RP)
        pred = clf.predict(X, ray_params=RP)
        assert (pred == y).mean() > 0.9
        proba = clf.predict_proba(X, ray_params=RP)
        assert proba.shape == (2000, 2)
        np.testing.assert_allclose(proba.sum(axis=1), 1.0, rtol=1e-5)

    def test_multiclass_label_encoding(self):
        X, y = create_data(1500, 6, kind="multi")
        labels = np.array(["a", "b", "c", "d"])[y.astype(int)]
        clf = RayXGBClassifier(n_estimators=8, max_depth=4)
        clf.fit(X, labels, ray_params=RP)
        assert clf.n_classes_ == 4
        pred = clf.predict(X, ray_params=RP)
        assert set(pred) <= {"a", "b", "c", "d"}
        assert (pred == labels).mean() > 0.9

    def test_eval_set(self):
        X, y = create_data(1500, 5)
        Xv, yv = create_data(400, 5, seed=3)
        clf = RayXGBClassifier(n_estimators=8, max_depth=4,
                               eval_metric=["logloss"])
        clf.fit(X, y, eval_set=[(Xv, yv)], ray_params=RP, verbose=False)
        assert "validation_0" in clf.evals_result_
        assert len(clf.evals_result_["validation_0"]["logloss"]) == 8

    def test_feature_importances(self):
        X, y = create_data(1000, 5)
        clf = RayXGBClassifier(n_estimators=5, max_depth=3)
        clf.fit(X, y, ray_params=RP)
        imp = clf.feature_importances_
        assert imp.shape == (5,)
        assert abs(imp.sum() - 1.0) < 1e-6
        assert imp[0] > 0.1  # X0 drives the label


class TestRegressor:
    def test_fit_predict(self):
        X, y = create_data(2000, 6, kind="reg")
        reg = RayXGBRegressor(n_estimators=15, max_depth=4)
        reg.fit(X, y, ray_params=RP)
        pred = reg.predict(X, ray_params=RP)
        rmse = np.sqrt(np.mean((pred - y) ** 2))
        assert rmse < np.std(y) * 0.5

    def test_save_load_roundtrip(self, tmp_path):
        X, y = create_data(800, 4, kind="reg")
        reg = RayXGBRegressor(n_estimators=5, max_depth=3)
        reg.fit(X, y, ray_params=RP)
        path = str(tmp_path / "model.json")
        reg.save_model(path)
        reg2 = RayXGBRegressor()
        reg2.load_model(path)
        np.testing.assert_allclose(
            reg.get_booster().predict(X), reg2.get_booster().predict(X),
            rtol=1e-6,
        )

    def test_resume_xgb_model(self):
        X, y = create_data(800, 4, kind="reg")
        reg1 = RayXGBRegressor(n_estimators=5, max_depth=3)
        reg1.fit(X, y, ray_params=RP)
        reg2 = RayXGBRegressor(n_estimators=5, max_depth=3)
        reg2.fit(X, y, xgb_model=reg1, ray_params=RP)
        assert reg2.get_booster().num_boosted_rounds() == 10

    def test_pickle_estimator(self):
        X, y = create_data(500, 4, kind="reg")
        reg = RayXGBRegressor(n_estimators=3, max_depth=3)
        reg.fit(X, y, ray_params=RP)
        blob = pickle.dumps(reg)
        reg2 = pickle.loads(blob)
        np.testing.assert_array_equal(
            reg.get_booster().predict(X), reg2.get_booster().predict(X)
        )


class TestRF:
    def test_rf_variants_run(self):
        X, y = create_data(1000, 5)
        for cls in (RayXGBRFClassifier, RayXGBRFRegressor):
            model = cls(n_estimators=5, max_depth=4)
            model.fit(X, y, ray_params=RP)
            bst = model.get_booster()
            # RF: ONE boosting round of n_estimators parallel trees
            assert bst.num_boosted_rounds() == 1
            assert len(bst.trees) == 5
        # averaged forest predictions stay in a sane range
        reg = RayXGBRFRegressor(n_estimators=10, max_depth=4)
        Xr, yr = create_data(1500, 5, kind="reg")
        reg.fit(Xr, yr, ray_params=RP)
        pred = reg.get_booster().predict(Xr)
        assert abs(pred.mean() - yr.mean()) < 0.5
        assert np.corrcoef(pred, yr)[0, 1] > 0.6


class TestRanker:
    def test_rank_fit_predict(self):
        X, y, qid = create_labeled_sorted_rank_data(n_groups=16, group_size=25)
        rk = RayXGBRanker(n_estimators=8, max_depth=3, objective="rank:ndcg")
        rk.fit(X, y, qid=qid, ray_params=RayParams(num_actors=1))
        scores = rk.predict(X, ray_params=RayParams(num_actors=1))
        assert scores.shape == (400,)
        # scores should correlate positively with relevance
        corr = np.corrcoef(scores, y)[0, 1]
        assert corr > 0.3

    def test_requires_qid(self):
        X, y = create_data(100, 4)
        with pytest.raises(ValueError, match="qid"):
            RayXGBRanker().fit(X, y)

    def test_group_rejected(self):
        X, y = create_data(100, 4)
        with pytest.raises(ValueError, match="group"):
            RayXGBRanker().fit(X, y, group=[50, 50])


class TestSklearnInterop:
    def test_get_set_params_clone(self):
        from sklearn.base import clone

        clf = RayXGBClassifier(n_estimators=7, max_depth=3)
        params = clf.get_params()
        assert params["n_estimators"] == 7
        c2 = clone(clf)
        assert c2.get_params()["n_estimators"] == 7

    def test_ray_dmatrix_passthrough(self):
        X, y = create_data(800, 4)
        dm = RayDMatrix(X, label=y)
        # num_class is required for RayDMatrix classifier fits (labels
        # cannot be inspected; reference test_sklearn_matrix.py:48)
        clf = RayXGBClassifier(n_estimators=4, max_depth=3, num_class=2)
        clf.fit(dm, None, ray_params=RP)
        pred = clf.predict(RayDMatrix(X), ray_params=RP)
        assert len(pred) == 800


class TestSklearnEcosystem:
    def test_grid_search_cv(self):
        from sklearn.model_selection import GridSearchCV

        X, y = create_data(600, 4)
        clf = RayXGBClassifier(n_estimators=3, max_depth=3, n_jobs=1)
        gs = GridSearchCV(
            clf, {"max_depth": [2, 3]}, cv=2, scoring="accuracy"
        )
        gs.fit(X, y)
        assert gs.best_params_["max_depth"] in (2, 3)
        assert gs.best_score_ > 0.8

    def test_cross_val_score(self):
        from sklearn.model_selection import cross_val_score

        X, y = create_data(600, 4, kind="reg")
        reg = RayXGBRegressor(n_estimators=4, max_depth=3, n_jobs=1)
        scores = cross_val_score(reg, X, y, cv=2, scoring="r2")
        assert len(scores) == 2
        assert scores.mean() > 0.5


def test_sklearn_gblinear():
    """booster='gblinear' through the sklearn estimator (reference
    test_sklearn_api_gblinear)."""
    import numpy as np

    from xgboost_ray_amd import RayParams, RayXGBRegressor

    rng = np.random.RandomState(0)
    X = rng.randn(4000, 4).astype(np.float32)
    y = (X @ np.array([1.0, -1.0, 2.0, 0.0], np.float32)).astype(np.float32)
    reg = RayXGBRegressor(
        booster="gblinear", n_estimators=30, learning_rate=0.5,
        reg_lambda=0.0, base_score=0.0,
    )
    reg.fit(X, y, ray_params=RayParams(num_actors=2))
    pred = reg.predict(X[:500], ray_params=RayParams(num_actors=2))
    assert np.abs(pred - y[:500]).max() < 0.1
