"""SHAP explainability + alternative boosters on the distributed API.

Trains a small model with each booster family (gbtree, dart, gblinear),
then explains predictions with pred_contribs (exact TreeSHAP),
pred_interactions, and pred_leaf.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from xgboost_ray_amd import RayDMatrix, RayParams, predict, train


def main():
    rng = np.random.RandomState(0)
    X = rng.rand(20000, 5).astype(np.float32)
    # f0*f1 interaction + f2 main effect
    y = (3 * (X[:, 0] > 0.5) * (X[:, 1] > 0.5) + X[:, 2]).astype(np.float32)
    rp = RayParams(num_actors=2)

    for booster in ("gbtree", "dart", "gblinear"):
        params = {"objective": "reg:squarederror", "booster": booster,
                  "max_depth": 4, "eta": 0.3}
        if booster == "dart":
            params["rate_drop"] = 0.2
        bst = train(params, RayDMatrix(X, y), num_boost_round=20,
                    ray_params=rp)
        rmse = float(np.sqrt(np.mean(
            (predict(bst, RayDMatrix(X), ray_params=rp) - y) ** 2)))
        print(f"{booster:9s} rmse={rmse:.4f}")

    # explain the gbtree model
    bst = train({"objective": "reg:squarederror", "max_depth": 4,
                 "eta": 0.3}, RayDMatrix(X, y), num_boost_round=20,
                ray_params=rp)
    rows = X[:5]
    contribs = bst.predict(rows, pred_contribs=True)
    print("SHAP contribs (row 0):", np.round(contribs[0], 3))
    inter = bst.predict(rows, pred_interactions=True)
    print("strongest interaction pair:",
          np.unravel_index(np.abs(inter[0, :5, :5]
                                  - np.diag(np.diag(inter[0, :5, :5])))
                           .argmax(), (5, 5)))
    leaves = bst.predict(rows, pred_leaf=True)
    print("leaf ids (row 0):", leaves[0][:8])


if __name__ == "__main__":
    main()
