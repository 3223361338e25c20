"""Distributed prediction example (reference examples/simple_predict.py)."""

import numpy as np

from xgboost_ray_amd import RayDMatrix, RayParams, predict, train


def main():
    rng = np.random.RandomState(1234)
    n, f = 50_000, 10
    X = rng.randn(n, f).astype(np.float32)
    y = ((X[:, 0] - X[:, 2]) > 0).astype(np.float32)

    bst = train(
        {"objective": "binary:logistic"},
        RayDMatrix(X, label=y),
        20,
        ray_params=RayParams(num_actors=2),
    )
    bst.save_model("simple_predict.json")

    data = RayDMatrix(X)
    pred = predict(bst, data, ray_params=RayParams(num_actors=2))
    acc = ((pred > 0.5).astype(np.float32) == y).mean()
    print(f"Prediction accuracy: {acc:.4f}")


if __name__ == "__main__":
    main()
