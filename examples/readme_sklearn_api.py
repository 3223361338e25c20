"""sklearn API example (reference examples/readme_sklearn_api.py)."""

from sklearn.datasets import load_breast_cancer
from sklearn.model_selection import train_test_split

from xgboost_ray_amd import RayParams, RayXGBClassifier


def main():
    seed = 42
    X, y = load_breast_cancer(return_X_y=True)
    X_train, X_test, y_train, y_test = train_test_split(
        X, y, train_size=0.25, random_state=42
    )

    clf = RayXGBClassifier(
        n_jobs=2,  # number of actors
        random_state=seed,
    )
    clf.fit(X_train, y_train)

    pred_ray = clf.predict(X_test)
    print(pred_ray[:10])

    pred_proba_ray = clf.predict_proba(X_test)
    print(pred_proba_ray[:3])

    acc = (pred_ray == y_test).mean()
    print(f"Test accuracy: {acc:.4f}")


if __name__ == "__main__":
    main()
