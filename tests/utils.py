"""Shared test helpers (reference tests/utils.py analogue)."""

import numpy as np


def create_data(n: int = 2048, f: int = 8, seed: int = 0, kind: str = "binary"):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f).astype(np.float32)
    if kind == "binary":
        y = ((X[:, 0] + 0.5 * X[:, 1] + 0.2 * rng.randn(n)) > 0).astype(
            np.float32
        )
    elif kind == "reg":
        y = (X[:, 0] * 2 - X[:, 1] + 0.1 * rng.randn(n)).astype(np.float32)
    elif kind == "multi":
        y = (
            (X[:, 0] > 0).astype(np.int64) * 2 + (X[:, 1] > 0).astype(np.int64)
        ).astype(np.float32)
    else:
        raise ValueError(kind)
    return X, y


def create_labeled_sorted_rank_data(n_groups=20, group_size=30, f=6, seed=0):
    rng = np.random.RandomState(seed)
    n = n_groups * group_size
    X = rng.randn(n, f).astype(np.float32)
    rel = X[:, 0] + 0.5 * rng.randn(n)
    # 3-level relevance per group
    y = np.zeros(n, np.float32)
    qid = np.repeat(np.arange(n_groups), group_size)
    for g in range(n_groups):
        sl = slice(g * group_size, (g + 1) * group_size)
        r = rel[sl]
        q = np.quantile(r, [0.5, 0.8])
        y[sl] = np.digitize(r, q)
    return X, y, qid.astype(np.int64)


def one_hot_impossible_halves(repeat: int = 32):
    """The reference's canonical allreduce-correctness dataset
    (reference test_end_to_end.py:56-90): 4 one-hot rows with labels
    0..3, repeated; BATCH-sharded halves only ever see 2 of the 4
    classes, so each actor alone cannot learn the task - 100% accuracy
    is only reachable through correct histogram allreduce."""
    x = np.eye(4, dtype=np.float32)
    y = np.arange(4, dtype=np.float32)
    X = np.tile(x, (repeat, 1))
    Y = np.tile(y, repeat)
    # sort so BATCH sharding gives each half only 2 classes
    order = np.argsort(Y, kind="stable")
    return X[order], Y[order]
