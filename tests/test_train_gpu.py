"""End-to-end train()/predict() through the actor layer on a real GPU."""

import numpy as np
import pytest

from tests.utils import create_data
from xgboost_ray_amd import RayDMatrix, RayParams, predict, train

pytestmark = pytest.mark.gpu


def test_train_single_gpu_actor():
    X, y = create_data(200_000, 12)
    res, add = {}, {}
    bst = train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 6, "eta": 0.3, "eval_metric": ["logloss", "auc"]},
        RayDMatrix(X, label=y),
        10,
        evals_result=res,
        additional_results=add,
        ray_params=RayParams(num_actors=1, gpus_per_actor=1),
        evals=[(RayDMatrix(X, label=y), "train")],
    )
    assert bst.num_boosted_rounds() == 10
    assert res["train"]["auc"][-1] > 0.9
    pred = predict(
        bst, RayDMatrix(X), ray_params=RayParams(num_actors=1, gpus_per_actor=1)
    )
    acc = ((pred > 0.5) == (y > 0.5)).mean()
    assert acc > 0.85


def test_train_gpu_fault_restart(tmp_path):
    """SIGKILL mid-training + checkpoint restart on the GPU path."""
    from tests.fault_tolerance import DieCallback

    X, y = create_data(100_000, 8)
    bst = train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 5},
        RayDMatrix(X, label=y),
        12,
        ray_params=RayParams(num_actors=1, gpus_per_actor=1,
                             max_actor_restarts=1, checkpoint_frequency=3),
        callbacks=[DieCallback(die_round=6, die_rank=0,
                               lock_dir=str(tmp_path))],
    )
    assert bst.num_boosted_rounds() == 12


def test_streaming_train_gpu(tmp_path):
    """Out-of-core path (RayDeviceQuantileDMatrix) through GPU actors."""
    import pandas as pd

    from xgboost_ray_amd import RayDeviceQuantileDMatrix

    rng = np.random.RandomState(0)
    files = []
    for i in range(4):
        Xc = rng.randn(100_000, 10).astype(np.float32)
        yc = ((Xc[:, 0] + Xc[:, 1]) > 0).astype(np.float32)
        df = pd.DataFrame(Xc, columns=[f"f{j}" for j in range(10)])
        df["label"] = yc
        path = str(tmp_path / f"s{i}.parquet")
        df.to_parquet(path)
        files.append(path)
    res = {}
    dm = RayDeviceQuantileDMatrix(files, label="label")
    bst = train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 6, "eval_metric": ["auc"]},
        dm,
        8,
        evals=[(dm, "train")],
        evals_result=res,
        ray_params=RayParams(num_actors=1, gpus_per_actor=1),
    )
    assert bst.num_boosted_rounds() == 8
    assert res["train"]["auc"][-1] > 0.9


def test_deep_tree_gpu():
    """depth 12 / 4096-node frontiers exercise the chunked machinery."""
    X, y = create_data(300_000, 10)
    res = {}
    train(
        {"objective": "binary:logistic", "tree_method": "gpu_hist",
         "max_depth": 12, "eta": 0.3, "eval_metric": ["logloss"]},
        RayDMatrix(X, label=y), 4,
        evals_result=res,
        evals=[(RayDMatrix(X, label=y), "train")],
        ray_params=RayParams(num_actors=1, gpus_per_actor=1),
    )
    assert res["train"]["logloss"][-1] < 0.3


@pytest.mark.gpu
def test_interaction_constraints_gpu():
    """GPU-trained trees obey interaction constraints (scan-kernel gate)."""
    import torch
    from tests.test_engine_cpu import _paths_respect_constraints
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    rng = np.random.RandomState(3)
    n = 20000
    X = rng.rand(n, 6).astype(np.float32)
    y = (
        X[:, 0] * X[:, 3] + X[:, 1] * X[:, 4] + X[:, 2] * X[:, 5]
        + 0.05 * rng.randn(n)
    ).astype(np.float32)
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(),
        label=torch.from_numpy(y).cuda(),
        max_bin=64,
    )
    sets = [[0, 1, 2], [3, 4, 5]]
    bst = run_training(
        {"objective": "reg:squarederror", "max_depth": 5, "eta": 0.3,
         "interaction_constraints": sets},
        dm, 10,
    )
    assert _paths_respect_constraints(bst, sets)

    # and bitwise CPU == GPU with the constraint active
    dm_cpu = BinnedMatrix.build(
        torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
    )
    bst_cpu = run_training(
        {"objective": "reg:squarederror", "max_depth": 5, "eta": 0.3,
         "interaction_constraints": sets},
        dm_cpu, 10,
    )
    for tg, tc in zip(bst.trees, bst_cpu.trees):
        assert np.array_equal(tg.feat, tc.feat)
        assert np.array_equal(tg.thr, tc.thr)
        assert np.array_equal(tg.value, tc.value)


@pytest.mark.gpu
def test_lossguide_and_sampling_gpu():
    """lossguide growth + row/column sampling exercise the GPU K=1 path."""
    import torch
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    X, y = create_data(200000, 12, 7, "binary")
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(),
        label=torch.from_numpy(y).cuda(),
        max_bin=128,
    )
    bst = run_training(
        {"objective": "binary:logistic", "grow_policy": "lossguide",
         "max_leaves": 64, "max_depth": 0, "eta": 0.3,
         "subsample": 0.8, "colsample_bytree": 0.8, "seed": 5},
        dm, 8,
    )
    assert len(bst.trees) == 8
    p = bst.predict(X[:1000])
    assert np.isfinite(p).all() and 0 < p.mean() < 1
    # determinism: same seed, same trees
    bst2 = run_training(
        {"objective": "binary:logistic", "grow_policy": "lossguide",
         "max_leaves": 64, "max_depth": 0, "eta": 0.3,
         "subsample": 0.8, "colsample_bytree": 0.8, "seed": 5},
        dm, 8,
    )
    for ta, tb in zip(bst.trees, bst2.trees):
        assert np.array_equal(ta.thr, tb.thr)
        assert np.array_equal(ta.value, tb.value)


@pytest.mark.gpu
def test_gblinear_gpu():
    """gblinear on device tensors (chunked fp64 GEMV path)."""
    import torch
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    rng = np.random.RandomState(0)
    X = rng.randn(200000, 6).astype(np.float32)
    w_true = np.array([2.0, -1.0, 0.5, 0, 0, 3.0], np.float32)
    y = (X @ w_true + 0.7).astype(np.float32)
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(), label=torch.from_numpy(y).cuda(),
        max_bin=64, keep_raw=True,
    )
    bst = run_training(
        {"objective": "reg:squarederror", "booster": "gblinear",
         "eta": 0.5, "lambda": 0.0, "base_score": 0.0},
        dm, 40,
    )
    w = bst.linear_weights[:, 0]
    assert np.abs(w[:6] - w_true).max() < 0.05


@pytest.mark.gpu
def test_dart_gpu_matches_cpu():
    """DART on GPU grows the same trees as CPU (dropout replay path)."""
    import torch
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    X, y = create_data(100000, 8, 11, "reg")
    params = {"objective": "reg:squarederror", "booster": "dart",
              "rate_drop": 0.3, "max_depth": 4, "eta": 0.3, "seed": 2}
    dm_g = BinnedMatrix.build(
        torch.from_numpy(X).cuda(), label=torch.from_numpy(y).cuda(),
        max_bin=64,
    )
    bst_g = run_training(dict(params), dm_g, 10)
    dm_c = BinnedMatrix.build(
        torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
    )
    bst_c = run_training(dict(params), dm_c, 10)
    for tg, tc in zip(bst_g.trees, bst_c.trees):
        assert np.array_equal(tg.feat, tc.feat)
        assert np.array_equal(tg.thr, tc.thr)


@pytest.mark.gpu
def test_extended_objectives_gpu():
    """Poisson / tweedie / AFT / cox / hinge run on device tensors."""
    import torch
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    rng = np.random.RandomState(0)
    n = 100000
    X = rng.rand(n, 4).astype(np.float32)
    lam = np.exp(1.0 + 2 * X[:, 0])
    Xg = torch.from_numpy(X).cuda()

    yp = rng.poisson(lam).astype(np.float32)
    dmp = BinnedMatrix.build(
        Xg, label=torch.from_numpy(yp).cuda(), max_bin=64
    )
    # max_delta_step=0.7 (poisson default) caps early steps: needs ~20
    # rounds to converge (same on CPU)
    bp = run_training(
        {"objective": "count:poisson", "max_depth": 4, "eta": 0.3}, dmp, 20
    )
    assert np.corrcoef(bp.predict(X[:5000]), lam[:5000])[0, 1] > 0.9

    bt = run_training(
        {"objective": "reg:tweedie", "max_depth": 4, "eta": 0.3}, dmp, 15
    )
    assert np.corrcoef(bt.predict(X[:5000]), lam[:5000])[0, 1] > 0.9

    t = np.exp(1.0 + 2 * X[:, 0] + 0.3 * rng.randn(n)).astype(np.float32)
    right = rng.rand(n) < 0.3
    yl = t.copy()
    yu = np.where(right, np.inf, t).astype(np.float32)
    dma = BinnedMatrix.build(Xg, max_bin=64)
    dma.label_lower_bound = torch.from_numpy(yl).cuda()
    dma.label_upper_bound = torch.from_numpy(yu).cuda()
    ba = run_training(
        {"objective": "survival:aft", "max_depth": 4, "eta": 0.3,
         "base_score": 1.0}, dma, 10,
    )
    pr = ba.predict(X[:5000], output_margin=True)
    assert np.corrcoef(pr, 1 + 2 * X[:5000, 0])[0, 1] > 0.9

    risk = 2.0 * X[:, 0]
    tc = rng.exponential(1.0 / np.exp(risk)).astype(np.float32)
    ev = rng.rand(n) < 0.7
    lab = np.where(ev, tc, -tc).astype(np.float32)
    lab[lab == 0] = 1e-6
    dmc = BinnedMatrix.build(
        Xg, label=torch.from_numpy(lab).cuda(), max_bin=64
    )
    bc = run_training(
        {"objective": "survival:cox", "max_depth": 4, "eta": 0.3,
         "base_score": 1.0}, dmc, 10,
    )
    pc = bc.predict(X[:5000], output_margin=True)
    assert np.corrcoef(pc, risk[:5000])[0, 1] > 0.85


def _train_higgs_like(num_actors, rounds=8, env=None):
    import os

    # 60k rows: every shard AND the single-rank whole matrix stay under
    # the sketch's exact-summary limit (quantile.py _EXACT_LIMIT), where
    # merged cuts are exactly the global quantiles for every world size.
    # Above it the sketch is approximate and (like stock XGBoost's)
    # distributed cuts differ slightly from single-machine cuts, so
    # bitwise model equality is only contractual in the exact regime.
    X, y = create_data(60_000, 10)
    old = {}
    for k, v in (env or {}).items():
        old[k] = os.environ.get(k)
        os.environ[k] = v
    try:
        bst = train(
            {"objective": "binary:logistic", "tree_method": "gpu_hist",
             "max_depth": 6, "eta": 0.3},
            RayDMatrix(X, label=y),
            rounds,
            ray_params=RayParams(num_actors=num_actors, gpus_per_actor=1,
                                 max_actor_restarts=0),
        )
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v
    return bst, X


def test_multirank_gloo_staged_one_gpu():
    """World-size-2 training with BOTH ranks on this MI355X: GPU compute,
    host-staged gloo transport (RXGB_COLL_BACKEND=gloo) through the SAME
    chunked/overlapped per-depth histogram allreduce path
    (trainer.py _build_depth_histograms). The model must be bitwise
    identical to single-actor training (distributed==single invariant;
    reference's core collective semantics, reference README.md:345-349)."""
    bst1, X = _train_higgs_like(1)
    bst2, _ = _train_higgs_like(2, env={"RXGB_COLL_BACKEND": "gloo"})
    np.testing.assert_array_equal(
        bst1.predict(X, output_margin=True),
        bst2.predict(X, output_margin=True),
    )


def test_multirank_rccl_one_gpu():
    """Two RCCL ranks sharing one MI355X. RCCL may refuse duplicate
    devices in a communicator; skip (with the refusal recorded) where it
    does — the gloo-staged test above covers the trainer path either
    way, and the torchrun bench path covers 1-GPU-per-rank RCCL."""
    bst1, X = _train_higgs_like(1)
    try:
        bst2, _ = _train_higgs_like(2)
    except Exception as e:  # noqa: BLE001
        msg = repr(e)
        if any(s in msg.lower() for s in
               ("duplicate", "invalid", "nccl", "rccl", "actor")):
            pytest.skip(f"RCCL refused 2 ranks on one GPU: {msg[:200]}")
        raise
    np.testing.assert_array_equal(
        bst1.predict(X, output_margin=True),
        bst2.predict(X, output_margin=True),
    )


def test_sklearn_estimator_gpu():
    """RayXGBClassifier with tree_method=gpu_hist through the actor
    layer (reference sklearn GPU path)."""
    X, y = create_data(150_000, 8)
    clf_kwargs = dict(n_estimators=8, max_depth=5,
                      tree_method="gpu_hist")
    from xgboost_ray_amd import RayXGBClassifier

    clf = RayXGBClassifier(**clf_kwargs)
    clf.fit(X, y, ray_params=RayParams(num_actors=1, gpus_per_actor=1))
    acc = (clf.predict(
        X, ray_params=RayParams(num_actors=1, gpus_per_actor=1)) == y
    ).mean()
    assert acc > 0.9
