"""GPU kernel numerics: every HIP op vs the CPU torch oracle.

All comparisons on integer ops are exact; the split scan is engineered to
be bitwise-identical (same FP op order); the float predictors use
tolerances. The final test trains a full model on GPU and CPU and demands
identical trees.
"""

import numpy as np
import pytest
import torch

from tests.utils import create_data
from xgboost_ray_amd.ops import cpu as cpu_ops

pytestmark = pytest.mark.gpu


def _gpu_ops():
    from xgboost_ray_amd.ops import gpu as gpu_ops

    return gpu_ops


@pytest.fixture(scope="module")
def data():
    torch.manual_seed(0)
    rng = np.random.RandomState(0)
    n, F = 200_000, 12
    X = rng.randn(n, F).astype(np.float32)
    X[rng.rand(n, F) < 0.05] = np.nan  # missing values
    y = (np.nan_to_num(X[:, 0]) + 0.5 * np.nan_to_num(X[:, 1]) > 0).astype(
        np.float32
    )
    return X, y


@pytest.fixture(scope="module")
def binned(data):
    from xgboost_ray_amd.engine.quantile import build_cuts

    X, y = data
    Xt = torch.from_numpy(X)
    cuts = build_cuts(Xt, 256)
    bins_cpu = cpu_ops.bin_matrix(Xt, cuts.cuts_flat, cuts.cut_ptr)
    return Xt, torch.from_numpy(y), cuts, bins_cpu


def test_quantize_gpair_exact(data):
    gpu = _gpu_ops()
    X, y = data
    n = len(y)
    gp = torch.stack(
        [torch.from_numpy(y) - 0.3, torch.rand(n) + 0.1], dim=1
    ).float()
    ref = cpu_ops.quantize_gpair(gp, 2.0**28, 2.0**27)
    out = gpu.quantize_gpair(gp.cuda(), 2.0**28, 2.0**27).cpu()
    torch.testing.assert_close(ref, out, rtol=0, atol=0)


def test_bin_matrix_exact(binned):
    gpu = _gpu_ops()
    Xt, yt, cuts, bins_cpu = binned
    out = gpu.bin_matrix(
        Xt.cuda(), cuts.cuts_flat.cuda(), cuts.cut_ptr.cuda()
    ).cpu()
    torch.testing.assert_close(bins_cpu, out, rtol=0, atol=0)


def _segments(n, k, seed=0):
    """Random contiguous partition of [0, n) into k segments."""
    rng = np.random.RandomState(seed)
    cut = np.sort(rng.choice(n - 1, k - 1, replace=False) + 1)
    starts = np.concatenate([[0], cut])
    ends = np.concatenate([cut, [n]])
    return (
        torch.tensor(starts, dtype=torch.int64),
        torch.tensor(ends - starts, dtype=torch.int64),
    )


def test_build_histogram_exact(binned):
    gpu = _gpu_ops()
    Xt, yt, cuts, bins_cpu = binned
    n = Xt.shape[0]
    gp = torch.stack([yt - 0.3, torch.rand(n) + 0.1], dim=1).float()
    gq = cpu_ops.quantize_gpair(gp, 2.0**28, 2.0**27)
    rng = np.random.RandomState(1)
    ridx = torch.from_numpy(rng.permutation(n).astype(np.int32))
    starts, counts = _segments(n, 7)
    ref = cpu_ops.build_histogram(
        bins_cpu, gq, ridx, starts, counts, cuts.max_bins
    )
    out = gpu.build_histogram(
        bins_cpu.cuda(), gq.cuda(), ridx.cuda(), starts, counts, cuts.max_bins
    ).cpu()
    torch.testing.assert_close(ref, out, rtol=0, atol=0)


def test_build_histogram_padded_exact(binned):
    """The engine's padded 16B-stride layout (vectorized kernel path)."""
    gpu = _gpu_ops()
    Xt, yt, cuts, bins_cpu = binned
    n, F = bins_cpu.shape
    F_pad = (F + 15) // 16 * 16
    full = torch.full((n, F_pad), 255, dtype=torch.uint8, device="cuda")
    full[:, :F] = bins_cpu.cuda()
    bins_padded = full[:, :F]
    gp = torch.stack([yt - 0.3, torch.rand(n) + 0.1], dim=1).float()
    gq = cpu_ops.quantize_gpair(gp, 2.0**28, 2.0**27)
    rng = np.random.RandomState(5)
    ridx = torch.from_numpy(rng.permutation(n).astype(np.int32))
    starts, counts = _segments(n, 7, seed=5)
    ref = cpu_ops.build_histogram(
        bins_cpu, gq, ridx, starts, counts, cuts.max_bins
    )
    out = gpu.build_histogram(
        bins_padded, gq.cuda(), ridx.cuda(), starts, counts, cuts.max_bins
    ).cpu()
    torch.testing.assert_close(ref, out, rtol=0, atol=0)
    # partition must also honor the padded stride
    sf = torch.tensor(rng.randint(0, F, 7), dtype=torch.int32)
    sb = torch.tensor(rng.randint(0, 100, 7), dtype=torch.int32)
    dl = torch.tensor(rng.randint(0, 2, 7), dtype=torch.uint8)
    ref_ridx, ref_counts = cpu_ops.partition_rows(
        bins_cpu, ridx, starts, counts, sf, sb, dl
    )
    out_ridx, out_counts = gpu.partition_rows(
        bins_padded, ridx.cuda(), starts, counts, sf, sb, dl
    )
    torch.testing.assert_close(ref_counts, out_counts.cpu(), rtol=0, atol=0)
    torch.testing.assert_close(ref_ridx, out_ridx.cpu(), rtol=0, atol=0)


def test_build_histogram_feature_ranges():
    """Chunked (f_range) builds into one output == single full build."""
    gpu = _gpu_ops()
    torch.manual_seed(3)
    n, F = 50_000, 40
    F_pad = (F + 15) // 16 * 16
    full = torch.full((n, F_pad), 255, dtype=torch.uint8, device="cuda")
    full[:, :F] = torch.randint(
        0, 200, (n, F), dtype=torch.uint8, device="cuda"
    )
    bins_padded = full[:, :F]
    gp = torch.stack(
        [torch.randn(n), torch.rand(n) + 0.1], dim=1
    ).float()
    gq = cpu_ops.quantize_gpair(gp, 2.0**28, 2.0**27).cuda()
    ridx = torch.arange(n, dtype=torch.int32, device="cuda")
    starts, counts = _segments(n, 3)
    ref = gpu.build_histogram(bins_padded, gq, ridx, starts, counts, 256)
    out = torch.zeros_like(ref)
    for f0, f1 in ((0, 16), (16, 32), (32, F)):
        gpu.build_histogram(
            bins_padded, gq, ridx, starts, counts, 256,
            f_range=(f0, f1), out=out,
        )
    torch.testing.assert_close(ref, out, rtol=0, atol=0)


def test_find_splits_bitwise(binned):
    gpu = _gpu_ops()
    Xt, yt, cuts, bins_cpu = binned
    n = Xt.shape[0]
    gp = torch.stack([yt - 0.3, torch.rand(n) + 0.1], dim=1).float()
    gq = cpu_ops.quantize_gpair(gp, 2.0**28, 2.0**27)
    ridx = torch.arange(n, dtype=torch.int32)
    starts, counts = _segments(n, 5)
    hist = cpu_ops.build_histogram(
        bins_cpu, gq, ridx, starts, counts, cuts.max_bins
    )
    pg = torch.stack([gq[s : s + c, 0].sum() for s, c in zip(starts, counts)])
    ph = torch.stack([gq[s : s + c, 1].sum() for s, c in zip(starts, counts)])
    args = (hist, pg, ph, cuts.feat_bins(), 2.0**28, 2.0**27, 1.0, 0.0, 0.0, 1.0)
    ref = cpu_ops.find_splits(*args)
    gargs = (hist.cuda(), pg.cuda(), ph.cuda(), cuts.feat_bins().cuda(),
             2.0**28, 2.0**27, 1.0, 0.0, 0.0, 1.0)
    out = gpu.find_splits(*gargs)
    for key in ("gain", "feature", "bin", "default_left", "left_g", "left_h"):
        np.testing.assert_array_equal(ref[key], out[key], err_msg=key)


def test_partition_rows_exact(binned):
    gpu = _gpu_ops()
    Xt, yt, cuts, bins_cpu = binned
    n = Xt.shape[0]
    rng = np.random.RandomState(2)
    ridx = torch.from_numpy(rng.permutation(n).astype(np.int32))
    starts, counts = _segments(n, 6)
    sf = torch.tensor(rng.randint(0, Xt.shape[1], 6), dtype=torch.int32)
    sb = torch.tensor(rng.randint(0, 100, 6), dtype=torch.int32)
    dl = torch.tensor(rng.randint(0, 2, 6), dtype=torch.uint8)
    ref_ridx, ref_counts = cpu_ops.partition_rows(
        bins_cpu, ridx, starts, counts, sf, sb, dl
    )
    out_ridx, out_counts = gpu.partition_rows(
        bins_cpu.cuda(), ridx.cuda(), starts, counts, sf, sb, dl
    )
    torch.testing.assert_close(ref_counts, out_counts.cpu(), rtol=0, atol=0)
    torch.testing.assert_close(ref_ridx, out_ridx.cpu(), rtol=0, atol=0)


def test_update_margins_exact(binned):
    gpu = _gpu_ops()
    Xt, yt, cuts, bins_cpu = binned
    n = Xt.shape[0]
    ridx = torch.from_numpy(
        np.random.RandomState(3).permutation(n).astype(np.int32)
    )
    starts, counts = _segments(n, 9)
    vals = np.random.RandomState(4).randn(9).astype(np.float32)
    m_ref = torch.zeros(n)
    cpu_ops.update_margins(m_ref, ridx, starts, counts, vals)
    m_gpu = torch.zeros(n).cuda()
    gpu.update_margins(m_gpu, ridx.cuda(), starts, counts, vals)
    torch.testing.assert_close(m_ref, m_gpu.cpu(), rtol=0, atol=0)


def test_predict_trees_close(binned):
    gpu = _gpu_ops()
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    Xt, yt, cuts, bins_cpu = binned
    dm = BinnedMatrix.build(Xt[:50000], label=yt[:50000], max_bin=64)
    bst = run_training(
        {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3}, dm, 5
    )
    Xs = Xt[:10000]
    # binary:logistic with base_score 0.5 => base margin 0, so the raw
    # tree-walk sum IS the margin
    ref = bst.predict(Xs.numpy(), output_margin=True)
    flat = bst._flat_trees("cuda")
    out = torch.zeros(len(Xs)).cuda()
    gpu.predict_trees(
        Xs.cuda(), flat["feat"], flat["thr"], flat["left"],
        flat["default_left"], flat["value"], flat["tree_ptr"], out,
    )
    np.testing.assert_allclose(out.cpu().numpy(), ref, rtol=1e-5, atol=1e-5)


@pytest.mark.gpu
def test_lambdarank_gpu_matches_cpu():
    from tests.utils import create_labeled_sorted_rank_data
    from xgboost_ray_amd.engine.objectives import RankNDCG, RankPairwise

    X, y, qid = create_labeled_sorted_rank_data(n_groups=40, group_size=50)
    torch.manual_seed(0)
    margin = torch.randn(len(y))
    label = torch.from_numpy(y)
    qid_t = torch.from_numpy(qid)
    for obj_cls in (RankPairwise, RankNDCG):
        obj = obj_cls()
        ref = obj.gradients(margin, label, None, qid_t)
        out = obj.gradients(
            margin.cuda(), label.cuda(), None, qid_t.cuda()
        ).cpu()
        torch.testing.assert_close(ref, out, rtol=2e-4, atol=2e-5)


@pytest.mark.gpu
def test_rank_training_gpu():
    from tests.utils import create_labeled_sorted_rank_data
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import EvalPack, run_training

    X, y, qid = create_labeled_sorted_rank_data(n_groups=200, group_size=40)
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(),
        label=torch.from_numpy(y).cuda(),
        qid=torch.from_numpy(qid).cuda(),
        max_bin=64,
    )
    res = {}
    run_training(
        {"objective": "rank:ndcg", "max_depth": 4, "eta": 0.3,
         "eval_metric": ["ndcg"]},
        dm, 10, evals=[EvalPack(name="train", X=None)], evals_result=res,
    )
    assert res["train"]["ndcg"][-1] > res["train"]["ndcg"][0]


def test_full_training_gpu_equals_cpu(binned):
    """Integration check: CPU- and GPU-trained models agree to float32
    rounding. (Bitwise equality is impossible across devices: torch's
    CPU and ROCm sigmoid differ by 1 ulp, which feeds the gradients.
    The bitwise contracts that DO hold - op-level equality, GPU
    determinism, world-size invariance - are covered above.)"""
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    Xt, yt, cuts, _ = binned
    X_small, y_small = Xt[:100000], yt[:100000]
    params = {"objective": "binary:logistic", "max_depth": 6, "eta": 0.3,
              "eval_metric": ["logloss", "auc"]}
    preds = {}
    for dev in ("cpu", "cuda"):
        Xd = X_small.to(dev)
        dm = BinnedMatrix.build(Xd, label=y_small.to(dev), max_bin=256)
        bst = run_training(params, dm, 5)
        preds[dev] = bst.predict(X_small[:20000].numpy(), output_margin=True)
        del dm
    np.testing.assert_allclose(preds["cpu"], preds["cuda"], rtol=1e-4, atol=1e-5)


@pytest.mark.gpu
def test_full_training_gpu_deterministic(binned):
    """Two identical GPU runs -> bitwise-identical models (atomic order
    independence of the int64 histograms)."""
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    Xt, yt, cuts, _ = binned
    Xd, yd = Xt[:100000].cuda(), yt[:100000].cuda()
    params = {"objective": "binary:logistic", "max_depth": 6, "eta": 0.3}
    outs = []
    for _ in range(2):
        dm = BinnedMatrix.build(Xd, label=yd, max_bin=256)
        bst = run_training(params, dm, 5)
        outs.append(bst.predict(Xt[:20000].numpy(), output_margin=True))
        del dm
    np.testing.assert_array_equal(outs[0], outs[1])


@pytest.mark.gpu
def test_monotone_training_gpu():
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    rng = np.random.RandomState(0)
    n = 100_000
    X = rng.rand(n, 3).astype(np.float32)
    y = (np.sin(X[:, 0] * 6) + X[:, 1] + 0.1 * rng.randn(n)).astype(np.float32)
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(), label=torch.from_numpy(y).cuda(),
        max_bin=128,
    )
    bst = run_training(
        {"objective": "reg:squarederror", "max_depth": 6, "eta": 0.3,
         "monotone_constraints": "(1,0,0)"},
        dm, 20,
    )
    grid = np.zeros((100, 3), dtype=np.float32)
    grid[:, 0] = np.linspace(0, 1, 100)
    grid[:, 1] = 0.5
    grid[:, 2] = 0.5
    pred = bst.predict(grid, output_margin=True)
    assert (np.diff(pred) >= -1e-6).all()


def test_build_histogram_direct_small_nodes(binned, monkeypatch):
    """Tiny-node direct-to-global path must produce the EXACT histogram
    the LDS route produces (int64 atomics, any order). Many small
    segments force every node under the direct threshold."""
    gpu = _gpu_ops()
    Xt, yt, cuts, bins_cpu = binned
    n = Xt.shape[0]
    gp = torch.stack([yt - 0.3, torch.rand(n) + 0.1], dim=1).float()
    gq = cpu_ops.quantize_gpair(gp, 2.0**28, 2.0**27)
    rng = np.random.RandomState(3)
    ridx = torch.from_numpy(rng.permutation(n).astype(np.int32))
    # 512 random segments (~390 rows avg): most take the direct path,
    # the larger tail stays on the LDS route - both compose into one hist
    starts, counts = _segments(n, 512)
    ref = cpu_ops.build_histogram(
        bins_cpu, gq, ridx, starts, counts, cuts.max_bins
    )
    monkeypatch.setenv("RXGB_HIST_DIRECT_ROWS", "512")
    out = gpu.build_histogram(
        bins_cpu.cuda(), gq.cuda(), ridx.cuda(), starts, counts,
        cuts.max_bins,
    ).cpu()
    torch.testing.assert_close(ref, out, rtol=0, atol=0)
    # and disabled (LDS route) agrees too
    monkeypatch.setenv("RXGB_HIST_DIRECT_ROWS", "0")
    out2 = gpu.build_histogram(
        bins_cpu.cuda(), gq.cuda(), ridx.cuda(), starts, counts,
        cuts.max_bins,
    ).cpu()
    torch.testing.assert_close(ref, out2, rtol=0, atol=0)


def test_deep_tree_direct_ab_bitwise():
    """Depth-12 GPU training with the small-node direct path on vs off:
    identical models (the path only changes atomics' destination)."""
    import os

    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    X, y = create_data(300_000, 12, seed=5)
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(), label=torch.from_numpy(y).cuda(),
        max_bin=256,
    )
    preds = {}
    for thresh in ("0", "512"):
        os.environ["RXGB_HIST_DIRECT_ROWS"] = thresh
        try:
            bst = run_training(
                {"objective": "binary:logistic", "max_depth": 12,
                 "eta": 0.3, "tree_method": "gpu_hist"},
                dm, 4,
            )
            preds[thresh] = bst.predict(X, output_margin=True)
        finally:
            os.environ.pop("RXGB_HIST_DIRECT_ROWS", None)
    np.testing.assert_array_equal(preds["0"], preds["512"])


@pytest.mark.parametrize("mode,spw,weighted", [
    (1, 1.0, False), (1, 3.0, True), (0, 1.0, False), (0, 1.0, True),
])
def test_grad_fused_matches_torch_composition(mode, spw, weighted):
    """The fused gradient kernel must be BITWISE equal to the torch GPU
    composition it replaces (same f32 op order), and its absmax must
    equal abs().max()."""
    import os

    from xgboost_ray_amd import ops
    from xgboost_ray_amd.engine.objectives import Logistic, SquaredError

    torch.manual_seed(0)
    n = 1_000_003  # odd size: exercises the grid-stride tail
    margin = (torch.randn(n, device="cuda") * 4).float()
    label = (torch.rand(n, device="cuda") < 0.4).float()
    weight = (
        (torch.rand(n, device="cuda") * 2 + 0.1).float()
        if weighted else None
    )
    obj = Logistic(spw) if mode == 1 else SquaredError()
    fused = ops.grad_fused(margin, label, weight, spw, mode)
    assert fused is not None
    gp_f, mx = fused
    os.environ["RXGB_FUSED_GRAD"] = "0"
    try:
        gp_t = obj.gradients(margin, label, weight)
    finally:
        os.environ.pop("RXGB_FUSED_GRAD", None)
    torch.testing.assert_close(gp_f, gp_t, rtol=0, atol=0)
    torch.testing.assert_close(
        mx.cpu(),
        torch.stack([gp_t[:, 0].abs().max(), gp_t[:, 1].abs().max()]).cpu(),
        rtol=0, atol=0,
    )


def test_predict_lds_matches_plain():
    """Tree-tiled LDS predictor vs the plain walk: identical outputs
    (same per-row accumulation order), including NaN rows and
    deep (over-tile fallback) trees."""
    import os

    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training
    from xgboost_ray_amd.ops import gpu as gops

    X, y = create_data(60_000, 10, seed=2)
    X = X.copy()
    X[::13, 3] = np.nan
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(), label=torch.from_numpy(y).cuda(),
        max_bin=64,
    )
    for depth in (8, 12):  # depth 12 can exceed the 4096-node tile
        bst = run_training(
            {"objective": "binary:logistic", "max_depth": depth,
             "eta": 0.3, "tree_method": "gpu_hist"},
            dm, 6,
        )
        Xp = torch.from_numpy(X[:20_000]).cuda()
        flat = bst._flat_trees(Xp.device)
        outs = {}
        for m in ("0", "1"):
            os.environ["RXGB_PREDICT_LDS"] = m
            try:
                out = torch.zeros(20_000, device="cuda")
                gops.predict_trees(
                    Xp, flat["feat"], flat["thr"], flat["left"],
                    flat["default_left"], flat["value"],
                    flat["tree_ptr"], out,
                )
                outs[m] = out.cpu()
            finally:
                os.environ.pop("RXGB_PREDICT_LDS", None)
        torch.testing.assert_close(outs["0"], outs["1"], rtol=0, atol=0)


def test_one_sync_path_bitwise():
    """Opt-in single-sync depth loop (RXGB_ONE_SYNC=1) vs the default
    2-sync path: identical models (the device plan kernel's split
    predicate must replay exactly on the host)."""
    import os

    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    X, y = create_data(150_000, 10, seed=4)
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(), label=torch.from_numpy(y).cuda(),
        max_bin=256,
    )
    preds = {}
    for m in ("0", "1"):
        os.environ["RXGB_ONE_SYNC"] = m
        try:
            bst = run_training(
                {"objective": "binary:logistic", "max_depth": 9,
                 "eta": 0.3, "tree_method": "gpu_hist",
                 "subsample": 0.8},
                dm, 5,
            )
            preds[m] = bst.predict(X, output_margin=True)
        finally:
            os.environ.pop("RXGB_ONE_SYNC", None)
    np.testing.assert_array_equal(preds["0"], preds["1"])


def test_fused_eval_matches_torch():
    """Fused logloss/AUC eval kernels vs the torch composition: logloss
    to reduction-order tolerance, AUC histograms exactly."""
    import os

    from xgboost_ray_amd.engine.metrics import AUC, AUCPR, LogLoss

    torch.manual_seed(1)
    n = 2_000_003
    margin = (torch.randn(n, device="cuda") * 3).float()
    label = (torch.rand(n, device="cuda") < 0.35).float()
    for metric in (LogLoss(), AUC(), AUCPR()):
        vals = {}
        for m in ("0", "1"):
            os.environ["RXGB_FUSED_EVAL"] = "1" if m == "1" else "0"
            try:
                st = metric.local_stats(margin, label, None, None, None)
                vals[m] = metric.finalize(st.cpu())
            finally:
                os.environ.pop("RXGB_FUSED_EVAL", None)
        assert vals["1"] == pytest.approx(vals["0"], rel=1e-9), metric.name
    # weighted logloss path too
    w = (torch.rand(n, device="cuda") + 0.1).float()
    m = LogLoss()
    a = m.finalize(m.local_stats(margin, label, w, None, None).cpu())
    os.environ["RXGB_FUSED_EVAL"] = "0"
    try:
        b = m.finalize(m.local_stats(margin, label, w, None, None).cpu())
    finally:
        os.environ.pop("RXGB_FUSED_EVAL", None)
    assert a == pytest.approx(b, rel=1e-9)


def test_depth_step_path_bitwise():
    """Opt-in C++ depth_step vs the python-orchestrated fused loop:
    identical models, including monotone constraints (exercises the
    device mono/bounds pointer plumbing)."""
    import os

    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import run_training

    X, y = create_data(150_000, 8, seed=6, kind="reg")
    dm = BinnedMatrix.build(
        torch.from_numpy(X).cuda(), label=torch.from_numpy(y).cuda(),
        max_bin=256,
    )
    for params in (
        {"objective": "reg:squarederror", "max_depth": 8, "eta": 0.3},
        {"objective": "reg:squarederror", "max_depth": 6, "eta": 0.3,
         "monotone_constraints": [1, -1, 0, 0, 0, 0, 0, 0]},
    ):
        preds = {}
        for m in ("0", "1"):
            os.environ["RXGB_DEPTH_STEP"] = m
            try:
                bst = run_training(
                    dict(params, tree_method="gpu_hist"), dm, 5,
                )
                preds[m] = bst.predict(X, output_margin=True)
            finally:
                os.environ.pop("RXGB_DEPTH_STEP", None)
        np.testing.assert_array_equal(preds["0"], preds["1"])
