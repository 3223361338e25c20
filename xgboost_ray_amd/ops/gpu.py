"""GPU op implementations backed by the hand-written CDNA4 HIP extension.

The extension (``xgboost_ray_amd/csrc``) is compiled in-tree for gfx950
(``__graft_entry__.build()`` / ``python setup.py build_ext --inplace``).
On a GPU machine it is REQUIRED - there is no silent eager fallback for
CUDA tensors: if the .so is missing, ops raise immediately.
"""

import os

import torch

_ext = None
_load_error = None


def _load():
    global _ext, _load_error
    if _ext is not None:
        return _ext
    try:
        import importlib

        _ext = importlib.import_module("xgboost_ray_amd._hip_ops")
    except ImportError as e:
        _load_error = e
        raise RuntimeError(
            "xgboost_ray_amd HIP extension (_hip_ops) is not built. "
            "Run `python -m xgboost_ray_amd.build` or "
            "`python __graft_entry__.py build` to compile it for gfx950. "
            f"Original error: {e}"
        ) from e
    return _ext


def quantize_gpair(gpair, scale_g, scale_h):
    return _load().quantize_gpair(gpair.contiguous(), scale_g, scale_h)


def bin_matrix(values, cuts_flat, cut_ptr):
    return _load().bin_matrix(
        values.contiguous(),
        cuts_flat.to(values.device),
        cut_ptr.to(values.device),
    )


def build_histogram(bins, gpair_q, ridx, starts, counts, n_bins,
                    f_range=None, out=None, pregathered=False):
    K = len(starts)
    F = bins.shape[1]
    if out is None:
        out = torch.zeros(
            (K, F, int(n_bins), 2), dtype=torch.int64, device=bins.device
        )
    f_lo, f_hi = (0, F) if f_range is None else f_range
    return _load().build_histogram(
        bins,
        gpair_q,
        ridx,
        starts,
        counts,
        int(n_bins),
        int(f_lo),
        int(f_hi),
        out,
        bool(pregathered),
    )


def find_splits(
    hist,
    parent_g,
    parent_h,
    feat_bins,
    scale_g,
    scale_h,
    reg_lambda,
    reg_alpha,
    gamma,
    min_child_weight,
    monotone=None,
    bounds=None,
    allowed=None,
    pull=True,
):
    import numpy as np

    dev = hist.device
    if monotone is None:
        monotone = torch.zeros(0, dtype=torch.int8, device=dev)
        bounds = torch.zeros((0, 2), dtype=torch.float64, device=dev)
    if allowed is None:
        allowed = torch.zeros(0, dtype=torch.uint8, device=dev)
    else:
        allowed = allowed.to(dev).to(torch.uint8).contiguous()
    (packed,) = _load().find_splits(
        hist,
        parent_g.to(dev),
        parent_h.to(dev),
        feat_bins.to(dev),
        float(scale_g),
        float(scale_h),
        float(reg_lambda),
        float(reg_alpha),
        float(gamma),
        float(min_child_weight),
        monotone.to(dev).to(torch.int8),
        bounds.to(dev).to(torch.float64),
        allowed,
        pull,
    )
    if not pull:
        return packed  # device [K,6] for the fused single-sync partition
    arr = packed.cpu().numpy()  # ONE D2H for the whole depth's splits
    return {
        "gain": arr[:, 0].astype(np.int32).view(np.float32),
        "feature": arr[:, 1].astype(np.int32),
        "bin": arr[:, 2].astype(np.int32),
        "default_left": arr[:, 3].astype(np.uint8),
        "left_g": arr[:, 4].copy(),
        "left_h": arr[:, 5].copy(),
    }


def partition_rows(bins, ridx, starts, counts, split_feat, split_bin,
                   default_left, gpair_seg=None, bins_t=None):
    dev = bins.device
    if bins_t is None:
        bins_t = torch.zeros(0, dtype=torch.uint8, device=dev)
    if gpair_seg is None:
        gpair_seg = torch.zeros((0, 2), dtype=torch.int32, device=dev)
        r, lc, _ = _load().partition_rows(
            bins, ridx, starts, counts, split_feat, split_bin,
            default_left, gpair_seg, bins_t,
        )
        return r, lc
    return _load().partition_rows(
        bins, ridx, starts, counts, split_feat, split_bin, default_left,
        gpair_seg, bins_t,
    )


def predict_trees(X, feat, thr, left, default_left, value, tree_ptr, out, tree_weight=1.0):
    dev = X.device
    target = out if out.is_contiguous() else out.contiguous()
    _load().predict_trees(
        X.contiguous(),
        feat.to(dev),
        thr.to(dev),
        left.to(dev),
        default_left.to(dev),
        value.to(dev),
        tree_ptr.to(dev),
        target,
        float(tree_weight),
    )
    if target is not out:
        out.copy_(target)
    return out


def lambdarank_grad(margin, label, group_ptr, rank, idcg, use_ndcg):
    return _load().lambdarank_grad(
        margin, label, group_ptr, rank, idcg, bool(use_ndcg)
    )


def update_margins(margin, ridx, starts, counts, leaf_values,
                   ridx_b=None, parity=None):
    # starts/counts/leaf_values stay on the HOST: the extension stages
    # all control data through ONE pinned H2D (a .to(dev) here forced a
    # pointless H2D + synchronizing D2H round-trip per round).
    # ridx_b/parity: ping-pong leaf buffers (parity[k] selects which
    # buffer node k's rows last landed in).
    lv = torch.as_tensor(leaf_values, dtype=torch.float32)
    rb = ridx_b if ridx_b is not None else torch.zeros(
        0, dtype=torch.int32, device=margin.device)
    par = (torch.as_tensor(parity, dtype=torch.int64)
           if parity is not None
           else torch.zeros(0, dtype=torch.int64))
    target = margin if margin.is_contiguous() else margin.contiguous()
    _load().update_margins(target, ridx, starts, counts, lv, rb, par)
    if target is not margin:
        margin.copy_(target)
    return margin


def grad_fused(margin, label, weight, scale_pos_weight, mode):
    w = weight if weight is not None else torch.zeros(
        0, dtype=torch.float32, device=margin.device)
    gpair, absmax = _load().grad_fused(
        margin.contiguous(), label.contiguous().float(),
        w.contiguous().float() if w.numel() else w,
        float(scale_pos_weight), int(mode),
    )
    return gpair, absmax


# partition chunk geometry (PART_THREADS * PART_ROWS_PER_THREAD in
# kernels.hip); the python side needs it only to BOUND the fused
# partition grid
PART_CHUNK = 2048


def partition_rows_from_packed(bins, ridx, starts_ord, counts_ord, packed,
                               gseg, bins_t, chunk_bound, ridx_dest=None):
    """Single-sync partition: consumes find_splits' device packed output
    (plan + count + prefix + scatter all device-planned); returns
    (ridx_out, gseg_out, pull) where `pull` is a pinned i64 [7K] buffer
    [packed rows | left_counts] valid after a stream sync."""
    dev = bins.device
    if bins_t is None:
        bins_t = torch.zeros(0, dtype=torch.uint8, device=dev)
    if gseg is None:
        gseg = torch.zeros((0, 2), dtype=torch.int32, device=dev)
    rd = ridx_dest if ridx_dest is not None else torch.zeros(
        0, dtype=torch.int32, device=bins.device)
    out = _load().partition_rows_from_packed(
        bins, ridx, starts_ord, counts_ord, packed, gseg, bins_t,
        int(chunk_bound), rd,
    )
    # out: [ridx_out, gseg_out, pull, meta, scalars, left_before,
    #       node_left_total, flags, chunk_bound]
    ev = torch.cuda.Event()
    ev.record()  # after plan+count+prefix+pull, BEFORE the scatter
    _load().partition_scatter_from_packed(
        ridx, out[0], gseg, out[1], out[3], out[4], out[5], out[6],
        out[7], out[8],
    )
    return out[0], out[1], out[2], ev


def partition_begin(bins, ridx, starts, counts, split_feat, split_bin,
                    default_left, gpair_seg=None, bins_t=None,
                    gseg_full_rewrite=True):
    """Launch the partition's count+prefix kernels and return a context
    immediately: host bookkeeping between begin and finish overlaps
    them (and the tree writes after finish overlap the scatter)."""
    dev = bins.device
    if bins_t is None:
        bins_t = torch.zeros(0, dtype=torch.uint8, device=dev)
    if gpair_seg is None:
        gpair_seg = torch.zeros((0, 2), dtype=torch.int32, device=dev)
    st = _load().partition_rows_begin(
        bins, ridx, starts, counts, split_feat, split_bin, default_left,
        gpair_seg, bins_t, bool(gseg_full_rewrite),
    )
    return (ridx, gpair_seg, st)


def partition_finish(ctx):
    ridx, gseg, st = ctx
    if len(st) == 6:  # no chunks: nothing moved
        import torch as _t

        return st[0], _t.zeros(
            st[2].numel(), dtype=_t.int64
        ), st[1]
    r, lc, g = _load().partition_rows_finish(
        ridx, st[0], gseg, st[1], st[2], st[3], st[4], st[5], st[6]
    )
    return r, lc, g


def eval_logloss(margin, label, weight=None):
    w = weight if weight is not None else torch.zeros(
        0, dtype=torch.float32, device=margin.device)
    return _load().eval_logloss(
        margin.contiguous(), label.contiguous().float(),
        w.contiguous().float() if w.numel() else w,
    )


def eval_auc_hist(margin, label, n_bins):
    return _load().eval_auc_hist(
        margin.contiguous(), label.contiguous().float(), int(n_bins)
    )
