"""Device-dispatched compute primitives for the boosting engine.

Each op has two implementations:

- a pure-PyTorch CPU reference (``ops.cpu``) used for CPU training and as the
  numerics oracle in tests, and
- a hand-written CDNA4 HIP kernel (``ops.gpu`` -> ``xgboost_ray_amd/csrc``)
  used whenever tensors live on a GPU.

On a machine with a GPU the HIP extension is REQUIRED: if a CUDA tensor
reaches an op and the extension is not importable, we raise instead of
silently falling back to slow eager code (this is the MI355X-native compute
path, reference-equivalent of XGBoost's CUDA gpu_hist updater, see
reference main.py:745 / SURVEY.md #2.3).

Histogram determinism: gradients are quantized to int64 fixed point before
histogram accumulation, so per-bin sums are integer adds - associative and
therefore bitwise-identical regardless of atomic ordering, row partition
order, or world size. This is what makes checkpoint-resume and
elastic-restart models deterministic (reference test_fault_tolerance.py
testSameResultWithAndWithoutError semantics).
"""

import torch

from xgboost_ray_amd.ops import cpu as _cpu

MISSING_BIN = 255  # uint8 marker for "value missing" in the binned matrix

_gpu_mod = None
_gpu_checked = False


def _gpu():
    """Load the HIP extension, failing loudly if unavailable."""
    global _gpu_mod, _gpu_checked
    if not _gpu_checked:
        _gpu_checked = True
        from xgboost_ray_amd.ops import gpu as gpu_backend

        _gpu_mod = gpu_backend
    if _gpu_mod is None:  # pragma: no cover
        raise RuntimeError("GPU op requested but HIP extension failed to load")
    return _gpu_mod


def _impl(t: torch.Tensor):
    return _gpu() if t.is_cuda else _cpu


def quantize_gpair(gpair: torch.Tensor, scale_g: float, scale_h: float) -> torch.Tensor:
    """fp32 [n,2] gradient pairs -> int32 [n,2] fixed point (|q| <= 2^30
    by scale construction; int64 accumulators hold any sum)."""
    return _impl(gpair).quantize_gpair(gpair, scale_g, scale_h)


def bin_matrix(values, cuts_flat, cut_ptr):
    """fp32 [n, F] feature matrix -> uint8 [n, F] bin indices.

    bin = number of cut points strictly below the value, clamped to the
    feature's bin count - 1; NaN -> MISSING_BIN.
    """
    return _impl(values).bin_matrix(values, cuts_flat, cut_ptr)


def build_histogram(bins, gpair_q, ridx, starts, counts, n_bins,
                    f_range=None, out=None, pregathered=False):
    """Accumulate per-(node, feature, bin) int64 gradient-pair histograms.

    bins: uint8 [n, F]; gpair_q: int32 [n, 2] packed pairs; ridx: int32 [n]
    row index array partitioned into contiguous per-node segments;
    starts/counts: int64 [K] segment descriptors for the K nodes to build.
    f_range=(f_lo, f_hi) builds only that feature slice (into ``out``,
    which the caller allocates zeroed) so the driver can overlap the
    AllReduce of one block with the build of the next.
    Returns int64 [K, F, n_bins, 2]. Missing values (bin==MISSING_BIN) are
    skipped; their mass is recovered as node_total - feature_sum.
    """
    return _impl(bins).build_histogram(
        bins, gpair_q, ridx, starts, counts, n_bins, f_range, out,
        pregathered
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
    """Best-split scan over histograms.

    hist: int64 [K, F, B, 2]; parent_g/parent_h: int64 [K] quantized node
    sums; feat_bins: int32 [F] per-feature bin counts.
    Returns dict of tensors (on hist.device) each [K]:
      gain f32, feature i32, bin i32, default_left u8,
      left_g/left_h int64 (quantized left-child sums incl. missing if
      default_left).
    """
    kw = {}
    if hist.is_cuda:
        kw["pull"] = pull  # CPU impl always returns host arrays
    return _impl(hist).find_splits(
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
        monotone,
        bounds,
        allowed,
        **kw,
    )


def partition_begin(bins, ridx, starts, counts, split_feat, split_bin,
                    default_left, gpair_seg=None, bins_t=None):
    """Two-phase partition, phase 1 (GPU only): see ops.gpu."""
    from xgboost_ray_amd.ops import gpu

    return gpu.partition_begin(bins, ridx, starts, counts, split_feat,
                               split_bin, default_left, gpair_seg, bins_t)


def partition_finish(ctx):
    """Two-phase partition, phase 2 (GPU only): see ops.gpu."""
    from xgboost_ray_amd.ops import gpu

    return gpu.partition_finish(ctx)


def partition_rows_from_packed(bins, ridx, starts_ord, counts_ord, packed,
                               gseg, bins_t, chunk_bound, ridx_dest=None):
    """Single-sync fused partition (GPU only): see ops.gpu."""
    from xgboost_ray_amd.ops import gpu

    return gpu.partition_rows_from_packed(
        bins, ridx, starts_ord, counts_ord, packed, gseg, bins_t,
        chunk_bound, ridx_dest,
    )


def partition_rows(bins, ridx, starts, counts, split_feat, split_bin,
                   default_left, gpair_seg=None, bins_t=None):
    """Stable partition of each node's ridx segment by its split.

    Rows with bin <= split_bin (or missing & default_left) go left. When
    ``gpair_seg`` (int32 [n,2], segment-ordered gradient pairs) is given
    it is permuted alongside ridx - fusing the next depth's gradient
    gather into the scatter.
    Returns (ridx_out, left_counts[, gpair_seg_out]).
    """
    return _impl(bins).partition_rows(
        bins, ridx, starts, counts, split_feat, split_bin, default_left,
        gpair_seg, bins_t
    )


def predict_trees(
    X, feat, thr, left, default_left, value, tree_ptr, out, tree_weight=1.0
):
    """Accumulate tree-walk predictions for a forest slice into ``out``.

    X: fp32 [n, F] raw features (NaN = missing). Tree arrays are flat SoA
    concatenations with tree_ptr: int32 [T+1] node offsets.
    feat: int32 (-1 => leaf), thr: f32, left: int32 (right = left + 1),
    default_left: u8, value: f32 leaf weight. out: f32 [n] updated in place.
    Decision rule (XGBoost semantics): go left iff x < thr; missing follows
    default_left.
    """
    return _impl(X).predict_trees(
        X, feat, thr, left, default_left, value, tree_ptr, out, tree_weight
    )


def update_margins(margin, ridx, starts, counts, leaf_values,
                   ridx_b=None, parity=None):
    """margin[ridx[seg_k]] += leaf_values[k] for each final-leaf segment.

    ridx_b/parity (GPU): ping-pong leaf buffers - parity[k] selects the
    buffer node k's rows last landed in."""
    if margin.is_cuda:
        from xgboost_ray_amd.ops import gpu

        return gpu.update_margins(
            margin, ridx, starts, counts, leaf_values,
            ridx_b=ridx_b, parity=parity,
        )
    return _impl(margin).update_margins(
        margin, ridx, starts, counts, leaf_values
    )


def grad_fused(margin, label, weight, scale_pos_weight, mode):
    """Fused objective gradient + |g|/|h| max for the hot objectives
    (mode 0 = reg:squarederror, 1 = binary:logistic). GPU only: returns
    (fp32 gpair [n,2], f32 absmax [2]) or None on CPU (the torch
    composition is the CPU path and the numerics oracle)."""
    import os

    if not margin.is_cuda or os.environ.get("RXGB_FUSED_GRAD") == "0":
        return None
    from xgboost_ray_amd.ops import gpu

    return gpu.grad_fused(margin, label, weight, scale_pos_weight, mode)
