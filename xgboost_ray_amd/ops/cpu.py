"""Pure-PyTorch CPU reference implementations of the engine ops.

These are the numerics oracle: the HIP kernels in ``csrc/`` are tested for
bitwise (integer ops) / close (float ops) agreement against these functions.
Integer fixed-point histogram accumulation makes the CPU and GPU training
paths produce identical trees.
"""

import torch

MISSING_BIN = 255


def quantize_gpair(gpair: torch.Tensor, scale_g: float, scale_h: float) -> torch.Tensor:
    # int32 packed pairs: |q| <= 2^30 by scale construction; accumulators
    # are int64 so any sum is exact
    out = torch.empty(gpair.shape, dtype=torch.int32, device=gpair.device)
    out[:, 0] = torch.round(gpair[:, 0].double() * scale_g).to(torch.int32)
    out[:, 1] = torch.round(gpair[:, 1].double() * scale_h).to(torch.int32)
    return out


def bin_matrix(values: torch.Tensor, cuts_flat: torch.Tensor, cut_ptr: torch.Tensor):
    n, F = values.shape
    out = torch.empty((n, F), dtype=torch.uint8, device=values.device)
    for f in range(F):
        lo, hi = int(cut_ptr[f]), int(cut_ptr[f + 1])
        nb = hi - lo
        col = values[:, f]
        if nb == 0:
            out[:, f] = torch.where(
                torch.isnan(col),
                torch.tensor(MISSING_BIN, dtype=torch.uint8),
                torch.tensor(0, dtype=torch.uint8),
            )
            continue
        cuts = cuts_flat[lo:hi].contiguous()
        # XGBoost HistogramCuts::SearchBin semantics: bin = count of cuts
        # <= value (upper_bound), clamped to nb - 1. Bin b covers
        # [cut[b-1], cut[b]), so a split after bin b <=> "v < cut[b] goes
        # left" - exactly the float threshold rule used at predict time.
        b = torch.searchsorted(cuts, col.contiguous(), side="right")
        b = torch.clamp(b, max=nb - 1).to(torch.uint8)
        b = torch.where(torch.isnan(col), torch.full_like(b, MISSING_BIN), b)
        out[:, f] = b
    return out


def build_histogram(bins, gpair_q, ridx, starts, counts, n_bins,
                    f_range=None, out=None, pregathered=False):
    K = len(starts)
    n, F = bins.shape
    hist = out
    if hist is None:
        hist = torch.zeros(
            (K, F, n_bins, 2), dtype=torch.int64, device=bins.device
        )
    f_lo, f_hi = (0, F) if f_range is None else f_range
    return _build_histogram_range(
        bins, gpair_q, ridx, starts, counts, n_bins, f_lo, f_hi, hist,
        pregathered
    )


def _build_histogram_range(bins, gpair_q, ridx, starts, counts, n_bins,
                           f_lo, f_hi, hist, pregathered=False):
    K = len(starts)
    n, F = bins.shape
    foff = torch.arange(f_lo, f_hi, dtype=torch.int64) * n_bins
    for k in range(K):
        s, c = int(starts[k]), int(counts[k])
        if c == 0:
            continue
        idx = ridx[s : s + c].long()
        rb = bins[idx][:, f_lo:f_hi].long()  # [c, fr]
        valid = rb != MISSING_BIN
        flat = foff.unsqueeze(0) + rb  # [c, F]
        gsrc = gpair_q[s : s + c] if pregathered else gpair_q[idx]
        g = gsrc[:, 0].long().unsqueeze(1).expand_as(flat)
        h = gsrc[:, 1].long().unsqueeze(1).expand_as(flat)
        hk = hist[k].view(F * n_bins, 2)
        fidx = flat[valid]
        hk[:, 0].scatter_add_(0, fidx, g[valid])
        hk[:, 1].scatter_add_(0, fidx, h[valid])
    return hist


def _calc_score(G, H, reg_lambda, reg_alpha):
    """XGBoost split score: ThresholdL1(G, alpha)^2 / (H + lambda)."""
    if reg_alpha > 0:
        G = torch.sign(G) * torch.clamp(G.abs() - reg_alpha, min=0.0)
    denom = H + reg_lambda
    return torch.where(denom > 0, G * G / denom, torch.zeros_like(G))


def _calc_weight_t(G, H, reg_lambda, reg_alpha):
    if reg_alpha > 0:
        G = torch.sign(G) * torch.clamp(G.abs() - reg_alpha, min=0.0)
    denom = H + reg_lambda
    return torch.where(denom > 0, -G / denom, torch.zeros_like(G))


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
):
    K, F, B, _ = hist.shape
    dev = hist.device
    # Dequantize to double for the scan: scan math must be identical on CPU
    # and GPU, and the quantized ints are exactly representable in f64.
    # Multiply by the (identically computed) reciprocal instead of dividing:
    # f64 division is ~10x the cost of multiply on CDNA4 and the GPU kernel
    # mirrors this exact op order.
    inv_g = 1.0 / scale_g
    inv_h = 1.0 / scale_h
    gq = hist[..., 0]
    hq = hist[..., 1]

    # Left-sum prefixes are computed in INT64 (exact under any summation
    # order) and dequantized once per bin: every bin's gain is then
    # independent of its neighbours, which is what lets the GPU scan run
    # one 64-lane wave per (node, feature) with a parallel int wave-scan
    # and still be bitwise-identical to this oracle.
    GLq = torch.cumsum(gq, dim=2)
    HLq = torch.cumsum(hq, dim=2)
    GL = GLq.double() * inv_g
    HL = HLq.double() * inv_h
    Gp = (parent_g.double() * inv_g).view(K, 1, 1)
    Hp = (parent_h.double() * inv_h).view(K, 1, 1)
    # Missing mass per (node, feature) = parent - feature total.
    Gmiss = Gp - GL[:, :, -1:].clone()
    Hmiss = Hp - HL[:, :, -1:].clone()

    parent_score = _calc_score(Gp, Hp, reg_lambda, reg_alpha)

    # Candidate split after bin b (left = bins <= b); last bin excluded.
    bin_idx = torch.arange(B, device=dev).view(1, 1, B)
    valid_bin = bin_idx < (feat_bins.to(dev).view(1, F, 1) - 1)

    best = {
        "gain": torch.full((K,), -1.0, dtype=torch.float64, device=dev),
        "feature": torch.full((K,), -1, dtype=torch.int32, device=dev),
        "bin": torch.zeros((K,), dtype=torch.int32, device=dev),
        "default_left": torch.zeros((K,), dtype=torch.uint8, device=dev),
        "left_g": torch.zeros((K,), dtype=torch.int64, device=dev),
        "left_h": torch.zeros((K,), dtype=torch.int64, device=dev),
    }

    for default_left in (1, 0):
        if default_left:
            gl, hl = GL + Gmiss, HL + Hmiss
        else:
            gl, hl = GL, HL
        gr, hr = Gp - gl, Hp - hl
        ok = (
            valid_bin
            & (hl >= min_child_weight)
            & (hr >= min_child_weight)
        )
        if allowed is not None:
            # interaction constraints: per-(node, feature) gate
            ok = ok & (allowed.to(dev) != 0).view(K, F, 1)
        if monotone is not None:
            # monotone constraints: clamp child weights into the node's
            # bound interval, then require the constrained ordering
            c = monotone.to(dev).view(1, F, 1)
            lo = bounds[:, 0].to(dev).view(K, 1, 1)
            up = bounds[:, 1].to(dev).view(K, 1, 1)
            wl = torch.clamp(
                _calc_weight_t(gl, hl, reg_lambda, reg_alpha), lo, up
            )
            wr = torch.clamp(
                _calc_weight_t(gr, hr, reg_lambda, reg_alpha), lo, up
            )
            ok = ok & torch.where(
                c > 0, wl <= wr, torch.where(c < 0, wl >= wr,
                                             torch.ones_like(ok)),
            )
        gain = (
            0.5
            * (
                _calc_score(gl, hl, reg_lambda, reg_alpha)
                + _calc_score(gr, hr, reg_lambda, reg_alpha)
                - parent_score
            )
            - gamma
        )
        gain = torch.where(ok, gain, torch.full_like(gain, -float("inf")))
        flat = gain.view(K, F * B)
        mx, arg = flat.max(dim=1)
        upd = mx > best["gain"]
        f_sel = (arg // B).to(torch.int32)
        b_sel = (arg % B).to(torch.int32)
        best["gain"] = torch.where(upd, mx, best["gain"])
        best["feature"] = torch.where(upd, f_sel, best["feature"])
        best["bin"] = torch.where(upd, b_sel, best["bin"])
        best["default_left"] = torch.where(
            upd,
            torch.full((K,), default_left, dtype=torch.uint8, device=dev),
            best["default_left"],
        )
        karange = torch.arange(K, device=dev)
        lgq = GLq[karange, f_sel.long(), b_sel.long()]
        lhq = HLq[karange, f_sel.long(), b_sel.long()]
        if default_left:
            # left child also receives the missing mass (exact in quantized
            # space: parent_q - feature_total_q).
            lgq = lgq + (parent_g - GLq[:, :, -1][karange, f_sel.long()])
            lhq = lhq + (parent_h - HLq[:, :, -1][karange, f_sel.long()])
        best["left_g"] = torch.where(upd, lgq, best["left_g"])
        best["left_h"] = torch.where(upd, lhq, best["left_h"])

    # contract: outputs are host numpy arrays (the driver consumes them on
    # the host; returning numpy makes the GPU path's single packed D2H
    # transfer natural)
    return {
        "gain": best["gain"].float().cpu().numpy(),
        "feature": best["feature"].cpu().numpy(),
        "bin": best["bin"].cpu().numpy(),
        "default_left": best["default_left"].cpu().numpy(),
        "left_g": best["left_g"].cpu().numpy(),
        "left_h": best["left_h"].cpu().numpy(),
    }


def partition_rows(bins, ridx, starts, counts, split_feat, split_bin,
                   default_left, gpair_seg=None, bins_t=None):
    K = len(starts)
    out = ridx.clone()
    gout = None if gpair_seg is None else gpair_seg.clone()
    left_counts = torch.zeros(K, dtype=torch.int64)
    for k in range(K):
        s, c = int(starts[k]), int(counts[k])
        if c == 0:
            continue
        idx = ridx[s : s + c]
        fv = bins[idx.long(), int(split_feat[k])]
        miss = fv == MISSING_BIN
        go_left = fv <= int(split_bin[k])
        if int(default_left[k]):
            go_left = go_left | miss
        else:
            go_left = go_left & ~miss
        out[s : s + c] = torch.cat([idx[go_left], idx[~go_left]])
        if gout is not None:
            gseg = gpair_seg[s : s + c]
            gout[s : s + c] = torch.cat([gseg[go_left], gseg[~go_left]])
        left_counts[k] = int(go_left.sum())
    if gout is None:
        return out, left_counts
    return out, left_counts, gout


def predict_trees(X, feat, thr, left, default_left, value, tree_ptr, out, tree_weight=1.0):
    n = X.shape[0]
    T = len(tree_ptr) - 1
    rows = torch.arange(n)
    for t in range(T):
        base = int(tree_ptr[t])
        cur = torch.full((n,), base, dtype=torch.int64)
        while True:
            f = feat[cur].long()
            is_leaf = f < 0
            if bool(is_leaf.all()):
                break
            # full-width level step with self-looping leaves: the
            # boolean-mask variant paid a nonzero + gather + clone per
            # level, several times this version's cost
            x = X[rows, f.clamp(min=0)]
            go_left = torch.where(
                torch.isnan(x), default_left[cur].bool(), x < thr[cur]
            )
            la = left[cur].long()
            nxt = torch.where(go_left, la, la + 1) + base
            cur = torch.where(is_leaf, cur, nxt)
        out += value[cur] * tree_weight
    return out


def update_margins(margin, ridx, starts, counts, leaf_values):
    for k in range(len(starts)):
        s, c = int(starts[k]), int(counts[k])
        if c:
            margin[ridx[s : s + c].long()] += float(leaf_values[k])
    return margin
