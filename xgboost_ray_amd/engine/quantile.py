"""Distributed feature quantile sketch + binned (ELLPACK-style) matrix.

MI355X-native equivalent of XGBoost's DMatrix/QuantileDMatrix construction
(C++/CUDA weighted quantile sketch). Per-feature
quantile summaries are computed locally (GPU sort via rocPRIM-backed
torch.sort), merged across workers with ONE allgather per matrix build
(SURVEY.md #2.4 item 1), and every worker derives identical cut points -
a hard correctness requirement: workers with different cuts silently grow
divergent trees.

The binned matrix is stored row-major uint8 [n_rows][n_features] resident
in HBM (11M x 28 HIGGS = 0.3 GB; 1B x 200 = 200 GB across 8 x 288 GB GPUs),
missing values encoded as bin 255, real bins capped at 255 per feature.
"""

from dataclasses import dataclass
from typing import Optional

import numpy as np
import torch

from xgboost_ray_amd import ops
from xgboost_ray_amd.engine.collective import Collective

MISSING_BIN = 255
# local summary size per feature (points sent to the merge); >= 8x max_bin
# keeps merged-quantile error well under one bin.
_SUMMARY_FACTOR = 8
_MAX_SAMPLE = 1 << 21


@dataclass
class HistogramCuts:
    cuts_flat: torch.Tensor  # f32 [sum_f nb_f] ascending per feature
    cut_ptr: torch.Tensor  # i64 [F+1]
    max_bins: int  # histogram stride (max per-feature bin count)

    @property
    def num_features(self) -> int:
        return len(self.cut_ptr) - 1

    def feat_bins(self) -> torch.Tensor:
        return (self.cut_ptr[1:] - self.cut_ptr[:-1]).to(torch.int32)

    def to(self, device):
        return HistogramCuts(
            self.cuts_flat.to(device), self.cut_ptr.to(device), self.max_bins
        )


# below this row count a worker sends its exact per-value histogram, which
# makes the merged cuts EXACTLY the global weighted quantiles - identical
# for every world size / sharding (the distributed-equals-single contract)
_EXACT_LIMIT = 1 << 16


def _local_summary(col: torch.Tensor, n_points: int, seed: int):
    """(values, weights, min, max) summary of one feature column."""
    valid = col[~torch.isnan(col)]
    n = valid.numel()
    if n == 0:
        return None
    if n <= _EXACT_LIMIT:
        uniq, counts = torch.unique(valid, return_counts=True)
        return (
            uniq.cpu().numpy().astype(np.float64),
            counts.cpu().numpy().astype(np.float64),
            float(uniq[0]),
            float(uniq[-1]),
        )
    if n > _MAX_SAMPLE:
        # deterministic strided subsample (stable under re-runs)
        stride = n // _MAX_SAMPLE
        valid = valid[::stride][:_MAX_SAMPLE]
        n_rep = n
        n = valid.numel()
    else:
        n_rep = n
    vs, _ = torch.sort(valid)
    k = min(n_points, n)
    # evenly spaced ranks, inclusive of min and max
    ranks = torch.linspace(0, n - 1, k, device=col.device).round().long()
    pts = vs[ranks]
    w = float(n_rep) / k
    return (
        pts.cpu().numpy().astype(np.float64),
        np.full(k, w, dtype=np.float64),
        float(vs[0]),
        float(vs[-1]),
    )


def _merge_to_cuts(summaries, max_bin: int) -> np.ndarray:
    """Merge per-worker summaries of one feature into <= max_bin cut points."""
    parts = [s for s in summaries if s is not None]
    if not parts:
        return np.zeros(0, dtype=np.float32)
    vals = np.concatenate([p[0] for p in parts])
    wts = np.concatenate([p[1] for p in parts])
    fmax = max(p[3] for p in parts)
    order = np.argsort(vals, kind="stable")
    vals, wts = vals[order], wts[order]
    # collapse duplicates
    uniq, inv = np.unique(vals, return_inverse=True)
    uw = np.zeros(len(uniq))
    np.add.at(uw, inv, wts)
    total = uw.sum()
    nb = max_bin
    if len(uniq) <= nb - 1:
        # few distinct values: one bin boundary after each value
        cuts = uniq[1:].astype(np.float64)  # [v_{i-1}, v_i) boundaries
        cuts = np.concatenate([cuts, [_above(fmax)]])
    else:
        cw = np.cumsum(uw)
        targets = total * (np.arange(1, nb) / nb)
        idx = np.searchsorted(cw, targets, side="left")
        idx = np.clip(idx, 0, len(uniq) - 1)
        cuts = uniq[idx]
        cuts = np.unique(cuts)
        # drop any interior cut <= min value (empty first bin is fine) and
        # ensure the last cut is strictly above the max value
        cuts = cuts[cuts <= fmax]
        cuts = np.concatenate([cuts, [_above(fmax)]])
        cuts = np.unique(cuts)
    return cuts.astype(np.float32)


def _above(x: float) -> float:
    return float(np.nextafter(np.float32(x), np.float32(np.inf)))


def build_cuts(
    X: torch.Tensor,
    max_bin: int,
    collective: Optional[Collective] = None,
    seed: int = 0,
) -> HistogramCuts:
    """Compute per-feature histogram cut points, identical on all workers."""
    max_bin = min(int(max_bin), 255)
    F = X.shape[1]
    n_points = min(_SUMMARY_FACTOR * max_bin, 1 << 14)
    local = [_local_summary(X[:, f], n_points, seed + f) for f in range(F)]
    if collective is not None and collective.is_distributed:
        gathered = collective.allgather_obj(local)
    else:
        gathered = [local]
    cuts_list = []
    for f in range(F):
        cuts_list.append(_merge_to_cuts([g[f] for g in gathered], max_bin))
    ptr = np.zeros(F + 1, dtype=np.int64)
    for f in range(F):
        ptr[f + 1] = ptr[f] + len(cuts_list[f])
    flat = (
        np.concatenate(cuts_list)
        if ptr[-1] > 0
        else np.zeros(0, dtype=np.float32)
    )
    max_bins = int(max((len(c) for c in cuts_list), default=1)) or 1
    return HistogramCuts(
        cuts_flat=torch.from_numpy(flat.astype(np.float32)).to(X.device),
        cut_ptr=torch.from_numpy(ptr).to(X.device),
        max_bins=max_bins,
    )


class SketchAccumulator:
    """Streaming per-feature quantile sketch: per-chunk summaries merged at
    finalize time (and across workers with one allgather). The MI355X
    equivalent of XGBoost's DeviceQuantileDMatrix streaming sketch
    (reference RayDataIter, matrix.py:127-196)."""

    def __init__(self, n_features: int, max_bin: int):
        self.F = n_features
        self.max_bin = min(int(max_bin), 255)
        self.summaries = [[] for _ in range(n_features)]
        self.n_rows = 0

    def push_chunk(self, X: torch.Tensor):
        assert X.shape[1] == self.F
        n_points = min(_SUMMARY_FACTOR * self.max_bin, 1 << 14)
        for f in range(self.F):
            s = _local_summary(X[:, f], n_points, 0)
            if s is not None:
                self.summaries[f].append(s)
        self.n_rows += X.shape[0]

    def finalize(self, collective: Optional[Collective] = None) -> HistogramCuts:
        local = self.summaries
        if collective is not None and collective.is_distributed:
            gathered = collective.allgather_obj(local)
        else:
            gathered = [local]
        cuts_list = []
        for f in range(self.F):
            parts = []
            for worker in gathered:
                parts.extend(worker[f])
            cuts_list.append(_merge_to_cuts(parts, self.max_bin))
        ptr = np.zeros(self.F + 1, dtype=np.int64)
        for f in range(self.F):
            ptr[f + 1] = ptr[f] + len(cuts_list[f])
        flat = (
            np.concatenate(cuts_list)
            if ptr[-1] > 0
            else np.zeros(0, dtype=np.float32)
        )
        max_bins = int(max((len(c) for c in cuts_list), default=1)) or 1
        return HistogramCuts(
            cuts_flat=torch.from_numpy(flat.astype(np.float32)),
            cut_ptr=torch.from_numpy(ptr),
            max_bins=max_bins,
        )


def _make_bins_t(bins: torch.Tensor) -> Optional[torch.Tensor]:
    """Column-major copy of the bin matrix for the partition gather.

    The partition predicate reads ONE feature column per node; row-major
    that costs a full 64 B cache line per row. A [F, n] copy makes it a
    dense near-sequential read (ridx stays sorted within a segment under
    the stable partition). Costs 2x bin memory - cheap against 288 GB
    HBM3E - and is skipped (None) if the allocation does not fit.
    """
    if not bins.is_cuda:
        return None
    try:
        return bins.t().contiguous()
    except torch.cuda.OutOfMemoryError:
        return None


class BinnedMatrix:
    """Quantized training matrix: uint8 bins + labels/weights/margins."""

    def __init__(
        self,
        X: torch.Tensor,
        label: Optional[torch.Tensor],
        weight: Optional[torch.Tensor],
        base_margin: Optional[torch.Tensor],
        qid: Optional[torch.Tensor],
        cuts: HistogramCuts,
    ):
        self.cuts = cuts
        self.bins = ops.bin_matrix(X, cuts.cuts_flat, cuts.cut_ptr)
        self.bins_t = _make_bins_t(self.bins)
        self.raw_X = None  # kept only for the gblinear booster
        self.n_rows, self.n_features = X.shape
        self.label = label
        self.weight = weight
        self.base_margin = base_margin
        self.qid = qid

    @classmethod
    def build(
        cls,
        X: torch.Tensor,
        label=None,
        weight=None,
        base_margin=None,
        qid=None,
        max_bin: int = 256,
        collective: Optional[Collective] = None,
        cuts: Optional[HistogramCuts] = None,
        keep_raw: bool = False,
        seed: int = 0,
    ) -> "BinnedMatrix":
        if cuts is None:
            cuts = build_cuts(X, max_bin, collective, seed)
        obj = cls(X, label, weight, base_margin, qid, cuts)
        if keep_raw:
            obj.raw_X = X.contiguous()
        return obj

    @classmethod
    def build_streaming(
        cls,
        chunk_fn,
        n_features: int,
        device,
        max_bin: int = 256,
        collective: Optional[Collective] = None,
    ) -> "BinnedMatrix":
        """Out-of-core construction: two passes over a chunk iterator.

        ``chunk_fn()`` returns a fresh iterator of shard dicts (numpy
        arrays: data [+label/weight/base_margin/qid]). Pass 1 streams each
        chunk through the device sketch; pass 2 bins chunk-by-chunk into
        the preallocated uint8 matrix - the raw fp32 features are never
        resident in full, so a 1B x 200 shard costs ~25 GB HBM per GPU
        instead of ~800 GB (SURVEY.md #5 long-context note).
        """
        acc = None
        metas = []
        for chunk in chunk_fn():
            Xc = torch.from_numpy(
                np.ascontiguousarray(chunk["data"], dtype=np.float32)
            ).to(device)
            if acc is None:
                acc = SketchAccumulator(Xc.shape[1], max_bin)
            acc.push_chunk(Xc)
            metas.append(Xc.shape[0])
            del Xc
        if acc is None:
            acc = SketchAccumulator(n_features, max_bin)
        cuts = acc.finalize(collective).to(device)
        n = acc.n_rows

        obj = cls.__new__(cls)
        obj.cuts = cuts
        obj.n_rows = n
        obj.n_features = acc.F
        if device.type == "cuda":
            # padded 16B-aligned row stride (vectorized histogram path);
            # pad bytes = 255 (missing) are never accumulated
            F_pad = (acc.F + 15) // 16 * 16
            full = torch.full(
                (n, F_pad), 255, dtype=torch.uint8, device=device
            )
            obj.bins = full[:, : acc.F]
        else:
            obj.bins = torch.empty(
                (n, acc.F), dtype=torch.uint8, device=device
            )
        sides = {"label": [], "weight": [], "base_margin": [], "qid": []}
        pos = 0
        for chunk in chunk_fn():
            Xc = torch.from_numpy(
                np.ascontiguousarray(chunk["data"], dtype=np.float32)
            ).to(device)
            c = Xc.shape[0]
            obj.bins[pos : pos + c] = ops.bin_matrix(
                Xc, cuts.cuts_flat, cuts.cut_ptr
            )
            del Xc
            pos += c
            for key in sides:
                v = chunk.get(key)
                if v is not None:
                    sides[key].append(np.ascontiguousarray(v))
        for key, parts in sides.items():
            if parts:
                arr = np.concatenate(parts)
                t = torch.from_numpy(
                    arr.astype(np.float32) if key != "qid" else arr
                ).to(device)
                setattr(obj, key, t)
            else:
                setattr(obj, key, None)
        obj.bins_t = _make_bins_t(obj.bins)
        obj.raw_X = None
        return obj
