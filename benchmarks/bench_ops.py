"""Per-op steady-state microbenchmark on the HIGGS shape (GPU).

Times each engine op in isolation (20 reps, synced) to separate kernel
time from host-side overhead. Run on an MI355X box.
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import time

import numpy as np
import torch

from xgboost_ray_amd.engine.quantile import BinnedMatrix
from xgboost_ray_amd import ops


def timeit(name, fn, reps=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps * 1000
    print(f"{name:24s} {dt:9.3f} ms", flush=True)
    return dt


def main():
    n, F = 11_000_000, 28
    dev = torch.device("cuda")
    gen = torch.Generator(device=dev).manual_seed(0)
    X = torch.randn(n, F, device=dev, generator=gen)
    y = (X[:, 0] > 0).float()
    dm = BinnedMatrix.build(X, label=y, max_bin=256)
    del X
    bins = dm.bins
    print("bins stride:", bins.stride(0), "shape", tuple(bins.shape))

    gp = torch.stack([y - 0.5, torch.ones(n, device=dev)], dim=1)
    gq = ops.quantize_gpair(gp, 2.0**28, 2.0**27)
    ridx = torch.randperm(n, device=dev).to(torch.int32)

    # depth-8-like frontier: 128 nodes to build
    K = 128
    cut = torch.sort(
        torch.randint(1, n - 1, (K - 1,), generator=gen, device=dev)
    )[0].cpu()
    starts = torch.cat([torch.tensor([0]), cut]).long()
    ends = torch.cat([cut, torch.tensor([n])]).long()
    counts = ends - starts

    timeit("quantize", lambda: ops.quantize_gpair(gp, 2.0**28, 2.0**27))
    timeit(
        "hist K=1",
        lambda: ops.build_histogram(
            bins, gq, ridx, torch.tensor([0]), torch.tensor([n]),
            dm.cuts.max_bins,
        ),
    )
    timeit(
        "hist K=128",
        lambda: ops.build_histogram(
            bins, gq, ridx, starts, counts, dm.cuts.max_bins
        ),
    )
    hist = ops.build_histogram(bins, gq, ridx, starts, counts, dm.cuts.max_bins)
    pg = hist[:, 0, :, 0].sum(1).contiguous()
    ph = hist[:, 0, :, 1].sum(1).contiguous()
    timeit(
        "find_splits K=128",
        lambda: ops.find_splits(
            hist, pg, ph, dm.cuts.feat_bins().cuda(), 2.0**28, 2.0**27,
            1.0, 0.0, 0.0, 1.0,
        ),
    )
    small_hist = hist[:2].contiguous()
    timeit(
        "find_splits K=2",
        lambda: ops.find_splits(
            small_hist, pg[:2].contiguous(), ph[:2].contiguous(),
            dm.cuts.feat_bins().cuda(), 2.0**28, 2.0**27, 1.0, 0.0, 0.0, 1.0,
        ),
    )
    sf = torch.randint(0, F, (K,), dtype=torch.int32)
    sb = torch.randint(0, 100, (K,), dtype=torch.int32)
    dl = torch.randint(0, 2, (K,), dtype=torch.uint8)
    timeit(
        "partition K=128",
        lambda: ops.partition_rows(bins, ridx, starts, counts, sf, sb, dl),
    )
    lv = np.random.randn(K).astype(np.float32)
    m = torch.zeros(n, device=dev)
    timeit(
        "update_margins",
        lambda: ops.update_margins(m, ridx, starts, counts, lv),
    )
    # torch glue pieces
    timeit("stack 128 hists", lambda: torch.stack([hist[i] for i in range(K)]))
    timeit("root_sum gather", lambda: gq[ridx.long()].sum(dim=0))
    timeit(
        "gradients(sigmoid)",
        lambda: torch.sigmoid(m),
    )

    # serving: packed-node tree-walk over a realistic 100-tree depth-8 model
    from xgboost_ray_amd.engine.trainer import run_training

    small = BinnedMatrix.build(
        torch.randn(500_000, F, device=dev, generator=gen),
        label=(torch.rand(500_000, device=dev, generator=gen) > 0.5).float(),
        max_bin=256,
    )
    bst = run_training(
        {"objective": "binary:logistic", "max_depth": 8, "eta": 0.1}, small, 100
    )
    Xp = torch.randn(2_000_000, F, device=dev, generator=gen)
    flat = bst._flat_trees(dev)
    out = torch.zeros(2_000_000, device=dev)
    from xgboost_ray_amd.ops import gpu as gops

    dt = timeit(
        "predict 2M x 100 trees",
        lambda: gops.predict_trees(
            Xp, flat["feat"], flat["thr"], flat["left"],
            flat["default_left"], flat["value"], flat["tree_ptr"], out,
        ),
        reps=10,
    )
    print(f"serving throughput: {2_000_000 / dt * 1000 / 1e6:.1f} M rows/s "
          f"(100 trees, depth 8)")


if __name__ == "__main__":
    main()
