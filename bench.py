"""Flagship benchmark: HIGGS-class distributed gpu_hist training.

Measures the BASELINE.json headline metric - boosting rounds/sec (+ AUC)
on an 11M x 28 binary:logistic workload, tree_method gpu_hist, depth 8,
max_bin 256 - on N MI355X GPUs (weak scaling: 11M rows per GPU, synthetic
data of the HIGGS shape, random-init trees; no network for the real CSV).

Single process: `python bench.py --gpus 1 --steps K --warmup W`.
Multi-GPU: launched by the driver as one rank per GPU via
`python -m torch.distributed.run --nproc-per-node N bench.py ...`
(reads RANK/WORLD_SIZE/MASTER_* from the env; backend nccl == RCCL over
xGMI).
"""

import argparse
import json
import os
import time

import numpy as np
import torch


def synth_higgs(n_rows: int, n_features: int, device, seed: int):
    """Synthetic HIGGS-shaped binary task (fp32, learnable signal).

    Generated directly on the target device so 100M x 200-class shapes
    never materialize on the host (80 GB fits in 288 GB HBM, not in RAM).
    """
    gen = torch.Generator(device=device).manual_seed(seed)
    X = torch.empty(n_rows, n_features, dtype=torch.float32, device=device)
    X.normal_(generator=gen)
    w = torch.randn(n_features, generator=gen, device=device) * 0.5
    logits = X @ w + 0.4 * (X[:, 0] * X[:, 1]) - 0.2 * X[:, 2] ** 2
    noise = torch.randn(n_rows, generator=gen, device=device)
    y = (logits + noise > 0).to(torch.float32)
    return X, y


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--rows", type=int, default=11_000_000)
    p.add_argument("--features", type=int, default=28)
    p.add_argument("--max-depth", type=int, default=8)
    p.add_argument("--max-bin", type=int, default=256)
    p.add_argument("--objective", default="binary:logistic",
                   choices=["binary:logistic", "reg:squarederror"])
    p.add_argument("--cpu", action="store_true", help="force CPU (debug)")
    args = p.parse_args()

    import torch.distributed as dist

    from xgboost_ray_amd.engine.collective import Collective
    from xgboost_ray_amd.engine.quantile import BinnedMatrix
    from xgboost_ray_amd.engine.trainer import BoostingEngine, EvalPack

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))

    use_gpu = torch.cuda.is_available() and not args.cpu
    if use_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(
            backend="nccl" if use_gpu else "gloo",
            init_method="env://",
            rank=rank,
            world_size=world,
        )
    coll = Collective(rank=rank, world_size=world, device=device)

    # weak scaling: each rank owns `rows` of its own synthetic shard
    X, y = synth_higgs(args.rows, args.features, device, seed=1234 + rank)
    if args.objective == "reg:squarederror":
        # regression target: the underlying continuous signal
        gen = torch.Generator(device=device).manual_seed(99 + rank)
        y = (X[:, 0] * 2 - X[:, 1] +
             0.1 * torch.randn(args.rows, generator=gen, device=device))
    if use_gpu:
        torch.cuda.synchronize()

    build_t0 = time.perf_counter()
    dm = BinnedMatrix.build(
        X, label=y, max_bin=args.max_bin, collective=coll, seed=0
    )
    del X
    if use_gpu:
        torch.cuda.synchronize()
    build_s = time.perf_counter() - build_t0

    params = {
        "objective": args.objective,
        "tree_method": "gpu_hist" if use_gpu else "hist",
        "max_depth": args.max_depth,
        "max_bin": args.max_bin,
        "eta": 0.1,
    }
    engine = BoostingEngine(params, dm, collective=coll, rank=rank)

    for _ in range(args.warmup):
        engine.update()

    coll.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.update()
    if use_gpu:
        torch.cuda.synchronize()
    coll.barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    el_t = torch.tensor([elapsed], dtype=torch.float64, device=device if use_gpu else "cpu")
    if world > 1:
        coll.allreduce_(el_t, op="max")
    elapsed_max = float(el_t[0])

    # final train AUC (the metric's second half; rmse for regression)
    from xgboost_ray_amd.engine.metrics import get_metric

    quality_name = "auc" if args.objective == "binary:logistic" else "rmse"
    m = get_metric(quality_name)
    st = m.local_stats(engine.margin, dm.label, None, None, None)
    if world > 1:
        st_d = st.to(device) if use_gpu else st
        coll.allreduce_(st_d)
        st = st_d.cpu()
    quality = m.finalize(st.cpu())

    rounds_per_sec = args.steps / elapsed_max
    if rank == 0:
        out = {
            "metric": "boost_rounds_per_sec",
            "value": rounds_per_sec,
            "unit": "rounds/s",
            "n_gpus": world if use_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed_max * 1000.0 / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": (
                    "higgs-11m-x28-binary-logistic-gpu-hist"
                    if args.objective == "binary:logistic"
                    and args.rows == 11_000_000 and args.features == 28
                    else f"synthetic-{args.rows}x{args.features}-"
                         f"{args.objective}-gpu-hist"
                ),
                "global_batch": args.rows * world,
                "seq_len": args.features,
                "parallelism": f"dp{world}",
                "n_rows_per_gpu": args.rows,
                "n_features": args.features,
                "max_depth": args.max_depth,
                "max_bin": args.max_bin,
                f"train_{quality_name}": quality,
                "matrix_build_s": build_s,
                "rows_per_sec": args.rows * world * rounds_per_sec,
            },
        }
        print(json.dumps(out), flush=True)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
