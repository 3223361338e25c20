"""Ranking flagship: rank:ndcg lambdarank rounds/s on synthetic groups.

Usage: python benchmarks/bench_rank.py [--rows N] [--group G] [--steps K]
(BASELINE configs list 'RayXGBRanker rank:ndcg on synthetic MSLR-style
groups'; this measures the engine-level round time on one GPU.)
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import numpy as np
import torch

from xgboost_ray_amd.engine.quantile import BinnedMatrix
from xgboost_ray_amd.engine.trainer import BoostingEngine


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=10_000_000)
    ap.add_argument("--features", type=int, default=136)  # MSLR-style
    ap.add_argument("--group", type=int, default=100)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    args = ap.parse_args()
    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    g = torch.Generator(device=dev).manual_seed(7)
    X = torch.randn(args.rows, args.features, generator=g, device=dev)
    rel = (X[:, 0] * 2 + X[:, 1] + 0.5 * torch.randn(
        args.rows, generator=g, device=dev)).clamp(0, 4)
    y = rel.round().float()
    qid = torch.arange(args.rows, device=dev) // args.group
    dm = BinnedMatrix.build(X, label=y, qid=qid, max_bin=256)
    del X
    eng = BoostingEngine(
        {"objective": "rank:ndcg", "max_depth": 8, "eta": 0.1,
         "eval_metric": ["ndcg@10"]}, dm)
    for _ in range(args.warmup):
        eng.update()
    torch.cuda.synchronize() if dev.type == "cuda" else None
    t0 = time.perf_counter()
    for _ in range(args.steps):
        eng.update()
    torch.cuda.synchronize() if dev.type == "cuda" else None
    dt = (time.perf_counter() - t0) / args.steps
    print(f"rank:ndcg {args.rows}x{args.features} group={args.group}: "
          f"{dt * 1000:.2f} ms/round = {1 / dt:.1f} rounds/s")


if __name__ == "__main__":
    main()
