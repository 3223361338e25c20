"""Diagnose in-loop stalls: time consecutive 5-round windows."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import time

import torch

from xgboost_ray_amd.engine.quantile import BinnedMatrix
from xgboost_ray_amd.engine.trainer import BoostingEngine


def main():
    n, F = 11_000_000, 28
    dev = torch.device("cuda")
    gen = torch.Generator(device=dev).manual_seed(0)
    X = torch.randn(n, F, device=dev, generator=gen)
    y = (X[:, 0] > 0).float()
    dm = BinnedMatrix.build(X, label=y, max_bin=256)
    del X
    engine = BoostingEngine(
        {"objective": "binary:logistic", "max_depth": 8, "eta": 0.1,
         "tree_method": "gpu_hist"},
        dm,
    )
    for window in range(8):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            engine.update()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 5 * 1000
        print(f"window {window}: {dt:8.2f} ms/round", flush=True)


if __name__ == "__main__":
    main()
