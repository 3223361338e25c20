"""200-round HIGGS realism soak: per-round eval metrics + a lossguide
number (profiles evidence; not the driver-facing bench)."""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))

import time

import torch

from xgboost_ray_amd.engine.quantile import BinnedMatrix
from xgboost_ray_amd.engine.trainer import EvalPack, run_training


def main():
    gen = torch.Generator(device="cuda").manual_seed(7)
    n, F = 11_000_000, 28
    X = torch.empty(n, F, device="cuda").normal_(generator=gen)
    w = torch.randn(F, generator=gen, device="cuda") * 0.5
    y = ((X @ w + torch.randn(n, generator=gen, device="cuda")) > 0).float()
    dm = BinnedMatrix.build(X, label=y, max_bin=256)
    del X

    res = {}
    t0 = time.perf_counter()
    run_training(
        {"objective": "binary:logistic", "max_depth": 8, "eta": 0.1,
         "tree_method": "gpu_hist", "eval_metric": ["logloss", "auc"]},
        dm, 200, evals=[EvalPack(name="train", X=None)], evals_result=res,
    )
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    auc = res["train"]["auc"][-1]
    print("200-round soak WITH per-round eval: "
          "%.2f ms/round, final train auc=%.4f" % (dt * 1000 / 200, auc))

    t0 = time.perf_counter()
    run_training(
        {"objective": "binary:logistic", "grow_policy": "lossguide",
         "max_leaves": 255, "max_depth": 0, "eta": 0.1,
         "tree_method": "gpu_hist"},
        dm, 20,
    )
    torch.cuda.synchronize()
    print("lossguide 255-leaf: %.1f ms/round"
          % ((time.perf_counter() - t0) * 1000 / 20))


if __name__ == "__main__":
    main()
