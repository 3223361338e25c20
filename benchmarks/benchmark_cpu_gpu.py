"""Train benchmark over N actors (reference tests/release/benchmark_cpu_gpu.py).

Usage: python benchmarks/benchmark_cpu_gpu.py NUM_WORKERS NUM_ROUNDS NUM_FILES
       [--gpu] [--smoke-test] [--data-dir DIR]

Generates (once) NUM_FILES parquet shards of synthetic classification data,
trains a distributed model, prints TRAIN TIME TAKEN / TOTAL TIME TAKEN and
appends a CSV row to res.csv - the same protocol the reference uses
(reference benchmark_cpu_gpu.py:101-106, 173-197).
"""

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))


import argparse
import os
import time

import numpy as np
import pandas as pd

from xgboost_ray_amd import RayDMatrix, RayParams, train


def generate_files(data_dir, num_files, rows_per_file, cols, seed=7):
    os.makedirs(data_dir, exist_ok=True)
    files = []
    for i in range(num_files):
        path = os.path.join(data_dir, f"part_{i:04d}.parquet")
        files.append(path)
        if os.path.exists(path):
            continue
        rng = np.random.RandomState(seed + i)
        X = rng.randn(rows_per_file, cols).astype(np.float32)
        w = np.linspace(-1, 1, cols)
        y = ((X @ w + 0.5 * rng.randn(rows_per_file)) > 0).astype(np.float32)
        df = pd.DataFrame(X, columns=[f"f{c}" for c in range(cols)])
        df["label"] = y
        df.to_parquet(path)
    return files


def main():
    p = argparse.ArgumentParser()
    p.add_argument("num_workers", type=int)
    p.add_argument("num_rounds", type=int)
    p.add_argument("num_files", type=int)
    p.add_argument("--gpu", action="store_true")
    p.add_argument("--smoke-test", action="store_true")
    p.add_argument("--data-dir", default="/tmp/rxgb_bench_data")
    p.add_argument("--rows-per-file", type=int, default=100_000)
    p.add_argument("--cols", type=int, default=40)
    args = p.parse_args()

    if args.smoke_test:
        args.rows_per_file = 2_000

    t_total = time.time()
    files = generate_files(
        args.data_dir if not args.smoke_test else args.data_dir + "_smoke",
        args.num_files, args.rows_per_file, args.cols,
    )
    init_taken = time.time() - t_total

    dtrain = RayDMatrix(files, label="label")
    params = {
        "objective": "binary:logistic",
        "tree_method": "gpu_hist" if args.gpu else "hist",
        "max_depth": 8,
        "eval_metric": ["logloss", "error"],
    }
    t_train = time.time()
    res = {}
    bst = train(
        params,
        dtrain,
        args.num_rounds,
        evals=[(dtrain, "train")],
        evals_result=res,
        ray_params=RayParams(
            num_actors=args.num_workers,
            checkpoint_frequency=max(1, args.num_rounds // 2),
        ),
    )
    train_taken = time.time() - t_train
    total_taken = time.time() - t_total

    print(f"TRAIN TIME TAKEN: {train_taken:.2f} seconds")
    print(f"TOTAL TIME TAKEN: {total_taken:.2f} seconds "
          f"({init_taken:.2f} for init)")
    print(f"Final training error: {res['train']['error'][-1]:.4f}")
    with open("res.csv", "at") as f:
        f.write(
            f"{args.num_workers},{args.num_files},{int(args.gpu)},"
            f"{args.num_rounds},{init_taken:.4f},{total_taken:.4f},"
            f"{train_taken:.4f}\n"
        )
    bst.save_model("benchmark_model.json")


if __name__ == "__main__":
    main()
