"""Out-of-core streaming matrix (RayDeviceQuantileDMatrix equivalent)."""

import numpy as np
import pandas as pd
import pytest
import torch

from tests.utils import create_data
from xgboost_ray_amd import (
    RayDeviceQuantileDMatrix,
    RayDMatrix,
    RayParams,
    predict,
    train,
)
from xgboost_ray_amd.engine.quantile import BinnedMatrix


def _write_parquet_shards(tmp_path, X, y, n_files):
    files = []
    per = len(X) // n_files
    for i in range(n_files):
        sl = slice(i * per, (i + 1) * per if i < n_files - 1 else len(X))
        df = pd.DataFrame(X[sl], columns=[f"f{j}" for j in range(X.shape[1])])
        df["label"] = y[sl]
        p = str(tmp_path / f"p{i}.parquet")
        df.to_parquet(p)
        files.append(p)
    return files


def test_streaming_build_matches_inmemory():
    X, y = create_data(5000, 6)

    def chunk_fn():
        for i in range(5):
            sl = slice(i * 1000, (i + 1) * 1000)
            yield {"data": X[sl], "label": y[sl]}

    dm_s = BinnedMatrix.build_streaming(
        chunk_fn, n_features=6, device=torch.device("cpu"), max_bin=64
    )
    dm_m = BinnedMatrix.build(
        torch.from_numpy(X), label=torch.from_numpy(y), max_bin=64
    )
    # exact sketch path -> identical cuts -> identical bins
    torch.testing.assert_close(dm_s.cuts.cuts_flat, dm_m.cuts.cuts_flat)
    torch.testing.assert_close(dm_s.bins, dm_m.bins)
    torch.testing.assert_close(dm_s.label, dm_m.label)


def test_streaming_train_equals_central(tmp_path):
    X, y = create_data(4000, 6)
    files = _write_parquet_shards(tmp_path, X, y, 4)
    params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3}

    bst_stream = train(
        params,
        RayDeviceQuantileDMatrix(files, label="label"),
        6,
        ray_params=RayParams(num_actors=2),
    )
    bst_central = train(
        params,
        RayDMatrix(X, label=y),
        6,
        ray_params=RayParams(num_actors=2),
    )
    np.testing.assert_allclose(
        bst_stream.predict(X, output_margin=True),
        bst_central.predict(X, output_margin=True),
        rtol=1e-6, atol=1e-7,
    )


def test_streaming_predict(tmp_path):
    X, y = create_data(3000, 5)
    files = _write_parquet_shards(tmp_path, X, y, 4)
    bst = train(
        {"objective": "binary:logistic", "max_depth": 4},
        RayDeviceQuantileDMatrix(files, label="label"),
        5,
        ray_params=RayParams(num_actors=2),
    )
    pred = predict(
        bst,
        RayDeviceQuantileDMatrix(files, label="label"),
        ray_params=RayParams(num_actors=2),
    )
    # FIXED sharding: rank0 gets files 0,2; rank1 gets 1,3
    reorder = np.concatenate([
        np.arange(0, 750), np.arange(1500, 2250),
        np.arange(750, 1500), np.arange(2250, 3000),
    ])
    local = bst.predict(X)[reorder]
    np.testing.assert_allclose(pred, local, rtol=1e-5, atol=1e-6)
