"""Data-layer tests (mirrors reference test_matrix.py strategy)."""

import os

import numpy as np
import pandas as pd
import pytest

from xgboost_ray_amd import shm_store
from xgboost_ray_amd.matrix import (
    RayDMatrix,
    RayFileType,
    RayShardingMode,
    _detect_distributed,
    _get_sharding_indices,
    combine_data,
    ensure_sorted_by_qid,
)


@pytest.fixture
def xy():
    rng = np.random.RandomState(0)
    X = rng.randn(100, 4).astype(np.float32)
    y = rng.randint(0, 2, 100).astype(np.float32)
    return X, y


def _gather_all(dm, num_actors):
    parts_x, parts_y = [], []
    for rank in range(num_actors):
        shard = dm.get_data(rank, num_actors)
        parts_x.append(shard["data"])
        parts_y.append(shard["label"])
    return parts_x, parts_y


def test_sharding_indices_batch():
    idx = [_get_sharding_indices(RayShardingMode.BATCH, r, 3, 10) for r in range(3)]
    assert [len(i) for i in idx] == [4, 3, 3]
    assert np.concatenate(idx).tolist() == list(range(10))


def test_sharding_indices_interleaved():
    idx = _get_sharding_indices(RayShardingMode.INTERLEAVED, 1, 3, 10)
    assert idx.tolist() == [1, 4, 7]


@pytest.mark.parametrize(
    "sharding", [RayShardingMode.INTERLEAVED, RayShardingMode.BATCH]
)
def test_combine_data_roundtrip(sharding):
    data = np.arange(103, dtype=np.float32)
    shards = [
        data[_get_sharding_indices(sharding, r, 4, len(data))] for r in range(4)
    ]
    out = combine_data(sharding, shards)
    np.testing.assert_array_equal(out, data)


def test_combine_data_2d_interleaved():
    data = np.arange(60, dtype=np.float32).reshape(20, 3)
    shards = [
        data[_get_sharding_indices(RayShardingMode.INTERLEAVED, r, 3, 20)]
        for r in range(3)
    ]
    out = combine_data(RayShardingMode.INTERLEAVED, shards)
    np.testing.assert_array_equal(out, data)


@pytest.mark.parametrize(
    "sharding", [RayShardingMode.INTERLEAVED, RayShardingMode.BATCH]
)
def test_numpy_central_loading(xy, sharding):
    X, y = xy
    dm = RayDMatrix(X, label=y, sharding=sharding)
    dm.load_data(num_actors=3)
    parts_x, parts_y = _gather_all(dm, 3)
    Xr = combine_data(sharding, parts_x)
    yr = combine_data(sharding, parts_y)
    np.testing.assert_allclose(Xr, X)
    np.testing.assert_allclose(yr, y)
    assert dm.has_label()


def test_pandas_label_column(xy):
    X, y = xy
    df = pd.DataFrame(X, columns=[f"c{i}" for i in range(4)])
    df["target"] = y
    dm = RayDMatrix(df, label="target")
    shard = dm.get_data(0, 2)
    assert shard["data"].shape[1] == 4  # label column removed
    assert "target" not in shard["feature_names"]


def test_pandas_weight_base_margin(xy):
    X, y = xy
    w = np.abs(np.random.RandomState(1).randn(100)).astype(np.float32)
    bm = np.zeros(100, dtype=np.float32)
    dm = RayDMatrix(X, label=y, weight=w, base_margin=bm)
    shard0 = dm.get_data(0, 2)
    shard1 = dm.get_data(1, 2)
    wr = combine_data(
        RayShardingMode.INTERLEAVED, [shard0["weight"], shard1["weight"]]
    )
    np.testing.assert_allclose(wr, w)


def test_csv_file_loading(tmp_path, xy):
    X, y = xy
    df = pd.DataFrame(X, columns=[f"c{i}" for i in range(4)])
    df["label"] = y
    path = str(tmp_path / "data.csv")
    df.to_csv(path, index=False)
    dm = RayDMatrix(path, label="label")
    assert dm.loader.filetype == RayFileType.CSV
    shard = dm.get_data(0, 1)
    np.testing.assert_allclose(shard["data"], X, rtol=1e-5)


def test_parquet_distributed_loading(tmp_path, xy):
    X, y = xy
    files = []
    for i in range(4):
        df = pd.DataFrame(
            X[i * 25 : (i + 1) * 25], columns=[f"c{j}" for j in range(4)]
        )
        df["label"] = y[i * 25 : (i + 1) * 25]
        p = str(tmp_path / f"part{i}.parquet")
        df.to_parquet(p)
        files.append(p)
    dm = RayDMatrix(files, label="label")
    assert dm.distributed
    assert dm.sharding == RayShardingMode.FIXED
    s0 = dm.get_data(0, 2)  # files 0, 2
    s1 = dm.get_data(1, 2)  # files 1, 3
    assert s0["data"].shape[0] == 50
    assert s1["data"].shape[0] == 50
    np.testing.assert_allclose(s0["data"][:25], X[:25], rtol=1e-6)
    np.testing.assert_allclose(s1["data"][:25], X[25:50], rtol=1e-6)


def test_too_many_actors_distributed(tmp_path, xy):
    X, y = xy
    df = pd.DataFrame(X, columns=[f"c{j}" for j in range(4)])
    df["label"] = y
    p = str(tmp_path / "single.parquet")
    df.to_parquet(p)
    dm = RayDMatrix([p], label="label", distributed=True)
    with pytest.raises(RuntimeError, match="only has 1 shards"):
        dm.get_data(0, 2)


def test_object_store_source(xy):
    X, y = xy
    refs = [shm_store.put(X[:50]), shm_store.put(X[50:])]
    dm = RayDMatrix(refs, label=None, distributed=True)
    shard = dm.get_data(0, 1)
    assert shard["data"].shape == (100, 4)


def test_detect_distributed(tmp_path):
    assert _detect_distributed(["a.csv", "b.csv"])
    assert not _detect_distributed(np.zeros((4, 4)))
    assert _detect_distributed(str(tmp_path))


def test_qid_sorting():
    df = pd.DataFrame({"a": [1.0, 2.0, 3.0, 4.0]})
    qid = np.array([2, 0, 1, 0])
    sorted_qid, sorted_df = ensure_sorted_by_qid(df, qid)
    assert list(np.asarray(sorted_qid)) == [0, 0, 1, 2]
    assert sorted_df["a"].tolist() == [2.0, 4.0, 3.0, 1.0]


def test_group_rejected(xy):
    X, y = xy
    with pytest.raises(ValueError, match="group"):
        RayDMatrix(X, label=y, group=[50, 50])


def test_qid_plus_weight_rejected(xy):
    X, y = xy
    with pytest.raises(RuntimeError):
        RayDMatrix(X, label=y, qid=np.zeros(100), weight=np.ones(100))


def test_uid_identity(xy):
    X, y = xy
    a = RayDMatrix(X, label=y)
    b = RayDMatrix(X, label=y)
    assert a != b
    assert a == a
    assert len({a, b}) == 2


def test_reshard_on_num_actors_change(xy):
    """A loaded matrix reused with a different world size must re-shard
    (reference matrix.py raises on the mismatch; we re-shard, which is
    strictly safer than silently reusing stale shards)."""
    X, y = xy
    dm = RayDMatrix(X, y, num_actors=2)
    assert dm.loaded and len(dm.refs) == 2
    # shrink the world: all 100 rows must still be covered by 3 shards
    dm.load_data(num_actors=3)
    assert len(dm.refs) == 3
    parts_x, parts_y = _gather_all(dm, 3)
    assert sum(p.shape[0] for p in parts_x) == 100
    np.testing.assert_array_equal(
        np.sort(np.concatenate(parts_y)), np.sort(y)
    )
    # grow back: no KeyError, 2 shards again
    dm.load_data(num_actors=2)
    assert len(dm.refs) == 2
    parts_x, _ = _gather_all(dm, 2)
    assert sum(p.shape[0] for p in parts_x) == 100


class TestShmStoreEdges:
    """shm object store robustness (the Ray object-store analogue)."""

    def test_put_get_free_cycle(self):
        import numpy as np

        from xgboost_ray_amd import shm_store

        store = shm_store.get_store()
        arr = np.arange(100_000, dtype=np.float64)
        ref = store.put(arr)
        out = shm_store.get(ref)
        np.testing.assert_array_equal(out, arr)
        store.free(ref)
        # double-free must be harmless (failure-path drains may race)
        store.free(ref)

    def test_many_small_objects(self):
        import numpy as np

        from xgboost_ray_amd import shm_store

        store = shm_store.get_store()
        refs = [store.put(np.full(10, i, np.int32)) for i in range(50)]
        for i, r in enumerate(refs):
            assert shm_store.get(r)[0] == i
        for r in refs:
            store.free(r)


def test_env_overrides(monkeypatch):
    """RXGB_* env tier: read-time typed overrides (reference _XGBoostEnv)."""
    from xgboost_ray_amd.env import ENV

    base = ENV.STATUS_FREQUENCY_S
    monkeypatch.setenv("RXGB_STATUS_FREQUENCY_S", "123")
    assert ENV.STATUS_FREQUENCY_S == 123  # int-typed field, coerced
    monkeypatch.delenv("RXGB_STATUS_FREQUENCY_S")
    assert ENV.STATUS_FREQUENCY_S == base
