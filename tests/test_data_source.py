"""Locality assignment unit tests with mocked partition->node maps
(reference test_data_source.py:16-166 technique: no cluster needed)."""

import numpy as np
import pandas as pd

from xgboost_ray_amd.data_sources._distributed import (
    assign_partitions_to_actors,
)


def _ips(*ips):
    return {rank: ip for rank, ip in enumerate(ips)}


class TestAssignPartitions:
    def test_even_colocated(self):
        ip_to_parts = {
            "n1": ["a", "b"],
            "n2": ["c", "d"],
        }
        out = assign_partitions_to_actors(ip_to_parts, _ips("n1", "n2"))
        assert out[0] == ["a", "b"]
        assert out[1] == ["c", "d"]

    def test_remainder_round_robin(self):
        ip_to_parts = {"n1": ["a", "b", "c", "d", "e"]}
        out = assign_partitions_to_actors(ip_to_parts, _ips("n1", "n2"))
        # invariant: max - min <= 1
        sizes = sorted(len(v) for v in out.values())
        assert sizes == [2, 3]
        # the co-located actor gets its max quota first
        assert len(out[0]) == 3

    def test_skewed_locations(self):
        ip_to_parts = {
            "n1": [f"p{i}" for i in range(6)],
            "n2": ["q0"],
        }
        out = assign_partitions_to_actors(
            ip_to_parts, _ips("n1", "n2", "n3")
        )
        sizes = {rank: len(parts) for rank, parts in out.items()}
        assert sum(sizes.values()) == 7
        assert max(sizes.values()) - min(sizes.values()) <= 1
        # actor 0 (on n1) should hold only n1 partitions
        assert all(p.startswith("p") for p in out[0])
        # actor 1 (on n2) gets its local partition
        assert "q0" in out[1]

    def test_no_locality_info(self):
        ip_to_parts = {"unknown": list(range(8))}
        out = assign_partitions_to_actors(
            ip_to_parts, _ips("n1", "n2", "n3", "n4")
        )
        sizes = [len(out[r]) for r in range(4)]
        assert sizes == [2, 2, 2, 2]
        # assignment preserves partition order overall
        flat = [p for r in range(4) for p in out[r]]
        assert sorted(flat) == list(range(8))

    def test_more_actors_than_parts(self):
        ip_to_parts = {"n1": ["a", "b"]}
        out = assign_partitions_to_actors(
            ip_to_parts, _ips("n1", "n2", "n3")
        )
        total = sum(len(v) for v in out.values())
        assert total == 2
        assert max(len(v) for v in out.values()) == 1


class _FakeFuture:
    def __init__(self, key):
        self.key = key


class _FakeDaskClient:
    """Mocked dask.distributed client with a multi-node who_has map."""

    def __init__(self, key_to_workers):
        self._who = key_to_workers

    def persist(self, data):
        return data

    def who_has(self, futures):
        return {f.key: self._who.get(f.key, ()) for f in futures}


class _FakeDaskDF:
    def __init__(self, n):
        self.npartitions = n

    def get_partition(self, i):
        return ("part", i)


def test_dask_ip_probe_multi_node(monkeypatch):
    """get_ip_to_parts must map partitions to their dask-worker hosts
    (reference dask.py:136-167 probes object locations the same way) and
    the greedy assigner must then keep local partitions local."""
    import sys
    import types

    from xgboost_ray_amd.data_sources import dask as dask_src

    keymap = {
        ("p", 0): ("tcp://10.0.0.1:4000",),
        ("p", 1): ("tcp://10.0.0.2:4000",),
        ("p", 2): ("tcp://10.0.0.1:4001",),
        ("p", 3): ("tcp://10.0.0.2:4001",),
    }
    client = _FakeDaskClient(keymap)
    fake_mod = types.ModuleType("dask.distributed")
    fake_mod.default_client = lambda: client
    fake_mod.futures_of = lambda persisted: [
        _FakeFuture(("p", i)) for i in range(persisted.npartitions)
    ]
    fake_pkg = types.ModuleType("dask")
    fake_pkg.distributed = fake_mod
    monkeypatch.setitem(sys.modules, "dask", fake_pkg)
    monkeypatch.setitem(sys.modules, "dask.distributed", fake_mod)

    data = _FakeDaskDF(4)
    actor_ips = {0: "10.0.0.1", 1: "10.0.0.2"}
    ip_to_parts = dask_src.get_ip_to_parts(data, actor_ips)
    assert set(ip_to_parts) == {"10.0.0.1", "10.0.0.2"}
    assert ip_to_parts["10.0.0.1"] == [("part", 0), ("part", 2)]
    assert ip_to_parts["10.0.0.2"] == [("part", 1), ("part", 3)]

    from xgboost_ray_amd.data_sources._distributed import (
        assign_partitions_to_actors,
    )

    assigned = assign_partitions_to_actors(ip_to_parts, actor_ips)
    assert sorted(assigned[0]) == [("part", 0), ("part", 2)]
    assert sorted(assigned[1]) == [("part", 1), ("part", 3)]


def test_dask_ip_probe_local_fallback():
    """No dask.distributed client: every partition maps to the first
    actor's IP (single-process scheduler == process-local data)."""
    from xgboost_ray_amd.data_sources import dask as dask_src

    data = _FakeDaskDF(3)
    out = dask_src.get_ip_to_parts(data, {0: "127.0.0.1", 1: "127.0.0.1"})
    assert list(out) == ["127.0.0.1"]
    assert len(out["127.0.0.1"]) == 3


class _FakePartitioned:
    """Object implementing the __partitioned__ dict protocol
    (reference data_sources/partitioned.py:18-99)."""

    def __init__(self, frames, locations):
        self._frames = frames
        self.__partitioned__ = {
            "get": lambda objs: objs,
            "shape": (sum(len(f) for f in frames), frames[0].shape[1]),
            "partition_tiling": (len(frames), 1),
            "partitions": {
                (i, 0): {
                    "start": (sum(len(f) for f in frames[:i]), 0),
                    "shape": f.shape,
                    "data": f,
                    "location": [locations[i]],
                }
                for i, f in enumerate(frames)
            },
        }


def _make_partitioned(n_parts=4, rows=50, locations=None):
    rng = np.random.RandomState(0)
    frames = []
    for i in range(n_parts):
        df = pd.DataFrame(
            rng.randn(rows, 3).astype(np.float32), columns=["a", "b", "c"]
        )
        df["label"] = (df["a"] > 0).astype(np.float32)
        frames.append(df)
    locations = locations or ["127.0.0.1"] * n_parts
    return _FakePartitioned(frames, locations), frames


def test_partitioned_protocol_load():
    from xgboost_ray_amd.data_sources.partitioned import Partitioned

    data, frames = _make_partitioned()
    assert Partitioned.is_data_type(data)
    assert Partitioned.get_n(data) == 4
    df = Partitioned.load_data(data)
    assert len(df) == 200
    pd.testing.assert_frame_equal(
        df.iloc[:50].reset_index(drop=True), frames[0]
    )


def test_partitioned_locality_assignment():
    from xgboost_ray_amd.data_sources.partitioned import Partitioned

    data, frames = _make_partitioned(
        locations=["10.0.0.1", "10.0.0.2", "10.0.0.1", "10.0.0.2"]
    )

    class _FakeActor:
        def __init__(self, ip):
            self._ip = ip

        def ip(self):
            return self._ip

    actors = [_FakeActor("10.0.0.1"), _FakeActor("10.0.0.2")]
    _, assigned = Partitioned.get_actor_shards(data, actors)
    # each actor gets exactly its two co-located partitions
    for rank, ip in ((0, "10.0.0.1"), (1, "10.0.0.2")):
        got = assigned[rank]
        assert len(got) == 2
        want_idx = [i for i, loc in enumerate(
            ["10.0.0.1", "10.0.0.2", "10.0.0.1", "10.0.0.2"]) if loc == ip]
        for g, wi in zip(got, want_idx):
            pd.testing.assert_frame_equal(g, frames[wi])


def test_partitioned_end_to_end_train():
    """A __partitioned__ object trains through RayDMatrix + train()."""
    from xgboost_ray_amd import RayDMatrix, RayParams, train

    data, _ = _make_partitioned(n_parts=4, rows=200)
    dm = RayDMatrix(data, label="label")
    bst = train(
        {"objective": "binary:logistic", "max_depth": 3},
        dm, 3, ray_params=RayParams(num_actors=2),
    )
    assert bst.num_boosted_rounds() == 3


class _FakeRayDataset:
    """Duck-typed ray.data.Dataset: split/num_blocks/to_pandas."""

    def __init__(self, df, blocks=4):
        self._df = df
        self._blocks = blocks

    def num_blocks(self):
        return self._blocks

    def to_pandas(self):
        return self._df.copy()

    def split(self, n, equal=False):
        chunks = np.array_split(np.arange(len(self._df)), n)
        return [_FakeRayDataset(self._df.iloc[c], 1) for c in chunks]


def test_ray_dataset_shard_semantics():
    """RayDataset source: split(len(actors)) one shard per rank, shards
    concat back to the full frame (reference ray_dataset.py:32-110)."""
    from xgboost_ray_amd.data_sources.ray_dataset import RayDataset

    rng = np.random.RandomState(0)
    df = pd.DataFrame(rng.randn(100, 3), columns=list("abc"))
    ds = _FakeRayDataset(df)
    assert RayDataset.get_n(ds) == 4
    _, shards = RayDataset.get_actor_shards(ds, [object(), object()])
    assert set(shards) == {0, 1}
    out0 = RayDataset.load_data(ds, indices=shards[0])
    out1 = RayDataset.load_data(ds, indices=shards[1])
    joined = pd.concat([out0, out1], ignore_index=True)
    pd.testing.assert_frame_equal(
        joined, df.reset_index(drop=True), check_dtype=False
    )


def test_petastorm_url_detection(monkeypatch):
    """Petastorm source: scheme detection + batch-reader load path with a
    mocked petastorm module (lib absent in this image)."""
    import sys
    import types
    from collections import namedtuple

    from xgboost_ray_amd.data_sources.petastorm import Petastorm
    from xgboost_ray_amd.matrix import RayFileType

    Row = namedtuple("Row", ["a", "b"])

    class _Reader:
        def __init__(self, urls):
            self.urls = urls

        def __enter__(self):
            return iter([Row(a=np.arange(3.0), b=np.ones(3))])

        def __exit__(self, *a):
            return False

    fake = types.ModuleType("petastorm")
    fake.make_batch_reader = lambda urls: _Reader(urls)
    monkeypatch.setitem(sys.modules, "petastorm", fake)

    assert Petastorm.get_filetype("file:///x/y.parquet") == \
        RayFileType.PETASTORM
    assert Petastorm.get_filetype(["s3://b/k.parquet"]) == \
        RayFileType.PETASTORM
    assert Petastorm.get_filetype("/plain/path.parquet") is None
    assert Petastorm.is_data_type(
        "file:///x.parquet", RayFileType.PETASTORM)
    df = Petastorm.load_data(["file:///x.parquet"])
    assert list(df.columns) == ["a", "b"] and len(df) == 3


def test_modin_locality_map(monkeypatch):
    """Modin source: unwrap_partitions(get_ip=True) partition->IP map
    feeds the greedy assigner (reference modin.py:48-143)."""
    import sys
    import types

    from xgboost_ray_amd.data_sources import modin as modin_src

    parts = {
        "10.0.0.1": [pd.DataFrame({"x": [1.0]}),
                     pd.DataFrame({"x": [2.0]})],
        "10.0.0.2": [pd.DataFrame({"x": [3.0]}),
                     pd.DataFrame({"x": [4.0]})],
    }
    flat = [(ip, p) for ip, ps in parts.items() for p in ps]

    dist_pkg = types.ModuleType("modin.distributed.dataframe.pandas")
    dist_pkg.unwrap_partitions = lambda data, axis=0, get_ip=False: flat
    for name in ("modin", "modin.distributed", "modin.distributed.dataframe"):
        monkeypatch.setitem(sys.modules, name, types.ModuleType(name))
    monkeypatch.setitem(
        sys.modules, "modin.distributed.dataframe.pandas", dist_pkg)

    class _FakeActor:
        def __init__(self, ip):
            self._ip = ip

        def ip(self):
            return self._ip

    actors = [_FakeActor("10.0.0.1"), _FakeActor("10.0.0.2")]
    _, assigned = modin_src.Modin.get_actor_shards(object(), actors)
    assert float(assigned[0][0]["x"][0]) in (1.0, 2.0)
    assert float(assigned[1][0]["x"][0]) in (3.0, 4.0)
    assert len(assigned[0]) == 2 and len(assigned[1]) == 2
