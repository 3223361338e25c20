"""Locality assignment unit tests with mocked partition->node maps
(reference test_data_source.py:16-166 technique: no cluster needed)."""

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
