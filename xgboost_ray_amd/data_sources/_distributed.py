"""Locality-aware partition -> actor assignment.

Two-phase greedy assignment with the reference's invariants
(reference data_sources/_distributed.py:24-112): first fill each actor with
co-located partitions up to the per-actor quota, then distribute the
remainder round-robin; the final counts differ by at most one.

On a single 8xMI355X node every actor is co-located with every partition,
but the algorithm is kept (and unit-tested with mocked IP maps) so
multi-node sources keep working unchanged.
"""

from collections import defaultdict
from typing import Any, Dict, Sequence


def get_actor_rank_ips(actors: Sequence) -> Dict[int, str]:
    """rank -> IP of each live actor (dead actors map to an empty string)."""
    ips = {}
    for rank, actor in enumerate(actors):
        if actor is None:
            ips[rank] = ""
        else:
            ips[rank] = actor.ip()
    return ips


def assign_partitions_to_actors(
    ip_to_parts: Dict[str, Sequence[Any]],
    actor_rank_ips: Dict[int, str],
) -> Dict[int, Sequence[Any]]:
    """Assign partitions to actor ranks, preferring co-located ones."""
    num_parts = sum(len(parts) for parts in ip_to_parts.values())
    num_actors = len(actor_rank_ips)
    min_parts_per_actor = max(0, num_parts // num_actors)
    max_parts_per_actor = min_parts_per_actor + int(
        num_parts % num_actors != 0
    )

    actor_to_parts: Dict[int, list] = defaultdict(list)

    # Phase 1a: fill every actor with local partitions up to the minimum.
    for rank, ip in actor_rank_ips.items():
        parts = ip_to_parts.get(ip, [])
        while parts and len(actor_to_parts[rank]) < min_parts_per_actor:
            actor_to_parts[rank].append(parts.pop(0))

    # Phase 1b: top up to the maximum with still-local partitions.
    for rank, ip in actor_rank_ips.items():
        parts = ip_to_parts.get(ip, [])
        while parts and len(actor_to_parts[rank]) < max_parts_per_actor:
            actor_to_parts[rank].append(parts.pop(0))

    # Phase 2: round-robin the remainder to the least-loaded actors.
    rest = []
    for parts in ip_to_parts.values():
        rest.extend(parts)
    while rest:
        rank = min(
            actor_rank_ips.keys(), key=lambda r: (len(actor_to_parts[r]), r)
        )
        actor_to_parts[rank].append(rest.pop(0))

    return dict(actor_to_parts)
