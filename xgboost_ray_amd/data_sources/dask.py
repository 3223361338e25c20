"""Dask dataframe data source (reference data_sources/dask.py:45-167).

Optional: requires `dask` (not installed in the base image). Partitions
are computed, located, and assigned with the greedy locality assigner.
"""

from typing import Any, Dict, Optional, Sequence, Tuple

import pandas as pd

from xgboost_ray_amd.data_sources._distributed import (
    assign_partitions_to_actors,
    get_actor_rank_ips,
)
from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType


def _dask_df_type():
    try:
        import dask.dataframe as dd

        return dd.DataFrame
    except Exception:
        return None


class Dask(DataSource):
    supports_central_loading = True
    supports_distributed_loading = True
    needs_partitions = True

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        cls = _dask_df_type()
        return cls is not None and isinstance(data, cls)

    @staticmethod
    def load_data(
        data: Any,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[Any]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        if indices is not None and indices and isinstance(
            indices[0], pd.DataFrame
        ):
            local_df = pd.concat(indices, copy=False)
        elif indices is not None:
            parts = [data.get_partition(i).compute() for i in indices]
            local_df = pd.concat(parts, copy=False)
        else:
            local_df = data.compute()
        if ignore:
            keep = [c for c in local_df.columns if c not in ignore]
            local_df = local_df[keep]
        return local_df.reset_index(drop=True)

    @staticmethod
    def get_actor_shards(
        data: Any, actors: Sequence
    ) -> Tuple[Any, Optional[Dict[int, Any]]]:
        actor_rank_ips = get_actor_rank_ips(actors)
        # single-node deployment: all partitions are local to every actor
        parts = [data.get_partition(i) for i in range(data.npartitions)]
        ip = next(iter(actor_rank_ips.values()), "127.0.0.1")
        return data, assign_partitions_to_actors({ip: parts}, actor_rank_ips)

    @staticmethod
    def get_n(data: Any) -> int:
        return data.npartitions
