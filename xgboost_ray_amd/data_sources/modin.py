"""Modin data source (reference data_sources/modin.py:48-143).

Optional: requires `modin` (not installed in the base image). Partitions
are unwrapped with their node IPs and assigned to actors with the greedy
locality assigner (_distributed.py).
"""

from typing import Any, Dict, Optional, Sequence, Tuple

import pandas as pd

from xgboost_ray_amd.data_sources._distributed import (
    assign_partitions_to_actors,
    get_actor_rank_ips,
)
from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType


def _modin_df_type():
    try:
        from modin.pandas import DataFrame as ModinDF

        return ModinDF
    except Exception:
        return None


class Modin(DataSource):
    supports_central_loading = True
    supports_distributed_loading = True
    needs_partitions = True

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        cls = _modin_df_type()
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
        else:
            local_df = data._to_pandas()
            if indices is not None:
                local_df = local_df.iloc[indices]
        if ignore:
            keep = [c for c in local_df.columns if c not in ignore]
            local_df = local_df[keep]
        return local_df.reset_index(drop=True)

    @staticmethod
    def get_actor_shards(
        data: Any, actors: Sequence
    ) -> Tuple[Any, Optional[Dict[int, Any]]]:
        from modin.distributed.dataframe.pandas import unwrap_partitions

        actor_rank_ips = get_actor_rank_ips(actors)
        parts_with_ip = unwrap_partitions(data, axis=0, get_ip=True)
        ip_to_parts: Dict[str, list] = {}
        for ip, part in parts_with_ip:
            ip_to_parts.setdefault(ip, []).append(part)
        return data, assign_partitions_to_actors(ip_to_parts, actor_rank_ips)

    @staticmethod
    def get_n(data: Any) -> int:
        return len(data)
