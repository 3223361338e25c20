"""__partitioned__ protocol data source
(reference data_sources/partitioned.py:18-99).

Supports any object exposing the `__partitioned__` dict protocol
(row-tiled 1d/2d partitions with location info).
"""

from typing import Any, Dict, Optional, Sequence, Tuple

import numpy as np
import pandas as pd

from xgboost_ray_amd.data_sources._distributed import (
    assign_partitions_to_actors,
    get_actor_rank_ips,
)
from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType
from xgboost_ray_amd.data_sources.numpy import Numpy
from xgboost_ray_amd.data_sources.pandas import Pandas


class Partitioned(DataSource):
    supports_central_loading = True
    supports_distributed_loading = True
    needs_partitions = True

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        return hasattr(data, "__partitioned__")

    @staticmethod
    def load_data(
        data: Any,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[Any]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        parts_meta = data.__partitioned__
        get = parts_meta["get"]
        all_parts = list(parts_meta["partitions"].values())
        if indices is not None:
            objs = indices if not all(
                isinstance(i, int) for i in indices
            ) else [all_parts[i]["data"] for i in indices]
        else:
            objs = [p["data"] for p in all_parts]
        frames = []
        for obj in objs:
            obj = get([obj])[0] if callable(get) else obj
            if isinstance(obj, pd.DataFrame):
                frames.append(obj)
            else:
                frames.append(Numpy.load_data(np.asarray(obj)))
        local_df = pd.concat(frames, ignore_index=True, copy=False)
        return Pandas.load_data(local_df, ignore=ignore)

    @staticmethod
    def get_actor_shards(
        data: Any, actors: Sequence
    ) -> Tuple[Any, Optional[Dict[int, Any]]]:
        parts_meta = data.__partitioned__
        actor_rank_ips = get_actor_rank_ips(actors)
        ip_to_parts: Dict[str, list] = {}
        for part in parts_meta["partitions"].values():
            location = part.get("location", ["127.0.0.1"])
            ip = location[0] if location else "127.0.0.1"
            ip_to_parts.setdefault(ip, []).append(part["data"])
        return data, assign_partitions_to_actors(ip_to_parts, actor_rank_ips)

    @staticmethod
    def get_n(data: Any) -> int:
        return len(data.__partitioned__["partitions"])
