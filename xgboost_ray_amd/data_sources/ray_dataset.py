"""Ray Dataset source (reference data_sources/ray_dataset.py:32-110).

Optional: requires `ray` with ray.data. Datasets are split into one
shard per actor with locality hints; without Ray installed the source
never matches.
"""

from typing import Any, Dict, Optional, Sequence, Tuple

import pandas as pd

from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType


def _ray_dataset_type():
    try:
        import ray.data

        return ray.data.Dataset
    except Exception:
        return None


class RayDataset(DataSource):
    supports_central_loading = True
    supports_distributed_loading = True
    needs_partitions = False

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        cls = _ray_dataset_type()
        return cls is not None and isinstance(data, cls)

    @staticmethod
    def load_data(
        data: Any,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[Any]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        if indices is not None and indices and hasattr(indices[0], "to_pandas"):
            frames = [shard.to_pandas() for shard in indices]
            local_df = pd.concat(frames, ignore_index=True, copy=False)
        else:
            local_df = data.to_pandas()
        if ignore:
            keep = [c for c in local_df.columns if c not in ignore]
            local_df = local_df[keep]
        return local_df

    @staticmethod
    def get_actor_shards(
        data: Any, actors: Sequence
    ) -> Tuple[Any, Optional[Dict[int, Any]]]:
        splits = data.split(len(actors), equal=True)
        return data, {rank: [splits[rank]] for rank in range(len(actors))}

    @staticmethod
    def get_n(data: Any) -> int:
        return data.num_blocks()
