"""Parquet data source (reference data_sources/parquet.py:9-48)."""

from typing import Any, Optional, Sequence, Union

import pandas as pd

from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType
from xgboost_ray_amd.data_sources.pandas import Pandas


class Parquet(DataSource):
    supports_distributed_loading = True

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        if isinstance(data, str):
            return filetype == RayFileType.PARQUET
        if isinstance(data, Sequence) and data and all(
            isinstance(x, str) for x in data
        ):
            return filetype == RayFileType.PARQUET
        return False

    @staticmethod
    def get_filetype(data: Any) -> Optional[RayFileType]:
        if isinstance(data, str) and data.endswith(".parquet"):
            return RayFileType.PARQUET
        if isinstance(data, (list, tuple)) and data:
            return Parquet.get_filetype(data[0])
        return None

    @staticmethod
    def load_data(
        data: Union[str, Sequence[str]],
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[int]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        if isinstance(data, (list, tuple)):
            shards = list(data)
            if indices is not None:
                shards = [shards[i] for i in indices]
            dfs = [pd.read_parquet(s, **kwargs) for s in shards]
            local_df = pd.concat(dfs, ignore_index=True, copy=False)
        else:
            local_df = pd.read_parquet(data, **kwargs)
        return Pandas.load_data(local_df, ignore=ignore)
