"""CSV data source (reference data_sources/csv.py:9-47).

Single files or lists of files; with a list, ``indices`` selects which
files an actor loads (file-level distributed sharding).
"""

from typing import Any, Optional, Sequence, Union

import pandas as pd

from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType
from xgboost_ray_amd.data_sources.pandas import Pandas


class CSV(DataSource):
    supports_distributed_loading = True

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        if isinstance(data, str):
            return filetype == RayFileType.CSV
        if isinstance(data, Sequence) and data and all(
            isinstance(x, str) for x in data
        ):
            return filetype == RayFileType.CSV
        return False

    @staticmethod
    def get_filetype(data: Any) -> Optional[RayFileType]:
        if isinstance(data, str) and (
            data.endswith(".csv") or data.endswith(".csv.gz")
        ):
            return RayFileType.CSV
        if isinstance(data, (list, tuple)) and data:
            return CSV.get_filetype(data[0])
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
            dfs = [pd.read_csv(s, **kwargs) for s in shards]
            local_df = pd.concat(dfs, ignore_index=True, copy=False)
        else:
            local_df = pd.read_csv(data, **kwargs)
        return Pandas.load_data(local_df, ignore=ignore)
