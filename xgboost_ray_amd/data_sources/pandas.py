"""Pandas data source (reference data_sources/pandas.py:8-30)."""

from typing import Any, Optional, Sequence

import pandas as pd

from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType


class Pandas(DataSource):
    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        return isinstance(data, pd.DataFrame)

    @staticmethod
    def load_data(
        data: pd.DataFrame,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[int]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        local_df = data
        if ignore:
            keep = [c for c in local_df.columns if c not in ignore]
            local_df = local_df[keep]
        if indices is not None:
            local_df = local_df.iloc[indices]
        return local_df
