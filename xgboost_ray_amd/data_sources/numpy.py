"""Numpy data source (reference data_sources/numpy.py:13-33)."""

from typing import Any, Optional, Sequence

import numpy as np
import pandas as pd

from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType
from xgboost_ray_amd.data_sources.pandas import Pandas


class Numpy(DataSource):
    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        return isinstance(data, np.ndarray)

    @staticmethod
    def load_data(
        data: np.ndarray,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[int]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        arr = data if data.ndim > 1 else data.reshape(-1, 1)
        local_df = pd.DataFrame(
            arr, columns=[f"f{i}" for i in range(arr.shape[1])]
        )
        return Pandas.load_data(local_df, ignore=ignore, indices=indices)
