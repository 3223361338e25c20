"""Object-store data source: lists of shared-memory ObjectRefs.

Reference equivalent: lists of ``ray.ObjectRef`` partitions
(data_sources/object_store.py:11-40); here partitions live in the POSIX
shared-memory store (:mod:`xgboost_ray_amd.shm_store`).
"""

from typing import Any, Optional, Sequence

import numpy as np
import pandas as pd

from xgboost_ray_amd import shm_store
from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType
from xgboost_ray_amd.data_sources.pandas import Pandas
from xgboost_ray_amd.data_sources.numpy import Numpy


class ObjectStore(DataSource):
    supports_distributed_loading = True

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        return isinstance(data, Sequence) and bool(data) and all(
            isinstance(x, shm_store.ObjectRef) for x in data
        )

    @staticmethod
    def load_data(
        data: Sequence,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[int]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        parts = list(data)
        if indices is not None:
            parts = [parts[i] for i in indices]
        objs = [shm_store.get(ref) for ref in parts]
        dfs = []
        for obj in objs:
            if isinstance(obj, pd.DataFrame):
                dfs.append(obj)
            elif isinstance(obj, np.ndarray):
                dfs.append(Numpy.load_data(obj))
            else:
                raise ValueError(f"Unsupported object-store partition: {type(obj)}")
        local_df = pd.concat(dfs, ignore_index=True, copy=False)
        return Pandas.load_data(local_df, ignore=ignore)
