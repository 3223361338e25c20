"""Pluggable data sources (reference data_sources/__init__.py).

Order matters: sources are probed in order and the first match wins.
Distributed-dataframe sources (Modin/Dask/Petastorm/ray.data) from the
reference are not applicable in this Ray-less single-node environment;
their loading capability is covered by the file-based (CSV/Parquet
multi-file) and ObjectStore sources.
"""

from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType
from xgboost_ray_amd.data_sources.numpy import Numpy
from xgboost_ray_amd.data_sources.pandas import Pandas
from xgboost_ray_amd.data_sources.modin import Modin
from xgboost_ray_amd.data_sources.dask import Dask
from xgboost_ray_amd.data_sources.partitioned import Partitioned
from xgboost_ray_amd.data_sources.csv import CSV
from xgboost_ray_amd.data_sources.parquet import Parquet
from xgboost_ray_amd.data_sources.petastorm import Petastorm
from xgboost_ray_amd.data_sources.object_store import ObjectStore
from xgboost_ray_amd.data_sources.ray_dataset import RayDataset

data_sources = [
    Numpy,
    Pandas,
    Modin,
    Dask,
    Partitioned,
    CSV,
    Parquet,
    Petastorm,
    RayDataset,
    ObjectStore,
]

__all__ = [
    "DataSource",
    "RayFileType",
    "Numpy",
    "Pandas",
    "Modin",
    "Dask",
    "Partitioned",
    "CSV",
    "Parquet",
    "Petastorm",
    "ObjectStore",
    "data_sources",
]
