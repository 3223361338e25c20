"""DataSource abstraction (reference data_sources/data_source.py:22-155)."""

from enum import Enum
from typing import Any, Dict, List, Optional, Sequence, Tuple

import pandas as pd


class RayFileType(Enum):
    """Enum for different file types (used for overrides)."""

    CSV = 1
    PARQUET = 2
    PETASTORM = 3


class DataSource:
    """Abstract class for pluggable data sources.

    A data source knows how to recognize its data type, load it (optionally
    only selected shard indices), and extract label/weight columns from it.
    """

    supports_central_loading = True
    supports_distributed_loading = False
    needs_partitions = False

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        return False

    @staticmethod
    def get_filetype(data: Any) -> Optional[RayFileType]:
        return None

    @staticmethod
    def load_data(
        data: Any,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[int]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        raise NotImplementedError

    @staticmethod
    def convert_to_series(data: Any) -> pd.Series:
        if isinstance(data, pd.DataFrame):
            return pd.Series(data.squeeze())
        if not isinstance(data, pd.Series):
            return pd.Series(data)
        return data

    @classmethod
    def get_column(
        cls, data: pd.DataFrame, column: Any
    ) -> Tuple[pd.Series, Optional[str]]:
        """Resolve a column spec (name or array-like) against a dataframe."""
        if isinstance(column, str):
            return data[column], column
        if column is not None:
            return cls.convert_to_series(column), None
        return column, None

    @staticmethod
    def get_n(data: Any) -> int:
        return len(data)

    @staticmethod
    def get_actor_shards(
        data: Any, actors: Sequence
    ) -> Tuple[Any, Optional[Dict[int, Any]]]:
        """Distribute data partitions to actors (locality-aware sources)."""
        return data, None

    @staticmethod
    def update_feature_names(matrix, feature_names: Optional[List[str]]):
        return matrix
