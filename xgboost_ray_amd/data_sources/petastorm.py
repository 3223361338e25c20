"""Petastorm data source (reference data_sources/petastorm.py:27-89).

Optional: requires `petastorm` (not installed in the base image). Reads
parquet datasets behind file:// / s3:// / gs:// / hdfs:// URLs.
"""

from typing import Any, Optional, Sequence

import pandas as pd

from xgboost_ray_amd.data_sources.data_source import DataSource, RayFileType

_SCHEMES = ("file://", "s3://", "gs://", "hdfs://")


def _petastorm_available() -> bool:
    try:
        import petastorm  # noqa

        return True
    except Exception:
        return False


class Petastorm(DataSource):
    supports_central_loading = True
    supports_distributed_loading = True

    @staticmethod
    def is_data_type(data: Any, filetype: Optional[RayFileType] = None) -> bool:
        if not _petastorm_available():
            return False
        if isinstance(data, str):
            return filetype == RayFileType.PETASTORM
        if isinstance(data, Sequence) and data and all(
            isinstance(x, str) for x in data
        ):
            return filetype == RayFileType.PETASTORM
        return False

    @staticmethod
    def get_filetype(data: Any) -> Optional[RayFileType]:
        if not _petastorm_available():
            return None
        if isinstance(data, str) and any(
            data.startswith(s) for s in _SCHEMES
        ):
            return RayFileType.PETASTORM
        if isinstance(data, (list, tuple)) and data:
            return Petastorm.get_filetype(data[0])
        return None

    @staticmethod
    def load_data(
        data: Any,
        ignore: Optional[Sequence[str]] = None,
        indices: Optional[Sequence[int]] = None,
        **kwargs,
    ) -> pd.DataFrame:
        from petastorm import make_batch_reader

        urls = [data] if isinstance(data, str) else list(data)
        if indices is not None:
            urls = [urls[i] for i in indices]
        frames = []
        with make_batch_reader(urls) as reader:
            for batch in reader:
                frames.append(pd.DataFrame(batch._asdict()))
        local_df = pd.concat(frames, ignore_index=True, copy=False)
        if ignore:
            keep = [c for c in local_df.columns if c not in ignore]
            local_df = local_df[keep]
        return local_df
