"""xgboost_ray_amd: MI355X-native distributed gradient-boosted-tree trainer.

A from-scratch re-implementation of the capabilities of ray-project/xgboost_ray
(reference: /root/reference/xgboost_ray/__init__.py:1-41) designed MI355X-first:

- one actor process per MI355X GPU, row-sharded data resident in 288 GB HBM3E
- the gpu_hist hot path (quantile sketch, binning, gradient histogram build,
  histogram subtraction, best-split scan, row partition, prediction tree-walk)
  is hand-written CDNA4 HIP (gfx950) with LDS-tiled histogram accumulation
- per-depth histogram AllReduce over RCCL/xGMI via torch.distributed
  (int64 fixed-point gradient pairs => bitwise-deterministic models,
  independent of atomic ordering and world size)
- no Ray and no libxgboost dependency: the control plane (actors, queue,
  events, fault tolerance, elastic restart) is implemented over
  multiprocessing + shared memory; the boosting engine is this package's own.
"""

from xgboost_ray_amd.main import RayParams, predict, train
from xgboost_ray_amd.matrix import (
    Data,
    RayDeviceQuantileDMatrix,
    RayDMatrix,
    RayFileType,
    RayQuantileDMatrix,
    RayShardingMode,
    combine_data,
)
from xgboost_ray_amd.booster import Booster, DMatrix
from xgboost_ray_amd.sklearn import (
    RayXGBClassifier,
    RayXGBRanker,
    RayXGBRegressor,
    RayXGBRFClassifier,
    RayXGBRFRegressor,
)

__version__ = "0.1.0"

__all__ = [
    "__version__",
    "RayParams",
    "RayDMatrix",
    "RayQuantileDMatrix",
    "RayDeviceQuantileDMatrix",
    "RayFileType",
    "RayShardingMode",
    "Data",
    "combine_data",
    "train",
    "predict",
    "Booster",
    "DMatrix",
    "RayXGBClassifier",
    "RayXGBRegressor",
    "RayXGBRFClassifier",
    "RayXGBRFRegressor",
    "RayXGBRanker",
]
