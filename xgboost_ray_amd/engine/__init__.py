"""The MI355X-native boosting engine.

Replaces the external libxgboost (C++/CUDA gpu_hist + Rabit/NCCL) the
reference delegates its entire hot loop to: quantile
sketch + binning, depth-wise histogram tree growing, objectives, metrics,
and the RCCL collective layer.
"""
