from .ddp import BucketedDataParallel, wrap_data_parallel
from .moe import CosineRouter, MoEMlp, expert_params  # noqa: F401
from .syncbn import all_reduce_norm, convert_sync_batchnorm

__all__ = ["BucketedDataParallel", "wrap_data_parallel", "all_reduce_norm",
           "convert_sync_batchnorm"]
