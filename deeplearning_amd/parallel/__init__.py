from .ddp import BucketedDataParallel, wrap_data_parallel
from .syncbn import all_reduce_norm, convert_sync_batchnorm

__all__ = ["BucketedDataParallel", "wrap_data_parallel", "all_reduce_norm",
           "convert_sync_batchnorm"]
