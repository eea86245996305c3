from .config import CfgNode, load_config
from .env import (gpu_mem_usage_mb, increment_path, seed_everything,
                  select_device, time_sync)
from .checkpoint import (auto_resume_helper, load_checkpoint, load_pretrained,
                         save_checkpoint, save_weights, strip_module_prefix,
                         unwrap_model)
from .dist import (all_gather_object_list, barrier, cleanup, get_local_rank,
                   get_rank, get_world_size, init_distributed, is_dist,
                   is_main_process, reduce_dict, reduce_value,
                   shared_random_seed, zero_first)
from .logging import TensorBoardWriter, create_logger
from .meters import AverageMeter, MetricLogger, SmoothedValue

__all__ = [
    "CfgNode", "load_config", "select_device", "seed_everything",
    "increment_path", "time_sync", "gpu_mem_usage_mb",
    "save_weights", "save_checkpoint", "load_checkpoint", "load_pretrained",
    "auto_resume_helper", "strip_module_prefix", "unwrap_model",
    "init_distributed", "cleanup", "is_dist", "get_rank", "get_world_size",
    "get_local_rank", "is_main_process", "barrier", "reduce_value",
    "reduce_dict", "all_gather_object_list", "zero_first", "shared_random_seed",
    "create_logger", "TensorBoardWriter",
    "AverageMeter", "SmoothedValue", "MetricLogger",
]
