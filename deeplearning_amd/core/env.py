"""Device selection, seeding, run-dir helpers.

Reference parity: select_device (classification/swin_transformer/utils/torch_utils.py:35),
seed_everything (Image_segmentation/DeepLabV3 utils/utils.py:10), increment_path
(others/train_with_DDP/train.py:86 pattern).
"""
from __future__ import annotations

import os
import random
import re
from pathlib import Path

import numpy as np
import torch


def select_device(device: str = "") -> torch.device:
    """Pick a device. '' -> cuda:0 if available else cpu; 'cpu'; '0'/'cuda:1' etc."""
    if device in ("cpu", "CPU"):
        return torch.device("cpu")
    if device and device not in ("cuda",):
        d = device if device.startswith("cuda") else f"cuda:{device}"
    else:
        d = "cuda"
    if torch.cuda.is_available():
        return torch.device(d)
    return torch.device("cpu")


def seed_everything(seed: int = 0, rank: int = 0, deterministic: bool = False) -> None:
    """Seed python/numpy/torch. Per-rank offset keeps DP augmentations decorrelated
    (reference: swin main.py:322 seed = SEED + rank)."""
    s = seed + rank
    random.seed(s)
    np.random.seed(s % (2**32 - 1))
    torch.manual_seed(s)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(s)
    if deterministic:
        torch.backends.cudnn.deterministic = True
        torch.backends.cudnn.benchmark = False
    else:
        torch.backends.cudnn.benchmark = True


def increment_path(path: str | Path, exist_ok: bool = False, sep: str = "") -> Path:
    """runs/exp -> runs/exp, runs/exp2, runs/exp3, ... (reference yolov5/train_with_DDP)."""
    path = Path(path)
    if path.exists() and not exist_ok:
        dirs = [d for d in path.parent.glob(f"{path.name}{sep}*")]
        matches = [re.search(rf"%s{sep}(\d+)" % re.escape(path.name), str(d)) for d in dirs]
        idx = [int(m.group(1)) for m in matches if m]
        n = max(idx) + 1 if idx else 2
        path = path.parent / f"{path.name}{sep}{n}"
    return path


def time_sync() -> float:
    """cuda-accurate wall time (reference swin utils/torch_utils.py:71)."""
    import time

    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return time.time()


def gpu_mem_usage_mb() -> float:
    if not torch.cuda.is_available():
        return 0.0
    return torch.cuda.max_memory_allocated() / (1024.0 * 1024.0)


def get_env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v is not None and v != "" else default
