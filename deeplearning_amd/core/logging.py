"""Rank-aware logging + optional TensorBoard.

Reference parity: create_logger (swin utils/logger.py:9), setup_for_distributed
print-gating (RetinaNet train_utils/distributed_utils.py:255-268), SummaryWriter
usage throughout (mnist/train.py:29 etc.).
"""
from __future__ import annotations

import functools
import logging
import os
import sys
from pathlib import Path


@functools.lru_cache()
def create_logger(output_dir: str | None = None, dist_rank: int = 0,
                  name: str = "dla") -> logging.Logger:
    logger = logging.getLogger(name)
    logger.setLevel(logging.DEBUG)
    logger.propagate = False
    fmt = f"[%(asctime)s] rank{dist_rank} %(levelname)s: %(message)s"

    if dist_rank == 0:
        ch = logging.StreamHandler(sys.stdout)
        ch.setLevel(logging.DEBUG)
        ch.setFormatter(logging.Formatter(fmt, datefmt="%H:%M:%S"))
        logger.addHandler(ch)

    if output_dir is not None:
        Path(output_dir).mkdir(parents=True, exist_ok=True)
        fh = logging.FileHandler(os.path.join(output_dir, f"log_rank{dist_rank}.txt"))
        fh.setLevel(logging.DEBUG)
        fh.setFormatter(logging.Formatter(fmt, datefmt="%Y-%m-%d %H:%M:%S"))
        logger.addHandler(fh)
    if not logger.handlers:
        logger.addHandler(logging.NullHandler())
    return logger


class TensorBoardWriter:
    """Thin rank-0-only wrapper; no-op when tensorboard isn't importable or off-rank."""

    def __init__(self, log_dir: str | None, rank: int = 0):
        self.writer = None
        if rank == 0 and log_dir is not None:
            try:
                from torch.utils.tensorboard import SummaryWriter

                self.writer = SummaryWriter(log_dir=log_dir)
            except Exception:
                self.writer = None

    def add_scalar(self, tag, value, step):
        if self.writer:
            self.writer.add_scalar(tag, value, step)

    def add_scalars(self, tag, d, step):
        if self.writer:
            self.writer.add_scalars(tag, d, step)

    def add_histogram(self, tag, values, step):
        if self.writer:
            self.writer.add_histogram(tag, values, step)

    def add_image(self, tag, img, step):
        if self.writer:
            self.writer.add_image(tag, img, step)

    def flush(self):
        if self.writer:
            self.writer.flush()

    def close(self):
        if self.writer:
            self.writer.close()
