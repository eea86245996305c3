"""Per-iteration LR schedulers: cosine/linear/step with linear warmup, and
poly (segmentation).

Reference parity: swin utils/lr_scheduler.py:7 (cosine+warmup per-iter),
FCN poly-LR (utils/train_and_eval.py:65-85), yolov5 warmup-interp
(train.py:330), linear scaling rule (swin main.py:329-343).
"""
from __future__ import annotations

import math

import torch


class WarmupScheduler:
    """Step once per ITERATION. Wraps an optimizer, sets lr for every group
    (honoring per-group base lr ratio)."""

    def __init__(self, optimizer: torch.optim.Optimizer, total_steps: int,
                 warmup_steps: int = 0, warmup_lr: float = 1e-6,
                 min_lr: float = 0.0, mode: str = "cosine",
                 milestones=(), gamma: float = 0.1, power: float = 0.9):
        self.optimizer = optimizer
        self.total_steps = max(1, total_steps)
        self.warmup_steps = warmup_steps
        self.warmup_lr = warmup_lr
        self.min_lr = min_lr
        self.mode = mode
        self.milestones = sorted(milestones)
        self.gamma = gamma
        self.power = power
        self.base_lrs = [g["lr"] for g in optimizer.param_groups]
        self.last_step = -1
        self.step()  # initialize lr at step 0

    def _factor(self, step: int) -> float:
        if step < self.warmup_steps:
            return -1.0  # sentinel: handled per-group with warmup_lr offset
        t = (step - self.warmup_steps) / max(1, self.total_steps - self.warmup_steps)
        t = min(t, 1.0)
        if self.mode == "cosine":
            return 0.5 * (1 + math.cos(math.pi * t))
        if self.mode == "linear":
            return 1.0 - t
        if self.mode == "poly":
            return (1.0 - t) ** self.power
        if self.mode == "step":
            k = sum(1 for m in self.milestones if step >= m)
            return self.gamma ** k
        if self.mode == "constant":
            return 1.0
        raise ValueError(f"unknown scheduler mode {self.mode}")

    def step(self, step: int | None = None):
        self.last_step = self.last_step + 1 if step is None else step
        f = self._factor(self.last_step)
        for g, base in zip(self.optimizer.param_groups, self.base_lrs):
            if f < 0:  # warmup: linear from warmup_lr to base
                a = self.last_step / max(1, self.warmup_steps)
                g["lr"] = self.warmup_lr + (base - self.warmup_lr) * a
            else:
                g["lr"] = self.min_lr + (base - self.min_lr) * f

    def state_dict(self):
        return {"last_step": self.last_step, "base_lrs": self.base_lrs}

    def load_state_dict(self, sd):
        self.base_lrs = sd["base_lrs"]
        self.step(sd["last_step"])


def scale_lr_linear(base_lr: float, global_batch: int, base_batch: int = 512) -> float:
    """swin main.py:329-343 linear scaling rule."""
    return base_lr * global_batch / base_batch
