"""Model EMA with fused multi-tensor update (torch._foreach_lerp_, one fused
launch per dtype-device group) and the yolov5/YOLOX decay ramp.

Reference parity: yolov5 utils/torch_utils.py:308+ (ModelEMA, decay ramp
1 - e^{-x/2000}), yolox/utils/ema.py:22-48.
"""
from __future__ import annotations

import math
from copy import deepcopy

import torch
import torch.nn as nn

from ..core.checkpoint import unwrap_model


class ModelEMA:
    def __init__(self, model: nn.Module, decay: float = 0.9999, tau: float = 2000.0,
                 updates: int = 0):
        self.ema = deepcopy(unwrap_model(model)).eval()
        for p in self.ema.parameters():
            p.requires_grad_(False)
        self.updates = updates
        self.decay_max = decay
        self.tau = tau

    def decay(self, updates: int) -> float:
        return self.decay_max * (1 - math.exp(-updates / self.tau))

    @torch.no_grad()
    def update(self, model: nn.Module) -> None:
        self.updates += 1
        d = self.decay(self.updates)
        msd = unwrap_model(model).state_dict()
        ema_f, model_f = [], []
        for k, v in self.ema.state_dict().items():
            if v.dtype.is_floating_point:
                ema_f.append(v)
                model_f.append(msd[k].detach().to(v.dtype))
            else:
                v.copy_(msd[k])
        if ema_f:
            # ema = d*ema + (1-d)*model  ==  lerp(ema, model, 1-d)
            torch._foreach_lerp_(ema_f, model_f, 1.0 - d)

    def state_dict(self) -> dict:
        return {"ema": self.ema.state_dict(), "updates": self.updates}

    def load_state_dict(self, sd: dict) -> None:
        self.ema.load_state_dict(sd["ema"])
        self.updates = sd.get("updates", self.updates)
