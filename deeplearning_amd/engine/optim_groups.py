"""Weight-decay parameter grouping.

Reference parity: swin utils/optimizer.py set_weight_decay — biases, 1-D
params (norm scales) and model-declared keys (no_weight_decay()/
no_weight_decay_keywords()) train WITHOUT weight decay; everything else
keeps it. Decaying LayerNorm scales / relative-position tables measurably
hurts transformer fine-tuning, which is why the reference splits groups.
"""
from __future__ import annotations

import torch.nn as nn


def param_groups_weight_decay(model: nn.Module, weight_decay: float):
    """Two param groups: [{decay}, {no decay (wd=0)}] for the optimizer."""
    from ..core.checkpoint import unwrap_model

    m = unwrap_model(model)
    skip = set()
    if hasattr(m, "no_weight_decay"):
        skip = set(m.no_weight_decay())
    skip_kw = set()
    if hasattr(m, "no_weight_decay_keywords"):
        skip_kw = set(m.no_weight_decay_keywords())
    decay, no_decay = [], []
    for name, p in m.named_parameters():
        if not p.requires_grad:
            continue
        if (p.ndim <= 1 or name.endswith(".bias") or name in skip or
                any(k in name for k in skip_kw)):
            no_decay.append(p)
        else:
            decay.append(p)
    groups = []
    if decay:
        groups.append({"params": decay, "weight_decay": weight_decay})
    if no_decay:
        groups.append({"params": no_decay, "weight_decay": 0.0})
    return groups
