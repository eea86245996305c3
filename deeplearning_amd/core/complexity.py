"""Model complexity: parameter and FLOP counting.

Reference parity: swin main.py logs `n_parameters` and `model.flops()`
(classification/swin_transformer/models/swin_transformer.py flops() methods
per module). Rather than hand-maintaining per-module formulas for 97 zoo
models, this counts MACs generically with forward hooks on the layers that
dominate (Conv1d/2d, ConvTranspose2d, Linear, matmul-bearing attention is
covered through its Linear projections; the softmax(QK^T)V term is added for
modules exposing `num_heads` + `head_dim` conventions is NOT attempted —
elementwise/norm terms are <1% and ignored, as the reference's flops() also
does for norms).
"""
from __future__ import annotations

import torch
import torch.nn as nn


def count_params(model: nn.Module, trainable_only: bool = False) -> int:
    return sum(p.numel() for p in model.parameters()
               if p.requires_grad or not trainable_only)


@torch.no_grad()
def estimate_macs(model: nn.Module, *inputs) -> int:
    """Run one forward with hooks; returns multiply-accumulate count.

    FLOPs (as commonly quoted, e.g. resnet50 ~4.1 GFLOPs) == MACs here;
    double for the multiply+add convention.
    """
    total = [0]
    hooks = []

    def conv_hook(m, inp, out):
        o = out[0] if isinstance(out, tuple) else out
        spatial = o.numel() // (o.shape[0] * o.shape[1]) if o.dim() > 2 else 1
        k = 1
        for s in m.kernel_size if isinstance(m.kernel_size, tuple) \
                else (m.kernel_size,):
            k *= s
        total[0] += (o.shape[0] * o.shape[1] * spatial *
                     (m.in_channels // m.groups) * k)

    def linear_hook(m, inp, out):
        o = out[0] if isinstance(out, tuple) else out
        rows = o.numel() // o.shape[-1]
        total[0] += rows * m.in_features * m.out_features

    for mod in model.modules():
        if isinstance(mod, (nn.Conv1d, nn.Conv2d, nn.Conv3d,
                            nn.ConvTranspose2d)):
            hooks.append(mod.register_forward_hook(conv_hook))
        elif isinstance(mod, nn.Linear):
            hooks.append(mod.register_forward_hook(linear_hook))
    was_training = model.training
    model.eval()
    try:
        model(*inputs)
    finally:
        for h in hooks:
            h.remove()
        model.train(was_training)
    return total[0]


def complexity_str(model: nn.Module, *inputs) -> str:
    """'params 25.6M  MACs 4.09G' — logged by the classification trainer at
    startup (swin main.py parity)."""
    p = count_params(model)
    try:
        m = estimate_macs(model, *inputs)
        return f"params {p / 1e6:.1f}M  MACs {m / 1e9:.2f}G"
    except Exception:
        return f"params {p / 1e6:.1f}M"
