"""LayerNorm backed by the fused HIP kernel (csrc/layernorm.hip) on GPU,
eager reference on CPU. Drop-in for nn.LayerNorm over the last dim.

Reference parity: nn.LayerNorm call sites in ViT (vit_model.py), Swin
(models/swin_transformer.py:19-36, --fused_layernorm main.py:72), ConvNeXt.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import ext, use_hip


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        x = x.contiguous()
        y, mean, rstd = ext().layernorm_fwd(x, weight.contiguous(), bias.contiguous(), eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext().layernorm_bwd(dy.contiguous(), x, weight.contiguous(), mean, rstd)
        return dx, dw, db, None


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    if use_hip(x, weight, bias):
        return _LayerNormFn.apply(x, weight, bias, eps)
    return torch.nn.functional.layer_norm(x, (x.shape[-1],), weight, bias, eps)


class LayerNorm(nn.Module):
    """LayerNorm over the last dimension; HIP-fused on GPU."""

    def __init__(self, normalized_shape: int, eps: float = 1e-5):
        super().__init__()
        if isinstance(normalized_shape, (tuple, list)):
            assert len(normalized_shape) == 1
            normalized_shape = normalized_shape[0]
        self.normalized_shape = (normalized_shape,)
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(normalized_shape))
        self.bias = nn.Parameter(torch.zeros(normalized_shape))

    def forward(self, x):
        return layer_norm(x, self.weight, self.bias, self.eps)

    def extra_repr(self):
        return f"{self.normalized_shape}, eps={self.eps}"


class LayerNorm2d(nn.Module):
    """Channels-first LayerNorm for NCHW maps (ConvNeXt style): permutes to
    channels-last, runs the fused row kernel, permutes back."""

    def __init__(self, num_channels: int, eps: float = 1e-6):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_channels))
        self.bias = nn.Parameter(torch.zeros(num_channels))

    def forward(self, x):  # (N,C,H,W)
        x = x.permute(0, 2, 3, 1)
        x = layer_norm(x, self.weight, self.bias, self.eps)
        return x.permute(0, 3, 1, 2)
