"""Activation ops backed by HIP kernels (csrc/elementwise.hip) on GPU.

gelu (erf form, matches nn.GELU default), silu (yolov5 Conv blocks), fused
residual add+ReLU (ResNet block join, classification/resnet/models/networks.py).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import dense, ext, same_layout, use_hip


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = dense(x)
        ctx.save_for_backward(x)
        return ext().gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return ext().gelu_bwd(same_layout(x, dy), x)


class _SiluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = dense(x)
        ctx.save_for_backward(x)
        return ext().silu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return ext().silu_bwd(same_layout(x, dy), x)


class _AddReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        a = dense(a)
        y = ext().add_relu_fwd(a, same_layout(a, b))
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dx = ext().relu_mask_bwd(same_layout(y, dy), y)
        return dx, dx


def gelu(x: torch.Tensor) -> torch.Tensor:
    if use_hip(x):
        return _GeluFn.apply(x)
    return torch.nn.functional.gelu(x)


def silu(x: torch.Tensor) -> torch.Tensor:
    if use_hip(x):
        return _SiluFn.apply(x)
    return torch.nn.functional.silu(x)


def add_relu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """relu(a + b) fused — the ResNet residual join."""
    if use_hip(a, b):
        return _AddReluFn.apply(a, b)
    return torch.relu(a + b)


class GELU(nn.Module):
    def forward(self, x):
        return gelu(x)


class SiLU(nn.Module):
    def forward(self, x):
        return silu(x)
