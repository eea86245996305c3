"""Swin fused window ops (roll + partition / merge + roll) — HIP kernels in
csrc/window.hip, same semantics as the reference CUDA kernels
(classification/swin_transformer/kernels/window_process/swin_window_process_kernel.cu
and their autograd wrappers kernels/window_process/window_process.py).

Eager reference (CPU + numerics tests): torch.roll + view/permute.
"""
from __future__ import annotations

import torch

from ._ext import ext, use_hip


def window_partition_eager(x: torch.Tensor, window_size: int) -> torch.Tensor:
    B, H, W, C = x.shape
    x = x.view(B, H // window_size, window_size, W // window_size, window_size, C)
    return x.permute(0, 1, 3, 2, 4, 5).contiguous().view(-1, window_size, window_size, C)


def window_reverse_eager(windows: torch.Tensor, window_size: int, H: int, W: int) -> torch.Tensor:
    B = int(windows.shape[0] / (H * W / window_size / window_size))
    x = windows.view(B, H // window_size, W // window_size, window_size, window_size, -1)
    return x.permute(0, 1, 3, 2, 4, 5).contiguous().view(B, H, W, -1)


class _RollPartitionFn(torch.autograd.Function):
    """roll(-shift) + window_partition fused."""

    @staticmethod
    def forward(ctx, x, window_size, shift):
        ctx.dims = (*x.shape, window_size, shift)
        return ext().window_partition_fwd(x.contiguous(), window_size, shift)

    @staticmethod
    def backward(ctx, grad):
        B, H, W, C, ws, shift = ctx.dims
        return ext().window_partition_bwd(grad.contiguous(), B, H, W, ws, shift), None, None


class _MergeRollFn(torch.autograd.Function):
    """window_reverse + roll(+shift) fused."""

    @staticmethod
    def forward(ctx, windows, B, H, W, window_size, shift):
        ctx.dims = (window_size, shift)
        return ext().window_merge_fwd(windows.contiguous(), B, H, W, window_size, shift)

    @staticmethod
    def backward(ctx, grad):
        ws, shift = ctx.dims
        return ext().window_merge_bwd(grad.contiguous(), ws, shift), None, None, None, None, None


def roll_and_window_partition(x: torch.Tensor, window_size: int, shift: int) -> torch.Tensor:
    """[B,H,W,C] -> [B*nW, ws, ws, C], rolling by (-shift, -shift) first."""
    if use_hip(x):
        return _RollPartitionFn.apply(x, window_size, shift)
    if shift:
        x = torch.roll(x, shifts=(-shift, -shift), dims=(1, 2))
    return window_partition_eager(x, window_size)


def window_merge_and_roll(windows: torch.Tensor, B: int, H: int, W: int,
                          window_size: int, shift: int) -> torch.Tensor:
    """[B*nW, ws, ws, C] -> [B,H,W,C], rolling by (+shift, +shift) after merge."""
    if use_hip(windows):
        return _MergeRollFn.apply(windows, B, H, W, window_size, shift)
    x = window_reverse_eager(windows, window_size, H, W)
    if shift:
        x = torch.roll(x, shifts=(shift, shift), dims=(1, 2))
    return x
