"""1x1 convolution on the hand-written implicit-GEMM MFMA kernel
(csrc/conv1x1.hip) and its fusion with train-mode BatchNorm.

This is the BASELINE.json north-star conv path: NHWC 1x1 conv IS the GEMM
C[M,N] = A[M,K] @ W[N,K]^T on CDNA4 matrix cores. The same kernel computes
forward (A=x, B=W) and dgrad (A=dy, B=W^T); wgrad is a separate M-contraction
kernel. The forward epilogue can accumulate per-channel sum/sumsq so the
following BatchNorm needs no separate stats pass over the conv output, and in
eval mode the whole conv+BN(+ReLU) chain is ONE kernel (scale/shift epilogue).

Reference parity: nn.Conv2d 1x1 call sites — ResNet Bottleneck conv1/conv3
and downsample (classification/resnet/models/networks.py:78-133).
"""
from __future__ import annotations

import os

import torch
import torch.nn.functional as F
from torch import nn

from ._ext import ext, use_hip


def _flatten_nhwc(x: torch.Tensor) -> torch.Tensor:
    """[B,C,H,W] channels_last -> [B*H*W, C] view (no copy)."""
    B, C, H, W = x.shape
    return x.permute(0, 2, 3, 1).reshape(B * H * W, C)


def _unflatten_nhwc(y2d: torch.Tensor, B: int, H: int, W: int) -> torch.Tensor:
    """[B*H*W, N] -> [B,N,H,W] channels_last view."""
    return y2d.view(B, H, W, -1).permute(0, 3, 1, 2)


class _Conv1x1Fn(torch.autograd.Function):
    """y2d[M,N] = x2d[M,K] @ w[N,K]^T with optional fused epilogue.

    Inputs are 2D bf16 (the NHWC flattening happens in the callers). The
    optional second output is the per-channel {sum, sumsq} of the raw conv
    output (fp32 [2N]) for the downstream BatchNorm — non-differentiable.
    """

    @staticmethod
    def forward(ctx, x2d, w, bias, scale, shift, residual, relu, want_stats):
        xb = x2d if x2d.dtype == torch.bfloat16 else x2d.to(torch.bfloat16)
        wb = w if w.dtype == torch.bfloat16 else w.to(torch.bfloat16)
        wb = wb.contiguous()
        y2d, sums = ext().conv1x1_fwd(xb, wb, bias, scale, shift, residual,
                                      relu, want_stats)
        ctx.save_for_backward(xb, wb)
        ctx.x_dtype = x2d.dtype
        ctx.w_dtype = w.dtype
        if want_stats:
            ctx.mark_non_differentiable(sums)
            return y2d, sums
        return y2d

    @staticmethod
    def backward(ctx, dy2d, *unused):
        xb, wb = ctx.saved_tensors
        dy2d = dy2d.contiguous()
        if dy2d.dtype != torch.bfloat16:
            dy2d = dy2d.to(torch.bfloat16)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            # LDS-tiled transpose kernel (torch's strided copy is ~15x slower)
            wt = ext().transpose2d(wb)  # [K,N] bf16
            dx, _ = ext().conv1x1_fwd(dy2d, wt, None, None, None, None,
                                      False, False)
            if dx.dtype != ctx.x_dtype:
                dx = dx.to(ctx.x_dtype)
        if ctx.needs_input_grad[1]:
            dw = ext().conv1x1_wgrad(dy2d, xb)  # fp32 [N,K]
            if dw.dtype != ctx.w_dtype:
                dw = dw.to(ctx.w_dtype)
        if ctx.needs_input_grad[2]:
            # bias rides the fused epilogue in forward; its grad is the
            # per-channel column sum of dy
            db = dy2d.float().sum(0)
        return dx, dw, db, None, None, None, None, None, None


class _BNFromStatsFn(torch.autograd.Function):
    """Train-mode BN whose stats pass was fused into the producing conv
    kernel: forward = finalize + apply (one read of x instead of two),
    backward = the full batchnorm_bwd (including the d-mean/d-var terms)."""

    @staticmethod
    def forward(ctx, x, weight, bias, sums, running_mean, running_var,
                momentum, eps, relu):
        y, mean, rstd = ext().batchnorm_fwd_from_sums(
            x, weight, bias, sums, running_mean, running_var, momentum, eps,
            relu)
        ctx.save_for_backward(x, y, weight, mean, rstd)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext().batchnorm_bwd(
            dy, x, y if ctx.relu else None, weight, mean, rstd, ctx.relu)
        return dx, dw, db, None, None, None, None, None, None


class _Stride2Fn(torch.autograd.Function):
    """Stride-2 NHWC subsample via coalesced row-copy kernels; backward is a
    single-pass scatter that zero-fills unsampled rows (torch's strided
    slicing assign ran ~700us per call on these shapes)."""

    @staticmethod
    def forward(ctx, x):
        ctx.ihw = (x.shape[2], x.shape[3])
        return ext().stride2_gather(x)

    @staticmethod
    def backward(ctx, dy):
        if not dy.is_contiguous(memory_format=torch.channels_last):
            dy = dy.contiguous(memory_format=torch.channels_last)
        return ext().stride2_scatter(dy, *ctx.ihw)


def _conv1x1_disabled() -> bool:
    return os.environ.get("DLA_NO_CONV1X1", "0") == "1"


def can_fuse_conv1x1(x: torch.Tensor, conv: nn.Conv2d) -> bool:
    """True when the HIP conv1x1 GEMM path applies to this call."""
    if _conv1x1_disabled() or not use_hip(x):
        return False
    if x.dtype != torch.bfloat16:
        return False
    if conv.kernel_size != (1, 1) or conv.groups != 1:
        return False
    if conv.padding != (0, 0) or conv.dilation != (1, 1):
        return False
    if conv.stride not in ((1, 1), (2, 2)):
        return False
    cin, cout = conv.in_channels, conv.out_channels
    if cin % 64 != 0 or cout % 64 != 0:
        return False
    return x.dim() == 4 and x.is_contiguous(memory_format=torch.channels_last)


def _prep(x: torch.Tensor, conv: nn.Conv2d):
    """Subsample for stride 2 (autograd scatters the grad back), flatten
    NHWC; the bf16 weight cast happens inside _Conv1x1Fn so fp32 weight
    leaves get their fp32 wgrad directly."""
    stride = conv.stride[0]
    if stride == 2:
        x = _Stride2Fn.apply(x)
    B, C, H, W = x.shape
    x2d = _flatten_nhwc(x)
    w2d = conv.weight.reshape(conv.weight.shape[0], conv.weight.shape[1])
    bias = conv.bias.float() if conv.bias is not None else None
    return x2d, w2d, bias, (B, H, W)


def conv1x1(x: torch.Tensor, conv: nn.Conv2d) -> torch.Tensor:
    """Plain 1x1 conv through the MFMA GEMM kernel (bias fused)."""
    x2d, wb, bias, (B, H, W) = _prep(x, conv)
    y2d = _Conv1x1Fn.apply(x2d, wb, bias, None, None, None, False, False)
    return _unflatten_nhwc(y2d, B, H, W)


def conv1x1_bn(x: torch.Tensor, conv: nn.Conv2d, bn) -> torch.Tensor:
    """Fused conv1x1 + BatchNorm2d(+ReLU) chain.

    train: conv GEMM with fused channel-stats epilogue -> BN finalize+apply
           (skips the stats read pass over the conv output entirely)
    eval (no grad): ONE kernel — conv GEMM with scale/shift(+ReLU) epilogue.
    """
    relu = bool(getattr(bn, "relu", False))
    x2d, wb, bias, (B, H, W) = _prep(x, conv)

    if bn.running_mean.dtype != torch.float32:
        bn.running_mean.data = bn.running_mean.data.float()
        bn.running_var.data = bn.running_var.data.float()

    if not bn.training:
        rstd = torch.rsqrt(bn.running_var + bn.eps)
        scale = bn.weight.float() * rstd
        shift = bn.bias.float() - bn.running_mean * scale
        if not (torch.is_grad_enabled() and
                (x.requires_grad or conv.weight.requires_grad or
                 bn.weight.requires_grad)):
            y2d, _ = ext().conv1x1_fwd(
                x2d.to(torch.bfloat16), wb.to(torch.bfloat16).contiguous(),
                bias, scale.contiguous(), shift.contiguous(), None, relu,
                False)
            return _unflatten_nhwc(y2d, B, H, W)
        # eval with grad: conv kernel + differentiable scale/shift
        y2d = _Conv1x1Fn.apply(x2d, wb, bias, None, None, None, False, False)
        y = _unflatten_nhwc(y2d, B, H, W)
        y = y * scale.reshape(1, -1, 1, 1).to(y.dtype) + \
            shift.reshape(1, -1, 1, 1).to(y.dtype)
        return torch.relu(y) if relu else y

    # train: conv + fused stats, then BN finalize/apply
    if bn.num_batches_tracked is not None:
        bn.num_batches_tracked.add_(1)
    momentum = bn._eaf() if hasattr(bn, "_eaf") else (bn.momentum or 0.0)
    y2d, partials = _Conv1x1Fn.apply(x2d, wb, bias, None, None, None, False,
                                     True)
    sums = partials.sum(dim=0)  # [n_blocks, 2C] slab -> [2C]
    y_raw = _unflatten_nhwc(y2d, B, H, W)
    return _BNFromStatsFn.apply(y_raw, bn.weight, bn.bias, sums,
                                bn.running_mean, bn.running_var, momentum,
                                bn.eps, relu)


def conv_bn(x: torch.Tensor, conv: nn.Conv2d, bn) -> torch.Tensor:
    """conv -> bn chain, routed through the fused MFMA path when it applies
    (1x1 any stride, or 3x3 s1p1; GPU, bf16, channels_last, channels % 64);
    otherwise eager."""
    from .batchnorm import BatchNorm2d

    if isinstance(bn, BatchNorm2d):
        if can_fuse_conv1x1(x, conv):
            return conv1x1_bn(x, conv, bn)
        if can_fuse_conv3x3(x, conv):
            return conv3x3_bn(x, conv, bn)
    return bn(conv(x))


class _Conv3x3Fn(torch.autograd.Function):
    """3x3 s1 p1 conv: forward on the TAPS=9 implicit-GEMM MFMA kernel
    (beats MIOpen fwd on 3 of 4 ResNet shapes, up to 2.75x, and can fuse
    the BN-stats epilogue); dgrad reuses the SAME kernel with the flipped,
    channel-transposed weight; wgrad is the TAPS=9 variant of the
    M-contraction wgrad kernel — conv2d 3x3 is hand-written end to end."""

    @staticmethod
    def forward(ctx, x4d, w, want_stats):
        B, C, H, W = x4d.shape
        xb = x4d if x4d.dtype == torch.bfloat16 else x4d.to(torch.bfloat16)
        wb = w if w.dtype == torch.bfloat16 else w.to(torch.bfloat16)
        a = _flatten_nhwc(xb)
        w9 = wb.permute(0, 2, 3, 1).reshape(w.shape[0], -1).contiguous()
        y2d, sums = ext().conv3x3_fwd(a, w9, H, W, None, None, None, None,
                                      False, want_stats)
        ctx.save_for_backward(xb, wb)
        ctx.w_dtype = w.dtype
        ctx.x_dtype = x4d.dtype
        y = _unflatten_nhwc(y2d, B, H, W)
        if want_stats:
            ctx.mark_non_differentiable(sums)
            return y, sums
        return y

    @staticmethod
    def backward(ctx, dy, *unused):
        xb, wb = ctx.saved_tensors
        B, C, H, W = xb.shape
        if not dy.is_contiguous(memory_format=torch.channels_last):
            dy = dy.contiguous(memory_format=torch.channels_last)
        if dy.dtype != torch.bfloat16:
            dy = dy.to(torch.bfloat16)
        dy2d = _flatten_nhwc(dy)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            # dgrad IS the same TAPS=9 kernel: conv3x3 of dy with the
            # spatially-flipped, channel-transposed weight
            wd = wb.flip(2, 3).transpose(0, 1)       # [Ci, Co, 3, 3]
            wd9 = wd.permute(0, 2, 3, 1).reshape(C, -1).contiguous()
            dx2d, _ = ext().conv3x3_fwd(dy2d, wd9, H, W, None, None, None,
                                        None, False, False)
            dx = _unflatten_nhwc(dx2d, B, H, W)
            if dx.dtype != ctx.x_dtype:
                dx = dx.to(ctx.x_dtype)
        if ctx.needs_input_grad[1]:
            dw9 = ext().conv3x3_wgrad(dy2d, _flatten_nhwc(xb), H, W)
            dw = dw9.view(wb.shape[0], 3, 3, C).permute(0, 3, 1, 2)
            if dw.dtype != ctx.w_dtype:
                dw = dw.to(ctx.w_dtype)
            else:
                dw = dw.contiguous()
        return dx, dw, None


def _conv3x3_enabled() -> bool:
    # measured round 2: the TAPS=9 fwd kernel beats MIOpen fwd on 3 of 4
    # ResNet shapes in isolation (to 2.75x), but the integrated train step
    # regressed 7403 -> 7192 img/s (library-backward/permute overheads eat
    # the fwd win). Routed opt-in until the hand-written 3x3 dgrad/wgrad
    # land; the kernel stays fully tested either way.
    return os.environ.get("DLA_CONV3X3", "0") == "1"


def can_fuse_conv3x3(x: torch.Tensor, conv: nn.Conv2d) -> bool:
    if _conv1x1_disabled() or not _conv3x3_enabled() or not use_hip(x):
        return False
    if x.dtype != torch.bfloat16:
        return False
    if conv.kernel_size != (3, 3) or conv.groups != 1:
        return False
    if conv.padding != (1, 1) or conv.dilation != (1, 1) or \
            conv.stride != (1, 1) or conv.bias is not None:
        return False
    if conv.in_channels % 64 != 0 or conv.out_channels % 64 != 0:
        return False
    return x.dim() == 4 and x.is_contiguous(memory_format=torch.channels_last)


def conv3x3_bn(x: torch.Tensor, conv: nn.Conv2d, bn) -> torch.Tensor:
    """Fused 3x3 conv + BatchNorm(+ReLU): train fuses the stats epilogue
    into the conv (no separate stats pass); eval is one kernel."""
    relu = bool(getattr(bn, "relu", False))
    B, C, H, W = x.shape
    if bn.running_mean.dtype != torch.float32:
        bn.running_mean.data = bn.running_mean.data.float()
        bn.running_var.data = bn.running_var.data.float()
    if not bn.training:
        rstd = torch.rsqrt(bn.running_var + bn.eps)
        scale = bn.weight.float() * rstd
        shift = bn.bias.float() - bn.running_mean * scale
        if not (torch.is_grad_enabled() and
                (x.requires_grad or conv.weight.requires_grad or
                 bn.weight.requires_grad)):
            xb = x.to(torch.bfloat16)
            wb = conv.weight.to(torch.bfloat16)
            w9 = wb.permute(0, 2, 3, 1).reshape(wb.shape[0], -1).contiguous()
            y2d, _ = ext().conv3x3_fwd(_flatten_nhwc(xb), w9, H, W, None,
                                       scale.contiguous(), shift.contiguous(),
                                       None, relu, False)
            return _unflatten_nhwc(y2d, B, H, W)
        y = _Conv3x3Fn.apply(x, conv.weight, False)
        y = y * scale.reshape(1, -1, 1, 1).to(y.dtype) + \
            shift.reshape(1, -1, 1, 1).to(y.dtype)
        return torch.relu(y) if relu else y
    if bn.num_batches_tracked is not None:
        bn.num_batches_tracked.add_(1)
    momentum = bn._eaf() if hasattr(bn, "_eaf") else (bn.momentum or 0.0)
    y_raw, partials = _Conv3x3Fn.apply(x, conv.weight, True)
    sums = partials.sum(dim=0)
    return _BNFromStatsFn.apply(y_raw, bn.weight, bn.bias, sums,
                                bn.running_mean, bn.running_var, momentum,
                                bn.eps, relu)
