"""BatchNorm2d backed by the HIP kernels (csrc/batchnorm.hip): training-mode
batch statistics with optional fused ReLU, and FrozenBatchNorm2d.

Reference parity: nn.BatchNorm2d everywhere; FrozenBatchNorm2d
(detection/fasterRcnn/models/backbone/resnet50_fpn.py:5, FPN/fpn_model.py:8).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import dense, ext, use_hip


class _BNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, momentum, eps, relu):
        x = dense(x)
        y, mean, rstd = ext().batchnorm_fwd(
            x, weight, bias, running_mean, running_var, momentum, eps, relu)
        ctx.save_for_backward(x, y, weight, mean, rstd)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext().batchnorm_bwd(
            dy, x, y if ctx.relu else None, weight, mean, rstd, ctx.relu)
        return dx, dw, db, None, None, None, None, None


class BatchNorm2d(nn.BatchNorm2d):
    """Drop-in BatchNorm2d; optional fused ReLU via `relu=True`.

    GPU train mode uses the HIP kernel; eval mode uses the fused scale/shift
    apply kernel; CPU uses eager F.batch_norm. State-dict compatible with
    nn.BatchNorm2d.
    """

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False):
        super().__init__(num_features, eps=eps, momentum=momentum, affine=True,
                         track_running_stats=True)
        self.relu = relu

    def _eaf(self) -> float:
        """Exponential average factor; momentum=None means cumulative average
        (torch semantics — what SWA's update_bn relies on)."""
        if self.momentum is not None:
            return self.momentum
        if self.training and self.num_batches_tracked is not None:
            return 1.0 / float(max(int(self.num_batches_tracked), 1))
        return 0.0

    def forward(self, x):
        if not use_hip(x):
            if self.training and self.momentum is None and \
                    self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            y = F.batch_norm(x, self.running_mean, self.running_var, self.weight,
                             self.bias, self.training, self._eaf(), self.eps)
            return torch.relu(y) if self.relu else y
        if self.running_mean.dtype != torch.float32:
            # .to(bf16) on the module converts buffers; the HIP kernel keeps
            # running statistics in fp32 — restore them once.
            self.running_mean.data = self.running_mean.data.float()
            self.running_var.data = self.running_var.data.float()
        if self.training:
            if self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            return _BNFn.apply(x, self.weight, self.bias, self.running_mean,
                               self.running_var, self._eaf(), self.eps,
                               self.relu)
        rstd = torch.rsqrt(self.running_var.float() + self.eps)
        scale = self.weight.float() * rstd
        shift = self.bias.float() - self.running_mean.float() * scale
        if torch.is_grad_enabled() and (x.requires_grad or self.weight.requires_grad):
            y = x * scale.reshape(1, -1, 1, 1).to(x.dtype) + shift.reshape(1, -1, 1, 1).to(x.dtype)
            return torch.relu(y) if self.relu else y
        return ext().bn_apply(dense(x), scale, shift, self.relu)


class FrozenBatchNorm2d(nn.Module):
    """BatchNorm with fixed affine + stats (detection backbones)."""

    def __init__(self, num_features, eps=1e-5):
        super().__init__()
        self.eps = eps
        self.register_buffer("weight", torch.ones(num_features))
        self.register_buffer("bias", torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def _load_from_state_dict(self, state_dict, prefix, *args, **kwargs):
        state_dict.pop(prefix + "num_batches_tracked", None)
        super()._load_from_state_dict(state_dict, prefix, *args, **kwargs)

    def forward(self, x):
        rstd = torch.rsqrt(self.running_var.float() + self.eps)
        scale = self.weight.float() * rstd
        shift = self.bias.float() - self.running_mean.float() * scale
        if use_hip(x) and not (torch.is_grad_enabled() and x.requires_grad):
            return ext().bn_apply(dense(x), scale, shift, False)
        sc = scale.reshape(1, -1, 1, 1).to(x.dtype)
        sh = shift.reshape(1, -1, 1, 1).to(x.dtype)
        return x * sc + sh
