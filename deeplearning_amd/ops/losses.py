"""Loss ops backed by HIP kernels: fused softmax-CE (label smoothing, soft
targets) and sigmoid focal loss.

Reference parity: nn.CrossEntropyLoss everywhere; timm
LabelSmoothingCrossEntropy / SoftTargetCrossEntropy (swin main.py:111-117);
sigmoid_focal_loss (RetinaNet network_files/losses.py:5-50, FCOS
models/loss.py:344-364).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import ext, use_hip


class _SoftmaxCEFn(torch.autograd.Function):
    """Mean CE in 2 kernels fwd / 1 kernel bwd: the forward kernel folds the
    loss sum + valid-row count into a stats[2] tensor (one division follows),
    and the backward kernel reads its grad scale from device memory — no host
    sync anywhere, so the loss is hipGraph-capturable."""

    @staticmethod
    def forward(ctx, logits, target, soft_target, smoothing, ignore_index):
        logits = logits.contiguous()
        stats, lse = ext().softmax_ce_fwd(logits, target, soft_target,
                                          smoothing, ignore_index)
        ctx.save_for_backward(logits, lse,
                              target if target is not None else torch.empty(0),
                              soft_target if soft_target is not None else torch.empty(0),
                              stats)
        ctx.smoothing = smoothing
        ctx.ignore_index = ignore_index
        ctx.has_target = target is not None
        return stats[0] / stats[1].clamp(min=1)

    @staticmethod
    def backward(ctx, dloss):
        logits, lse, target, soft_target, stats = ctx.saved_tensors
        grad_out = dloss.detach().reshape(1)
        if grad_out.dtype != torch.float32:
            grad_out = grad_out.float()
        if not grad_out.is_cuda:  # backward of a CPU-held scalar grad
            grad_out = grad_out.to(logits.device)
        dlogits = ext().softmax_ce_bwd(
            logits, target if ctx.has_target else None,
            soft_target if not ctx.has_target else None, lse, ctx.smoothing,
            ctx.ignore_index, grad_out.contiguous(), stats)
        return dlogits, None, None, None, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor,
                  smoothing: float = 0.0, ignore_index: int = -100) -> torch.Tensor:
    """Mean CE over valid rows. target: int64 class ids (B,)."""
    if use_hip(logits) and logits.dim() == 2:
        return _SoftmaxCEFn.apply(logits, target, None, smoothing, ignore_index)
    return F.cross_entropy(logits, target, label_smoothing=smoothing,
                           ignore_index=ignore_index)


def soft_target_cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean over batch of sum(-q * log_softmax(z)) (timm SoftTargetCrossEntropy)."""
    if use_hip(logits) and logits.dim() == 2:
        return _SoftmaxCEFn.apply(logits, None, target.contiguous(), 0.0, -100)
    return torch.sum(-target * F.log_softmax(logits, dim=-1), dim=-1).mean()


class CrossEntropyLoss(nn.Module):
    def __init__(self, smoothing: float = 0.0, ignore_index: int = -100):
        super().__init__()
        self.smoothing = smoothing
        self.ignore_index = ignore_index

    def forward(self, logits, target):
        if target.dtype in (torch.float16, torch.bfloat16, torch.float32) and target.dim() == 2:
            return soft_target_cross_entropy(logits, target)
        return cross_entropy(logits, target, self.smoothing, self.ignore_index)


class _FocalFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, alpha, gamma):
        logits = logits.contiguous()
        targets = targets.contiguous()
        loss = ext().focal_loss_fwd(logits, targets, alpha, gamma)
        ctx.save_for_backward(logits, targets)
        ctx.alpha, ctx.gamma = alpha, gamma
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets = ctx.saved_tensors
        dlogits = ext().focal_loss_bwd(dloss.contiguous(), logits, targets,
                                       ctx.alpha, ctx.gamma)
        return dlogits, None, None, None


def sigmoid_focal_loss(logits: torch.Tensor, targets: torch.Tensor,
                       alpha: float = 0.25, gamma: float = 2.0,
                       reduction: str = "none") -> torch.Tensor:
    """torchvision-semantics sigmoid focal loss. targets are 0/1 floats."""
    if use_hip(logits, targets):
        loss = _FocalFn.apply(logits, targets.to(logits.dtype), alpha, gamma)
    else:
        p = torch.sigmoid(logits)
        ce = F.binary_cross_entropy_with_logits(logits, targets.to(logits.dtype),
                                                reduction="none")
        p_t = p * targets + (1 - p) * (1 - targets)
        loss = ce * ((1 - p_t) ** gamma)
        if alpha >= 0:
            loss = loss * (alpha * targets + (1 - alpha) * (1 - targets))
    if reduction == "mean":
        return loss.mean()
    if reduction == "sum":
        return loss.sum()
    return loss
