"""Evaluation metrics: top-k accuracy, segmentation ConfusionMatrix (mIoU),
Dice coefficient.

Reference parity: accuracy (swin utils/torch_utils.py:325), ConfusionMatrix w/
all_reduce (Image_segmentation/FCN/utils/distributed_utils.py:73-123), Dice
(U-Net/loss/dice_score.py:5-40).
"""
from __future__ import annotations

import torch

from ..core.dist import is_dist, reduce_value


def accuracy(output: torch.Tensor, target: torch.Tensor, topk=(1,)):
    """Top-k accuracy in percent."""
    maxk = max(topk)
    _, pred = output.topk(maxk, dim=1, largest=True, sorted=True)
    pred = pred.t()
    correct = pred.eq(target.reshape(1, -1).expand_as(pred))
    return [correct[:k].reshape(-1).float().sum() * (100.0 / target.size(0))
            for k in topk]


class ConfusionMatrix:
    """num_classes x num_classes confusion matrix via the bincount trick."""

    def __init__(self, num_classes: int):
        self.num_classes = num_classes
        self.mat: torch.Tensor | None = None

    def update(self, target: torch.Tensor, pred: torch.Tensor):
        n = self.num_classes
        if self.mat is None:
            self.mat = torch.zeros((n, n), dtype=torch.int64, device=target.device)
        with torch.no_grad():
            k = (target >= 0) & (target < n)
            inds = n * target[k].to(torch.int64) + pred[k]
            self.mat += torch.bincount(inds, minlength=n**2).reshape(n, n)

    def reset(self):
        if self.mat is not None:
            self.mat.zero_()

    def reduce_from_all_processes(self):
        if self.mat is not None and is_dist():
            self.mat = reduce_value(self.mat, average=False)

    def compute(self):
        h = self.mat.float()
        acc_global = torch.diag(h).sum() / h.sum()
        acc = torch.diag(h) / h.sum(1).clamp(min=1)
        iu = torch.diag(h) / (h.sum(1) + h.sum(0) - torch.diag(h)).clamp(min=1)
        return acc_global, acc, iu

    def __str__(self):
        acc_global, acc, iu = self.compute()
        return (f"global acc: {acc_global.item()*100:.1f} | "
                f"mean IoU: {iu.mean().item()*100:.1f}")


def dice_coeff(pred: torch.Tensor, target: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """Average Dice over batch; pred/target are (N, ...) binary masks."""
    pred = pred.flatten(1).float()
    target = target.flatten(1).float()
    inter = (pred * target).sum(dim=1)
    denom = pred.sum(dim=1) + target.sum(dim=1)
    return ((2 * inter + eps) / (denom + eps)).mean()


def multiclass_dice_coeff(pred: torch.Tensor, target: torch.Tensor,
                          eps: float = 1e-6) -> torch.Tensor:
    """pred/target one-hot (N, C, ...)."""
    return dice_coeff(pred.flatten(0, 1), target.flatten(0, 1), eps)


def dice_loss(logits: torch.Tensor, target_onehot: torch.Tensor) -> torch.Tensor:
    probs = logits.softmax(dim=1) if logits.shape[1] > 1 else logits.sigmoid()
    return 1 - multiclass_dice_coeff(probs, target_onehot)


@torch.no_grad()
def count_flops(model, input_shape=(1, 3, 224, 224), device="cpu",
                inputs=None):
    """Forward-pass FLOPs (multiply-adds x2 — reference
    SwinTransformer.flops / fvcore convention) and parameter count.
    Thin front over core/complexity.estimate_macs (the single counting
    implementation; also covers Conv1d/3d/ConvTranspose)."""
    from ..core.complexity import count_params, estimate_macs

    x = inputs if inputs is not None else torch.randn(*input_shape,
                                                      device=device)
    return 2 * estimate_macs(model, x), count_params(model)
