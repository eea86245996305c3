"""Supervised contrastive learning: SupCon loss + projection-head model.

Reference parity: self-supervised/SupCon/losses/SupConLoss.py:5-109 —
re-designed (same math: anchor-vs-contrast log-softmax over the masked
similarity matrix, mean over positives).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..classification.resnet import ResNet, Bottleneck
from ..registry import register_model


class SupConLoss(nn.Module):
    def __init__(self, temperature=0.07, contrast_mode="all",
                 base_temperature=0.07):
        super().__init__()
        self.temperature = temperature
        self.contrast_mode = contrast_mode
        self.base_temperature = base_temperature

    def forward(self, features, labels=None, mask=None):
        """features: [B, n_views, D] (L2-normalized); labels: [B]."""
        B = features.shape[0]
        if labels is not None and mask is not None:
            raise ValueError("provide labels or mask, not both")
        if labels is None and mask is None:
            mask = torch.eye(B, dtype=torch.float32, device=features.device)
        elif labels is not None:
            labels = labels.contiguous().view(-1, 1)
            mask = torch.eq(labels, labels.T).float()
        else:
            mask = mask.float()

        n_views = features.shape[1]
        contrast = torch.cat(torch.unbind(features, dim=1), dim=0)  # B*V, D
        if self.contrast_mode == "one":
            anchor, n_anchor = features[:, 0], 1
        else:
            anchor, n_anchor = contrast, n_views

        logits = anchor @ contrast.T / self.temperature
        logits = logits - logits.max(dim=1, keepdim=True)[0].detach()

        mask = mask.repeat(n_anchor, n_views)
        logits_mask = 1.0 - torch.eye(
            mask.shape[0], mask.shape[1], device=mask.device)
        mask = mask * logits_mask

        exp_logits = torch.exp(logits) * logits_mask
        log_prob = logits - torch.log(exp_logits.sum(1, keepdim=True) + 1e-12)
        mask_sum = mask.sum(1).clamp(min=1e-8)
        mean_log_prob_pos = (mask * log_prob).sum(1) / mask_sum
        loss = -(self.temperature / self.base_temperature) * mean_log_prob_pos
        return loss.mean()


class SupConResNet(nn.Module):
    """ResNet-50 encoder + 2-layer MLP projection head."""

    def __init__(self, feat_dim=128, head="mlp"):
        super().__init__()
        self.encoder = ResNet(Bottleneck, [3, 4, 6, 3], include_top=False)
        self.pool = nn.AdaptiveAvgPool2d(1)
        dim_in = 2048
        if head == "linear":
            self.head = nn.Linear(dim_in, feat_dim)
        else:
            self.head = nn.Sequential(
                nn.Linear(dim_in, dim_in), nn.ReLU(inplace=True),
                nn.Linear(dim_in, feat_dim))

    def forward(self, x):
        feat = self.pool(self.encoder(x)).flatten(1)
        return F.normalize(self.head(feat), dim=1)


@register_model
def supcon_resnet50(**kw):
    return SupConResNet(**kw)
