"""BDB: Batch-DropBlock network for metric learning / ReID, plus the
hard-mining triplet loss family.

Reference parity: metric_learning/BDB (models/, trainers/trainer.py:175
triplet+softmax criterion) and metric_learning/Happy-Whale
retrieval/models/triplet_loss.py:38-189 (hard mining, global/local loss) —
re-designed on this repo's ResNet.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..classification.resnet import Bottleneck, ResNet
from ..registry import register_model


class BatchDrop(nn.Module):
    """Drop the same spatial block across the whole batch (training only)."""

    def __init__(self, h_ratio=0.3, w_ratio=1.0):
        super().__init__()
        self.h_ratio = h_ratio
        self.w_ratio = w_ratio

    def forward(self, x):
        if not self.training:
            return x
        H, W = x.shape[-2:]
        rh, rw = round(self.h_ratio * H), round(self.w_ratio * W)
        sx = int(torch.randint(0, H - rh + 1, (1,)))
        sy = int(torch.randint(0, W - rw + 1, (1,)))
        mask = x.new_ones(x.shape[-2:])
        mask[sx:sx + rh, sy:sy + rw] = 0
        return x * mask


class BDBNetwork(nn.Module):
    """ResNet-50 trunk -> global branch (GAP+triplet/softmax) + part branch
    (BatchDrop + max-pool)."""

    def __init__(self, num_classes=751, feat_dim=512, h_ratio=0.3, w_ratio=1.0):
        super().__init__()
        trunk = ResNet(Bottleneck, [3, 4, 6, 3], include_top=False)
        # last stride 1 (standard ReID trick): rebuild layer4 with stride 1
        trunk.inplanes = 1024
        trunk.layer4 = trunk._make_layer(Bottleneck, 512, 3, stride=1)
        self.backbone = trunk

        self.global_pool = nn.AdaptiveAvgPool2d(1)
        self.global_reduce = nn.Sequential(
            nn.Linear(2048, feat_dim), nn.BatchNorm1d(feat_dim),
            nn.ReLU(inplace=True))
        self.global_softmax = nn.Linear(feat_dim, num_classes)

        self.part_drop = BatchDrop(h_ratio, w_ratio)
        self.part_pool = nn.AdaptiveMaxPool2d(1)
        self.part_reduce = nn.Sequential(
            nn.Linear(2048, feat_dim * 2), nn.BatchNorm1d(feat_dim * 2),
            nn.ReLU(inplace=True))
        self.part_softmax = nn.Linear(feat_dim * 2, num_classes)

    def forward(self, x):
        feat = self.backbone(x)
        g = self.global_reduce(self.global_pool(feat).flatten(1))
        p = self.part_reduce(self.part_pool(self.part_drop(feat)).flatten(1))
        if self.training:
            return {"global_feat": g, "part_feat": p,
                    "global_logits": self.global_softmax(g),
                    "part_logits": self.part_softmax(p)}
        return torch.cat([F.normalize(g, dim=1), F.normalize(p, dim=1)], dim=1)


def pairwise_dist(x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """Euclidean distance matrix (ref triplet_loss.py:21-37)."""
    d = x.pow(2).sum(1, keepdim=True) + y.pow(2).sum(1).unsqueeze(0) \
        - 2 * x @ y.t()
    return d.clamp(min=1e-12).sqrt()


def hard_example_mining(dist_mat: torch.Tensor, labels: torch.Tensor):
    """Per-anchor hardest positive / hardest negative (ref :38-90)."""
    same = labels.unsqueeze(0) == labels.unsqueeze(1)
    inf = torch.finfo(dist_mat.dtype).max
    dist_ap = dist_mat.masked_fill(~same, -inf).max(1).values
    dist_an = dist_mat.masked_fill(same, inf).min(1).values
    return dist_ap, dist_an


class TripletLoss(nn.Module):
    """Batch-hard triplet with margin (soft-margin if margin is None)."""

    def __init__(self, margin: float | None = 0.3):
        super().__init__()
        self.margin = margin

    def forward(self, feats, labels):
        dist = pairwise_dist(feats, feats)
        ap, an = hard_example_mining(dist, labels)
        if self.margin is None:
            return F.softplus(ap - an).mean()
        return F.relu(ap - an + self.margin).mean()


@register_model
def bdb_resnet50(num_classes=751, **kw):
    return BDBNetwork(num_classes=num_classes, **kw)
