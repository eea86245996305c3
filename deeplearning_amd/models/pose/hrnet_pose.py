"""HRNet keypoint detection: heatmap regression head on the shared HRNet trunk,
KeypointToHeatMap target generation, focal heatmap loss, and heatmap->keypoint
decode with local-NMS.

Reference parity: pose_estimation/Insulator (models/hrnet.py:5-299,
KeypointToHeatMap train.py:143-156, Kploss_focal train.py:254, decode
utils/train_and_eval.py:136-217 + _nms:307) — re-designed on the shared
segmentation/hrnet.py trunk.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..registry import register_model
from ..segmentation.hrnet import HRNetTrunk


class HRNetPose(nn.Module):
    """Highest-resolution branch -> 1x1 conv heatmap head (1/4 input res)."""

    def __init__(self, width=18, num_joints=17):
        super().__init__()
        self.trunk = HRNetTrunk(width)
        self.head = nn.Conv2d(self.trunk.channels[0], num_joints, 1)

    def forward(self, x):
        return self.head(self.trunk(x)[0])


class KeypointToHeatMap:
    """Render gaussian heatmaps from keypoint coords (ref train.py:143-156).

    keypoints: [N, K, 2] in input-image pixels; output [N, K, H/4, W/4].
    """

    def __init__(self, heatmap_hw, gaussian_sigma=2.0, stride=4):
        self.hw = heatmap_hw
        self.sigma = gaussian_sigma
        self.stride = stride
        size = int(6 * gaussian_sigma + 3)
        x = torch.arange(size, dtype=torch.float32)
        y = x.unsqueeze(-1)
        c = size // 2
        self.kernel = torch.exp(-((x - c) ** 2 + (y - c) ** 2) /
                                (2 * gaussian_sigma ** 2))

    def __call__(self, keypoints: torch.Tensor,
                 visible: torch.Tensor | None = None) -> torch.Tensor:
        N, K, _ = keypoints.shape
        H, W = self.hw
        heatmap = torch.zeros(N, K, H, W)
        kr = self.kernel.shape[0] // 2
        for n in range(N):
            for k in range(K):
                if visible is not None and not bool(visible[n, k]):
                    continue
                cx = int(keypoints[n, k, 0] / self.stride + 0.5)
                cy = int(keypoints[n, k, 1] / self.stride + 0.5)
                if not (0 <= cx < W and 0 <= cy < H):
                    continue
                x0, x1 = max(0, cx - kr), min(W, cx + kr + 1)
                y0, y1 = max(0, cy - kr), min(H, cy + kr + 1)
                kx0, ky0 = x0 - (cx - kr), y0 - (cy - kr)
                patch = self.kernel[ky0:ky0 + (y1 - y0), kx0:kx0 + (x1 - x0)]
                heatmap[n, k, y0:y1, x0:x1] = torch.maximum(
                    heatmap[n, k, y0:y1, x0:x1], patch)
        return heatmap


def heatmap_focal_loss(pred: torch.Tensor, gt: torch.Tensor,
                       alpha=2.0, beta=4.0) -> torch.Tensor:
    """CornerNet-style focal loss on gaussian heatmaps (ref Kploss_focal)."""
    pred = torch.sigmoid(pred).clamp(1e-6, 1 - 1e-6)
    pos = gt.eq(1.0)
    pos_loss = -((1 - pred) ** alpha) * torch.log(pred) * pos
    neg_loss = -((1 - gt) ** beta) * (pred ** alpha) * torch.log(1 - pred) * (~pos)
    num_pos = pos.sum().clamp(min=1)
    return (pos_loss.sum() + neg_loss.sum()) / num_pos


def heatmap_nms(heat: torch.Tensor, kernel=3) -> torch.Tensor:
    """Keep local maxima only (ref utils/train_and_eval.py _nms:307)."""
    pad = (kernel - 1) // 2
    hmax = F.max_pool2d(heat, kernel, stride=1, padding=pad)
    return heat * (hmax == heat).to(heat.dtype)


def decode_heatmaps(heat: torch.Tensor, stride=4, refine=True):
    """[N,K,H,W] -> coords [N,K,2] (input pixels) + scores [N,K]
    (ref utils/train_and_eval.py:136-217). refine: quarter-pixel shift
    toward the higher neighbor (simple-baselines post-processing)."""
    prob = torch.sigmoid(heat)
    peaks = heatmap_nms(prob)
    N, K, H, W = peaks.shape
    flat = peaks.view(N, K, -1)
    scores, idx = flat.max(-1)
    ys = (idx // W).float()
    xs = (idx % W).float()
    if refine and H > 2 and W > 2:
        pf = prob.view(N, K, -1)
        xi = idx % W
        yi = idx // W
        inner_x = (xi > 0) & (xi < W - 1)
        inner_y = (yi > 0) & (yi < H - 1)
        g = lambda off, lo, hi: pf.gather(
            -1, (idx + off).clamp(min=lo, max=hi).unsqueeze(-1)).squeeze(-1)
        right, left = g(1, 0, H * W - 1), g(-1, 0, H * W - 1)
        down, up = g(W, 0, H * W - 1), g(-W, 0, H * W - 1)
        xs = xs + 0.25 * torch.sign(right - left) * inner_x.float()
        ys = ys + 0.25 * torch.sign(down - up) * inner_y.float()
    coords = torch.stack([xs, ys], dim=-1) * stride
    return coords, scores


@register_model
def hrnet_w18_pose(num_joints=17, **kw):
    return HRNetPose(width=18, num_joints=num_joints)


@register_model
def hrnet_w48_pose(num_joints=17, **kw):
    return HRNetPose(width=48, num_joints=num_joints)
