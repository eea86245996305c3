"""Anchor generation + box encode/decode + IoU matcher + pos/neg sampler.

Reference parity: detection/fasterRcnn/models/rpn_function.py
(AnchorsGenerator:25-204, cached grid anchors :157), utils/det_utils.py
(BoxCoder, Matcher, BalancedPositiveNegativeSampler) and
detection/RetinaNet/network_files/anchor_utils.py — re-designed: anchors are
generated on-device in one meshgrid pass per level and cached per
(grid, stride) key.
"""
from __future__ import annotations

import math

import torch
from torch import nn


class AnchorGenerator(nn.Module):
    def __init__(self, sizes=((128, 256, 512),), aspect_ratios=((0.5, 1.0, 2.0),)):
        super().__init__()
        if not isinstance(sizes[0], (list, tuple)):
            sizes = tuple((s,) for s in sizes)
        if not isinstance(aspect_ratios[0], (list, tuple)):
            aspect_ratios = (aspect_ratios,) * len(sizes)
        self.sizes = sizes
        self.aspect_ratios = aspect_ratios
        self._cache = {}

    def num_anchors_per_location(self):
        return [len(s) * len(a) for s, a in zip(self.sizes, self.aspect_ratios)]

    @staticmethod
    def _base_anchors(scales, aspect_ratios, dtype, device):
        scales = torch.as_tensor(scales, dtype=dtype, device=device)
        aspect_ratios = torch.as_tensor(aspect_ratios, dtype=dtype, device=device)
        h_ratios = torch.sqrt(aspect_ratios)
        w_ratios = 1 / h_ratios
        ws = (w_ratios[:, None] * scales[None, :]).view(-1)
        hs = (h_ratios[:, None] * scales[None, :]).view(-1)
        return (torch.stack([-ws, -hs, ws, hs], dim=1) / 2).round()

    def grid_anchors(self, grid_sizes, strides, dtype, device):
        anchors = []
        for size, stride, scales, ratios in zip(grid_sizes, strides,
                                                self.sizes, self.aspect_ratios):
            key = (size, stride, str(dtype), str(device))
            if key in self._cache:
                anchors.append(self._cache[key])
                continue
            gh, gw = size
            sy, sx = stride
            base = self._base_anchors(scales, ratios, dtype, device)
            shifts_x = torch.arange(0, gw, dtype=dtype, device=device) * sx
            shifts_y = torch.arange(0, gh, dtype=dtype, device=device) * sy
            shift_y, shift_x = torch.meshgrid(shifts_y, shifts_x, indexing="ij")
            shifts = torch.stack([shift_x.reshape(-1), shift_y.reshape(-1),
                                  shift_x.reshape(-1), shift_y.reshape(-1)],
                                 dim=1)
            a = (shifts[:, None] + base[None]).reshape(-1, 4)
            self._cache[key] = a
            anchors.append(a)
        return anchors

    def forward(self, image_list, feature_maps):
        grid_sizes = [tuple(f.shape[-2:]) for f in feature_maps]
        image_size = image_list.tensors.shape[-2:]
        dtype, device = feature_maps[0].dtype, feature_maps[0].device
        strides = [(image_size[0] // g[0], image_size[1] // g[1])
                   for g in grid_sizes]
        per_level = self.grid_anchors(grid_sizes, strides, torch.float32, device)
        anchors_all = torch.cat(per_level, dim=0)
        return [anchors_all for _ in range(image_list.tensors.shape[0])]


class BoxCoder:
    """Encode boxes as (dx,dy,dw,dh) deltas w.r.t. anchors and back."""

    def __init__(self, weights=(1.0, 1.0, 1.0, 1.0),
                 bbox_xform_clip=math.log(1000.0 / 16)):
        self.weights = weights
        self.bbox_xform_clip = bbox_xform_clip

    def encode(self, reference_boxes, anchors):
        wx, wy, ww, wh = self.weights
        ax = (anchors[:, 0] + anchors[:, 2]) * 0.5
        ay = (anchors[:, 1] + anchors[:, 3]) * 0.5
        aw = anchors[:, 2] - anchors[:, 0]
        ah = anchors[:, 3] - anchors[:, 1]
        gx = (reference_boxes[:, 0] + reference_boxes[:, 2]) * 0.5
        gy = (reference_boxes[:, 1] + reference_boxes[:, 3]) * 0.5
        gw = reference_boxes[:, 2] - reference_boxes[:, 0]
        gh = reference_boxes[:, 3] - reference_boxes[:, 1]
        return torch.stack([wx * (gx - ax) / aw, wy * (gy - ay) / ah,
                            ww * torch.log(gw / aw),
                            wh * torch.log(gh / ah)], dim=1)

    def decode(self, deltas, anchors):
        wx, wy, ww, wh = self.weights
        ax = (anchors[:, 0] + anchors[:, 2]) * 0.5
        ay = (anchors[:, 1] + anchors[:, 3]) * 0.5
        aw = anchors[:, 2] - anchors[:, 0]
        ah = anchors[:, 3] - anchors[:, 1]
        deltas = deltas.reshape(-1, 4)
        dx = deltas[:, 0] / wx
        dy = deltas[:, 1] / wy
        dw = torch.clamp(deltas[:, 2] / ww, max=self.bbox_xform_clip)
        dh = torch.clamp(deltas[:, 3] / wh, max=self.bbox_xform_clip)
        cx = dx * aw + ax
        cy = dy * ah + ay
        w = torch.exp(dw) * aw
        h = torch.exp(dh) * ah
        return torch.stack([cx - w / 2, cy - h / 2, cx + w / 2, cy + h / 2],
                           dim=1)


class Matcher:
    """Assign each anchor the best-IoU gt (thresholds -> fg/bg/ignore)."""

    BELOW_LOW = -1
    BETWEEN = -2

    def __init__(self, high_threshold, low_threshold,
                 allow_low_quality_matches=False):
        self.high = high_threshold
        self.low = low_threshold
        self.allow_low_quality_matches = allow_low_quality_matches

    def __call__(self, match_quality_matrix):
        # match_quality_matrix: [num_gt, num_anchors]
        matched_vals, matches = match_quality_matrix.max(dim=0)
        all_matches = matches.clone() if self.allow_low_quality_matches else None
        matches[matched_vals < self.low] = self.BELOW_LOW
        matches[(matched_vals >= self.low) & (matched_vals < self.high)] = \
            self.BETWEEN
        if self.allow_low_quality_matches:
            # for each gt, force-keep its best anchor(s)
            highest_per_gt = match_quality_matrix.max(dim=1).values
            idx = torch.where(
                match_quality_matrix == highest_per_gt[:, None])[1]
            matches[idx] = all_matches[idx]
        return matches


class BalancedPositiveNegativeSampler:
    def __init__(self, batch_size_per_image, positive_fraction):
        self.batch_size_per_image = batch_size_per_image
        self.positive_fraction = positive_fraction

    def __call__(self, matched_idxs_per_image):
        pos_masks, neg_masks = [], []
        for matched in matched_idxs_per_image:
            positive = torch.where(matched >= 1)[0]
            negative = torch.where(matched == 0)[0]
            num_pos = int(self.batch_size_per_image * self.positive_fraction)
            num_pos = min(positive.numel(), num_pos)
            num_neg = min(negative.numel(),
                          self.batch_size_per_image - num_pos)
            perm1 = torch.randperm(positive.numel(),
                                   device=positive.device)[:num_pos]
            perm2 = torch.randperm(negative.numel(),
                                   device=negative.device)[:num_neg]
            pos_mask = torch.zeros_like(matched, dtype=torch.bool)
            neg_mask = torch.zeros_like(matched, dtype=torch.bool)
            pos_mask[positive[perm1]] = True
            neg_mask[negative[perm2]] = True
            pos_masks.append(pos_mask)
            neg_masks.append(neg_mask)
        return pos_masks, neg_masks
