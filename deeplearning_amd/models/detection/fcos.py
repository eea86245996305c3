"""FCOS: anchor-free per-pixel detection with centerness.

Reference parity: detection/FCOS (models/fcos.py FCOSDetector:85,
DetectHead:141, batched_nms:310; models/loss.py GenTargets:27-214,
focal/GIoU/centerness losses :250-414) — re-designed: focal loss is the
framework's HIP kernel; NMS the HIP batched-NMS; target generation is one
vectorized pass per level.
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F
from torch import nn

from ...ops import batched_nms, sigmoid_focal_loss
from ..registry import register_model
from .fpn import resnet_fpn_backbone
from .transform import GeneralizedRCNNTransform


class ScaleExp(nn.Module):
    def __init__(self, init=1.0):
        super().__init__()
        self.scale = nn.Parameter(torch.tensor(init))

    def forward(self, x):
        return torch.exp(x * self.scale)


class FCOSHead(nn.Module):
    def __init__(self, in_channels, num_classes, num_levels=5, prior=0.01):
        super().__init__()
        def tower():
            layers = []
            for _ in range(4):
                layers += [nn.Conv2d(in_channels, in_channels, 3, padding=1),
                           nn.GroupNorm(32, in_channels),
                           nn.ReLU(inplace=True)]
            return nn.Sequential(*layers)
        self.cls_tower = tower()
        self.reg_tower = tower()
        self.cls_logits = nn.Conv2d(in_channels, num_classes, 3, padding=1)
        self.reg_pred = nn.Conv2d(in_channels, 4, 3, padding=1)
        self.centerness = nn.Conv2d(in_channels, 1, 3, padding=1)
        self.scales = nn.ModuleList([ScaleExp() for _ in range(num_levels)])
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.normal_(m.weight, std=0.01)
                nn.init.zeros_(m.bias)
        nn.init.constant_(self.cls_logits.bias,
                          -math.log((1 - prior) / prior))

    def forward(self, features):
        cls_out, reg_out, ctr_out = [], [], []
        for i, f in enumerate(features):
            c = self.cls_tower(f)
            r = self.reg_tower(f)
            cls_out.append(self.cls_logits(c))
            ctr_out.append(self.centerness(r))
            reg_out.append(self.scales[i](self.reg_pred(r)))
        return cls_out, reg_out, ctr_out


def fcos_targets(points_per_level, strides, targets, num_classes,
                 ranges=((-1, 64), (64, 128), (128, 256), (256, 512),
                         (512, 1e8)), center_radius=1.5):
    """Assign (cls, ltrb, centerness) per location (ref models/loss.py:27-214)."""
    cls_t, reg_t = [], []
    for lvl, (pts, stride, rng) in enumerate(
            zip(points_per_level, strides, ranges)):
        xs, ys = pts[:, 0], pts[:, 1]
        per_img_cls, per_img_reg = [], []
        for t in targets:
            gt, labels = t["boxes"], t["labels"]
            if gt.numel() == 0:
                per_img_cls.append(torch.zeros(
                    pts.shape[0], dtype=torch.long, device=pts.device))
                per_img_reg.append(torch.zeros(pts.shape[0], 4,
                                               device=pts.device))
                continue
            l = xs[:, None] - gt[None, :, 0]
            t_ = ys[:, None] - gt[None, :, 1]
            r = gt[None, :, 2] - xs[:, None]
            b = gt[None, :, 3] - ys[:, None]
            ltrb = torch.stack([l, t_, r, b], dim=-1)  # P, G, 4
            inside = ltrb.min(-1).values > 0
            max_reg = ltrb.max(-1).values
            in_range = (max_reg >= rng[0]) & (max_reg <= rng[1])
            # center sampling
            cx = (gt[:, 0] + gt[:, 2]) / 2
            cy = (gt[:, 1] + gt[:, 3]) / 2
            radius = center_radius * stride
            near = (xs[:, None] - cx[None]).abs().le(radius) & \
                (ys[:, None] - cy[None]).abs().le(radius)
            mask = inside & in_range & near
            areas = (gt[:, 2] - gt[:, 0]) * (gt[:, 3] - gt[:, 1])
            area_mat = areas[None].expand(pts.shape[0], -1).clone()
            area_mat[~mask] = float("inf")
            min_area, min_idx = area_mat.min(1)
            pos = ~torch.isinf(min_area)
            cls = torch.zeros(pts.shape[0], dtype=torch.long,
                              device=pts.device)
            cls[pos] = labels[min_idx[pos]]
            reg = ltrb[torch.arange(pts.shape[0], device=pts.device), min_idx]
            reg[~pos] = 0
            per_img_cls.append(cls)
            per_img_reg.append(reg)
        cls_t.append(torch.stack(per_img_cls))
        reg_t.append(torch.stack(per_img_reg))
    return torch.cat(cls_t, dim=1), torch.cat(reg_t, dim=1)  # B,P  B,P,4


def centerness_target(reg):
    lr = reg[..., [0, 2]]
    tb = reg[..., [1, 3]]
    c = (lr.min(-1).values / lr.max(-1).values.clamp(min=1e-8)) * \
        (tb.min(-1).values / tb.max(-1).values.clamp(min=1e-8))
    return c.clamp(min=0).sqrt()


def giou_loss(pred_ltrb, gt_ltrb):
    """GIoU between two ltrb offsets at the same point (ref loss.py:388)."""
    p_area = (pred_ltrb[:, 0] + pred_ltrb[:, 2]) * \
        (pred_ltrb[:, 1] + pred_ltrb[:, 3])
    g_area = (gt_ltrb[:, 0] + gt_ltrb[:, 2]) * \
        (gt_ltrb[:, 1] + gt_ltrb[:, 3])
    w_i = torch.min(pred_ltrb[:, 0], gt_ltrb[:, 0]) + \
        torch.min(pred_ltrb[:, 2], gt_ltrb[:, 2])
    h_i = torch.min(pred_ltrb[:, 1], gt_ltrb[:, 1]) + \
        torch.min(pred_ltrb[:, 3], gt_ltrb[:, 3])
    inter = w_i.clamp(min=0) * h_i.clamp(min=0)
    union = p_area + g_area - inter
    iou = inter / union.clamp(min=1e-8)
    w_c = torch.max(pred_ltrb[:, 0], gt_ltrb[:, 0]) + \
        torch.max(pred_ltrb[:, 2], gt_ltrb[:, 2])
    h_c = torch.max(pred_ltrb[:, 1], gt_ltrb[:, 1]) + \
        torch.max(pred_ltrb[:, 3], gt_ltrb[:, 3])
    enclose = (w_c * h_c).clamp(min=1e-8)
    giou = iou - (enclose - union) / enclose
    return (1 - giou).mean() if pred_ltrb.numel() else pred_ltrb.sum()


class FCOS(nn.Module):
    strides = (8, 16, 32, 64, 128)

    def __init__(self, num_classes=80, min_size=800, max_size=1333,
                 score_thresh=0.05, nms_thresh=0.6, detections_per_img=100,
                 trainable_backbone_layers=3):
        super().__init__()
        self.transform = GeneralizedRCNNTransform(min_size, max_size)
        self.backbone = resnet_fpn_backbone(
            returned_layers=(2, 3, 4), extra_blocks="p6p7",
            trainable_layers=trainable_backbone_layers)
        self.head = FCOSHead(256, num_classes)
        self.num_classes = num_classes
        self.score_thresh = score_thresh
        self.nms_thresh = nms_thresh
        self.detections_per_img = detections_per_img

    @staticmethod
    def _points(features, strides):
        pts = []
        for f, s in zip(features, strides):
            H, W = f.shape[-2:]
            ys = (torch.arange(H, device=f.device, dtype=torch.float32) + 0.5) * s
            xs = (torch.arange(W, device=f.device, dtype=torch.float32) + 0.5) * s
            gy, gx = torch.meshgrid(ys, xs, indexing="ij")
            pts.append(torch.stack([gx.reshape(-1), gy.reshape(-1)], dim=1))
        return pts

    def forward(self, images, targets=None):
        original_sizes = [tuple(img.shape[-2:]) for img in images]
        image_list, targets = self.transform(images, targets)
        feats = list(self.backbone(image_list.tensors).values())
        cls_out, reg_out, ctr_out = self.head(feats)
        points = self._points(feats, self.strides)

        def flat(outs, c):
            return torch.cat([o.permute(0, 2, 3, 1).reshape(
                o.shape[0], -1, c) for o in outs], dim=1)
        cls_flat = flat(cls_out, self.num_classes)
        reg_flat = flat(reg_out, 4)
        ctr_flat = flat(ctr_out, 1).squeeze(-1)

        if self.training:
            cls_t, reg_t = fcos_targets(points, self.strides, targets,
                                        self.num_classes)
            pos = cls_t > 0
            num_pos = int(pos.sum().clamp(min=1))
            gt_onehot = torch.zeros_like(cls_flat)
            if pos.any():
                gt_onehot[pos] = F.one_hot(
                    cls_t[pos] - 1, self.num_classes).to(cls_flat.dtype)
            cls_loss = sigmoid_focal_loss(cls_flat, gt_onehot,
                                          reduction="sum") / num_pos
            if pos.any():
                ctr_t = centerness_target(reg_t[pos])
                reg_loss = giou_loss(reg_flat[pos], reg_t[pos])
                ctr_loss = F.binary_cross_entropy_with_logits(
                    ctr_flat[pos], ctr_t)
            else:
                reg_loss = reg_flat.sum() * 0
                ctr_loss = ctr_flat.sum() * 0
            return {"cls_loss": cls_loss, "reg_loss": reg_loss,
                    "centerness_loss": ctr_loss}

        # inference
        all_points = torch.cat(points)
        detections = []
        for i, (h, w) in enumerate(image_list.image_sizes):
            scores = torch.sigmoid(cls_flat[i]) * \
                torch.sigmoid(ctr_flat[i])[:, None]
            score_max, labels = scores.max(1)
            keep = score_max > self.score_thresh
            pts = all_points[keep]
            reg = reg_flat[i][keep]
            boxes = torch.stack([pts[:, 0] - reg[:, 0], pts[:, 1] - reg[:, 1],
                                 pts[:, 0] + reg[:, 2], pts[:, 1] + reg[:, 3]],
                                dim=1)
            boxes[:, 0::2].clamp_(0, w)
            boxes[:, 1::2].clamp_(0, h)
            s, l = score_max[keep], labels[keep]
            kept = batched_nms(boxes, s, l, self.nms_thresh)
            kept = kept[:self.detections_per_img]
            detections.append({"boxes": boxes[kept], "scores": s[kept],
                               "labels": l[kept] + 1})
        return self.transform.postprocess(detections, image_list.image_sizes,
                                          original_sizes)


@register_model
def fcos_resnet50_fpn(num_classes=80, **kw):
    return FCOS(num_classes=num_classes, **kw)
