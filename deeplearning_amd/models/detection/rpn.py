"""Region Proposal Network.

Reference parity: detection/fasterRcnn/models/rpn_function.py
(RPNHead:207-239, RegionProposalNetwork:304, assign_targets_to_anchors:375,
filter_proposals:454-522, compute_loss:523-565) — re-designed on the
framework's HIP batched-NMS and shared anchors.py utilities.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from ...ops import batched_nms, box_iou, clip_boxes_to_image, \
    remove_small_boxes
from .anchors import (BalancedPositiveNegativeSampler, BoxCoder, Matcher)


class RPNHead(nn.Module):
    def __init__(self, in_channels, num_anchors):
        super().__init__()
        self.conv = nn.Conv2d(in_channels, in_channels, 3, padding=1)
        self.cls_logits = nn.Conv2d(in_channels, num_anchors, 1)
        self.bbox_pred = nn.Conv2d(in_channels, num_anchors * 4, 1)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.normal_(m.weight, std=0.01)
                nn.init.zeros_(m.bias)

    def forward(self, features):
        logits, bbox_reg = [], []
        for f in features:
            t = F.relu(self.conv(f))
            logits.append(self.cls_logits(t))
            bbox_reg.append(self.bbox_pred(t))
        return logits, bbox_reg


def _flatten_head_outputs(cls_per_level, reg_per_level):
    """[N,A,H,W]/[N,A*4,H,W] per level -> [N, sum(HWA)] and [N, sum(HWA), 4]."""
    cls_flat, reg_flat = [], []
    for cls, reg in zip(cls_per_level, reg_per_level):
        N, A, H, W = cls.shape
        cls_flat.append(cls.permute(0, 2, 3, 1).reshape(N, -1))
        reg_flat.append(reg.view(N, A, 4, H, W).permute(0, 3, 4, 1, 2)
                        .reshape(N, -1, 4))
    return torch.cat(cls_flat, dim=1), torch.cat(reg_flat, dim=1)


class RegionProposalNetwork(nn.Module):
    def __init__(self, anchor_generator, head,
                 fg_iou_thresh=0.7, bg_iou_thresh=0.3,
                 batch_size_per_image=256, positive_fraction=0.5,
                 pre_nms_top_n=(2000, 1000), post_nms_top_n=(2000, 1000),
                 nms_thresh=0.7, min_size=1e-3):
        super().__init__()
        self.anchor_generator = anchor_generator
        self.head = head
        self.box_coder = BoxCoder()
        self.matcher = Matcher(fg_iou_thresh, bg_iou_thresh,
                               allow_low_quality_matches=True)
        self.sampler = BalancedPositiveNegativeSampler(
            batch_size_per_image, positive_fraction)
        self._pre_nms_top_n = pre_nms_top_n  # (train, test)
        self._post_nms_top_n = post_nms_top_n
        self.nms_thresh = nms_thresh
        self.min_size = min_size

    def pre_nms_top_n(self):
        return self._pre_nms_top_n[0 if self.training else 1]

    def post_nms_top_n(self):
        return self._post_nms_top_n[0 if self.training else 1]

    def filter_proposals(self, proposals, scores, image_sizes,
                         num_anchors_per_level):
        N = proposals.shape[0]
        device = proposals.device
        scores = scores.detach()
        # per-level top-n
        levels = torch.cat([
            torch.full((n,), i, dtype=torch.long, device=device)
            for i, n in enumerate(num_anchors_per_level)])
        top_idx = []
        offset = 0
        for n in num_anchors_per_level:
            k = min(self.pre_nms_top_n(), n)
            _, idx = scores[:, offset:offset + n].topk(k, dim=1)
            top_idx.append(idx + offset)
            offset += n
        top_idx = torch.cat(top_idx, dim=1)
        batch_idx = torch.arange(N, device=device)[:, None]
        scores = scores[batch_idx, top_idx]
        levels = levels[top_idx]
        proposals = proposals[batch_idx, top_idx]

        out_boxes, out_scores = [], []
        for i, (h, w) in enumerate(image_sizes):
            boxes = clip_boxes_to_image(proposals[i], (h, w))
            keep = remove_small_boxes(boxes, self.min_size)
            boxes, s, lvl = boxes[keep], scores[i][keep], levels[i][keep]
            keep = batched_nms(boxes, s, lvl, self.nms_thresh)
            keep = keep[:self.post_nms_top_n()]
            out_boxes.append(boxes[keep])
            out_scores.append(s[keep])
        return out_boxes, out_scores

    def assign_targets_to_anchors(self, anchors, targets):
        labels, matched_gt_boxes = [], []
        for anchors_i, t in zip(anchors, targets):
            gt = t["boxes"]
            if gt.numel() == 0:
                labels.append(torch.zeros(anchors_i.shape[0],
                                          device=anchors_i.device))
                matched_gt_boxes.append(torch.zeros_like(anchors_i))
                continue
            matched = self.matcher(box_iou(gt, anchors_i))
            matched_gt_boxes.append(gt[matched.clamp(min=0)])
            lbl = (matched >= 0).float()
            lbl[matched == Matcher.BELOW_LOW] = 0.0
            lbl[matched == Matcher.BETWEEN] = -1.0
            labels.append(lbl)
        return labels, matched_gt_boxes

    def compute_loss(self, objectness, pred_bbox_deltas, labels,
                     regression_targets):
        # sampler convention: >=1 pos, ==0 neg, <0 ignored — labels already are
        pos_masks, neg_masks = self.sampler([l.long() for l in labels])
        pos = torch.cat(pos_masks).nonzero().flatten()
        neg = torch.cat(neg_masks).nonzero().flatten()
        sampled = torch.cat([pos, neg])
        objectness = objectness.flatten()
        labels_cat = torch.cat(labels)
        regression_targets = torch.cat(regression_targets)
        box_loss = F.smooth_l1_loss(
            pred_bbox_deltas.reshape(-1, 4)[pos], regression_targets[pos],
            beta=1 / 9, reduction="sum") / max(sampled.numel(), 1)
        cls_loss = F.binary_cross_entropy_with_logits(
            objectness[sampled], labels_cat[sampled])
        return cls_loss, box_loss

    def forward(self, image_list, features, targets=None):
        features = list(features.values())
        cls_per_level, reg_per_level = self.head(features)
        anchors = self.anchor_generator(image_list, features)
        num_anchors_per_level = [c.shape[1] * c.shape[2] * c.shape[3]
                                 for c in cls_per_level]
        objectness, pred_bbox_deltas = _flatten_head_outputs(
            cls_per_level, reg_per_level)
        N = objectness.shape[0]
        with torch.no_grad():
            proposals = self.box_coder.decode(
                pred_bbox_deltas.reshape(-1, 4),
                torch.cat(anchors)).view(N, -1, 4)
        boxes, scores = self.filter_proposals(
            proposals, objectness, image_list.image_sizes,
            num_anchors_per_level)
        losses = {}
        if self.training:
            assert targets is not None
            labels, matched_gt = self.assign_targets_to_anchors(anchors,
                                                                targets)
            regression_targets = [
                self.box_coder.encode(m, a) if a.numel() else m
                for m, a in zip(matched_gt, anchors)]
            cls_loss, box_loss = self.compute_loss(
                objectness, pred_bbox_deltas, labels, regression_targets)
            losses = {"loss_objectness": cls_loss, "loss_rpn_box_reg": box_loss}
        return boxes, losses
