"""RetinaNet: one-stage focal-loss detector on ResNet-FPN (P3-P7).

Reference parity: detection/RetinaNet/network_files/retinanet.py:23-480
(cls/reg heads :23-237, GIoU/L1 reg loss :153, forward :480, postprocess
:418-469) and losses.py:5-50 — re-designed: sigmoid focal loss is the
framework's HIP kernel (ops.sigmoid_focal_loss), NMS is the HIP batched-NMS.
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as F
from torch import nn

from ...ops import batched_nms, box_iou, sigmoid_focal_loss
from ..registry import register_model
from .anchors import AnchorGenerator, BoxCoder, Matcher
from .fpn import resnet_fpn_backbone
from .transform import GeneralizedRCNNTransform


class RetinaNetHead(nn.Module):
    def __init__(self, in_channels, num_anchors, num_classes):
        super().__init__()
        def tower():
            layers = []
            for _ in range(4):
                layers += [nn.Conv2d(in_channels, in_channels, 3, padding=1),
                           nn.ReLU(inplace=True)]
            return nn.Sequential(*layers)
        self.cls_tower = tower()
        self.bbox_tower = tower()
        self.cls_logits = nn.Conv2d(in_channels, num_anchors * num_classes,
                                    3, padding=1)
        self.bbox_pred = nn.Conv2d(in_channels, num_anchors * 4, 3, padding=1)
        self.num_classes = num_classes
        self.num_anchors = num_anchors
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.normal_(m.weight, std=0.01)
                nn.init.zeros_(m.bias)
        # focal-loss prior so background starts confident
        nn.init.constant_(self.cls_logits.bias, -math.log((1 - 0.01) / 0.01))

    def forward(self, features):
        cls_out, reg_out = [], []
        for f in features:
            cls = self.cls_logits(self.cls_tower(f))
            reg = self.bbox_pred(self.bbox_tower(f))
            N, _, H, W = cls.shape
            cls = cls.view(N, self.num_anchors, self.num_classes, H, W)
            cls = cls.permute(0, 3, 4, 1, 2).reshape(N, -1, self.num_classes)
            reg = reg.view(N, self.num_anchors, 4, H, W)
            reg = reg.permute(0, 3, 4, 1, 2).reshape(N, -1, 4)
            cls_out.append(cls)
            reg_out.append(reg)
        return torch.cat(cls_out, dim=1), torch.cat(reg_out, dim=1)


class RetinaNet(nn.Module):
    def __init__(self, num_classes=80, min_size=800, max_size=1333,
                 score_thresh=0.05, nms_thresh=0.5, detections_per_img=300,
                 topk_candidates=1000, fg_iou_thresh=0.5, bg_iou_thresh=0.4,
                 trainable_backbone_layers=3, reg_loss="l1"):
        super().__init__()
        self.backbone = resnet_fpn_backbone(
            returned_layers=(2, 3, 4), extra_blocks="p6p7",
            trainable_layers=trainable_backbone_layers)
        sizes = tuple((int(x), int(x * 2 ** (1 / 3)), int(x * 2 ** (2 / 3)))
                      for x in [32, 64, 128, 256, 512])
        ratios = ((0.5, 1.0, 2.0),) * len(sizes)
        self.anchor_generator = AnchorGenerator(sizes, ratios)
        num_anchors = self.anchor_generator.num_anchors_per_location()[0]
        self.head = RetinaNetHead(256, num_anchors, num_classes)
        self.box_coder = BoxCoder()
        self.matcher = Matcher(fg_iou_thresh, bg_iou_thresh,
                               allow_low_quality_matches=True)
        self.transform = GeneralizedRCNNTransform(min_size, max_size)
        self.score_thresh = score_thresh
        self.nms_thresh = nms_thresh
        self.detections_per_img = detections_per_img
        self.topk_candidates = topk_candidates
        self.num_classes = num_classes
        assert reg_loss in ("l1", "giou"), reg_loss
        self.reg_loss = reg_loss  # ref network_files/retinanet.py:153

    def compute_loss(self, targets, cls_logits, bbox_regression, anchors):
        cls_losses, reg_losses = [], []
        for i, t in enumerate(targets):
            anchors_i = anchors[i]
            if t["boxes"].numel() == 0:
                matched = torch.full((anchors_i.shape[0],), -1,
                                     dtype=torch.long,
                                     device=anchors_i.device)
            else:
                iou = box_iou(t["boxes"], anchors_i)
                matched = self.matcher(iou)
            fg = matched >= 0
            num_fg = int(fg.sum().clamp(min=1))
            gt_cls = torch.zeros_like(cls_logits[i])
            if fg.any():
                gt_cls[fg, t["labels"][matched[fg]]] = 1.0
            valid = matched != Matcher.BETWEEN
            cls_losses.append(sigmoid_focal_loss(
                cls_logits[i][valid], gt_cls[valid], alpha=0.25, gamma=2.0,
                reduction="sum") / num_fg)
            if fg.any():
                if self.reg_loss == "giou":
                    from ...ops.boxes import bbox_iou_aligned
                    pred = self.box_coder.decode(bbox_regression[i][fg],
                                                 anchors_i[fg])
                    g = bbox_iou_aligned(pred, t["boxes"][matched[fg]],
                                         xywh=False, GIoU=True).squeeze(-1)
                    reg_losses.append((1.0 - g).sum() / num_fg)
                else:
                    gt_deltas = self.box_coder.encode(
                        t["boxes"][matched[fg]], anchors_i[fg])
                    reg_losses.append(F.l1_loss(
                        bbox_regression[i][fg], gt_deltas,
                        reduction="sum") / num_fg)
            else:
                reg_losses.append(bbox_regression[i].sum() * 0)
        return {"classification": torch.stack(cls_losses).mean(),
                "bbox_regression": torch.stack(reg_losses).mean()}

    def postprocess_detections(self, cls_logits, bbox_regression, anchors,
                               image_sizes):
        detections = []
        for i, (h, w) in enumerate(image_sizes):
            scores_all = torch.sigmoid(cls_logits[i]).flatten()
            n = min(self.topk_candidates, scores_all.numel())
            scores, idxs = scores_all.topk(n)
            keep = scores > self.score_thresh
            scores, idxs = scores[keep], idxs[keep]
            anchor_idxs = idxs // self.num_classes
            labels = idxs % self.num_classes
            boxes = self.box_coder.decode(
                bbox_regression[i][anchor_idxs], anchors[i][anchor_idxs])
            boxes[:, 0::2].clamp_(0, w)
            boxes[:, 1::2].clamp_(0, h)
            keep = batched_nms(boxes, scores, labels, self.nms_thresh)
            keep = keep[:self.detections_per_img]
            detections.append({"boxes": boxes[keep], "scores": scores[keep],
                               "labels": labels[keep]})
        return detections

    def forward(self, images, targets=None):
        original_sizes = [tuple(img.shape[-2:]) for img in images]
        image_list, targets = self.transform(images, targets)
        feats = self.backbone(image_list.tensors)
        features = list(feats.values())
        cls_logits, bbox_regression = self.head(features)
        anchors = self.anchor_generator(image_list, features)
        if self.training:
            assert targets is not None
            return self.compute_loss(targets, cls_logits, bbox_regression,
                                     anchors)
        detections = self.postprocess_detections(
            cls_logits, bbox_regression, anchors, image_list.image_sizes)
        return self.transform.postprocess(detections, image_list.image_sizes,
                                          original_sizes)


@register_model
def retinanet_resnet50_fpn(num_classes=80, **kw):
    return RetinaNet(num_classes=num_classes, **kw)
