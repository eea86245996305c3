"""RoI heads: sample proposals, multi-scale RoIAlign, two-MLP head, Fast R-CNN
classification/regression losses, per-class NMS postprocess.

Reference parity: detection/fasterRcnn/models/roi_head.py:57-401
(select_training_samples:188, fastrcnn_loss:11-55, postprocess_detections
:248-338) and faster_rcnn.py TwoMLPHead:129 — re-designed on the framework's
HIP RoIAlign + batched-NMS kernels.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from ...ops import MultiScaleRoIAlign, batched_nms, box_iou
from .anchors import BalancedPositiveNegativeSampler, BoxCoder, Matcher


class TwoMLPHead(nn.Module):
    def __init__(self, in_channels, representation_size=1024):
        super().__init__()
        self.fc6 = nn.Linear(in_channels, representation_size)
        self.fc7 = nn.Linear(representation_size, representation_size)

    def forward(self, x):
        x = x.flatten(start_dim=1)
        return F.relu(self.fc7(F.relu(self.fc6(x))))


class FastRCNNPredictor(nn.Module):
    def __init__(self, in_channels, num_classes):
        super().__init__()
        self.cls_score = nn.Linear(in_channels, num_classes)
        self.bbox_pred = nn.Linear(in_channels, num_classes * 4)

    def forward(self, x):
        return self.cls_score(x), self.bbox_pred(x)


def fastrcnn_loss(class_logits, box_regression, labels, regression_targets):
    labels = torch.cat(labels)
    regression_targets = torch.cat(regression_targets)
    classification_loss = F.cross_entropy(class_logits, labels)
    pos = torch.where(labels > 0)[0]
    labels_pos = labels[pos]
    N = class_logits.shape[0]
    box_regression = box_regression.reshape(N, -1, 4)
    box_loss = F.smooth_l1_loss(
        box_regression[pos, labels_pos], regression_targets[pos],
        beta=1 / 9, reduction="sum") / labels.numel()
    return classification_loss, box_loss


class RoIHeads(nn.Module):
    def __init__(self, num_classes, out_channels=256,
                 fg_iou_thresh=0.5, bg_iou_thresh=0.5,
                 batch_size_per_image=512, positive_fraction=0.25,
                 score_thresh=0.05, nms_thresh=0.5, detections_per_img=100,
                 featmap_names=("0", "1", "2", "3"), output_size=7,
                 sampling_ratio=2):
        super().__init__()
        self.box_roi_pool = MultiScaleRoIAlign(
            featmap_names=list(featmap_names), output_size=output_size,
            sampling_ratio=sampling_ratio)
        self.box_head = TwoMLPHead(out_channels * output_size * output_size)
        self.box_predictor = FastRCNNPredictor(1024, num_classes)
        self.matcher = Matcher(fg_iou_thresh, bg_iou_thresh)
        self.sampler = BalancedPositiveNegativeSampler(
            batch_size_per_image, positive_fraction)
        self.box_coder = BoxCoder(weights=(10.0, 10.0, 5.0, 5.0))
        self.score_thresh = score_thresh
        self.nms_thresh = nms_thresh
        self.detections_per_img = detections_per_img
        self.num_classes = num_classes

    def select_training_samples(self, proposals, targets):
        # append gt boxes so every gt has at least one positive
        proposals = [torch.cat([p, t["boxes"]])
                     for p, t in zip(proposals, targets)]
        matched_idxs, labels = [], []
        for p, t in zip(proposals, targets):
            gt = t["boxes"]
            if gt.numel() == 0:
                matched_idxs.append(torch.zeros(p.shape[0], dtype=torch.long,
                                                device=p.device))
                labels.append(torch.zeros(p.shape[0], dtype=torch.long,
                                          device=p.device))
                continue
            matched = self.matcher(box_iou(gt, p))
            clamped = matched.clamp(min=0)
            lbl = t["labels"][clamped]
            lbl[matched == Matcher.BELOW_LOW] = 0
            lbl[matched == Matcher.BETWEEN] = -1
            matched_idxs.append(clamped)
            labels.append(lbl)
        # sampler convention: >=1 pos, ==0 neg, <0 ignored
        pos_masks, neg_masks = self.sampler(
            [(l > 0).long() - (l < 0).long() for l in labels])
        result_props, result_labels, regression_targets, matched_gt = \
            [], [], [], []
        for i, (p, t) in enumerate(zip(proposals, targets)):
            sampled = torch.where(pos_masks[i] | neg_masks[i])[0]
            props = p[sampled]
            lbl = labels[i][sampled].clamp(min=0)
            gt = t["boxes"]
            if gt.numel() == 0:
                reg = torch.zeros_like(props)
            else:
                matched_boxes = gt[matched_idxs[i][sampled]]
                reg = self.box_coder.encode(matched_boxes, props)
            result_props.append(props)
            result_labels.append(lbl)
            regression_targets.append(reg)
        return result_props, result_labels, regression_targets

    def postprocess_detections(self, class_logits, box_regression, proposals,
                               image_sizes):
        device = class_logits.device
        boxes_per_image = [p.shape[0] for p in proposals]
        pred_boxes = self.box_coder.decode(
            box_regression.reshape(-1, 4),
            torch.cat(proposals).repeat_interleave(self.num_classes, dim=0))
        pred_boxes = pred_boxes.view(-1, self.num_classes, 4)
        pred_scores = F.softmax(class_logits, -1)
        pred_boxes = pred_boxes.split(boxes_per_image, 0)
        pred_scores = pred_scores.split(boxes_per_image, 0)

        results = []
        for boxes, scores, (h, w) in zip(pred_boxes, pred_scores, image_sizes):
            boxes = boxes[:, 1:]  # drop background
            scores = scores[:, 1:]
            labels = torch.arange(1, self.num_classes, device=device)
            labels = labels.view(1, -1).expand_as(scores)
            boxes = boxes.reshape(-1, 4)
            scores = scores.reshape(-1)
            labels = labels.reshape(-1)
            boxes[:, 0::2].clamp_(0, w)
            boxes[:, 1::2].clamp_(0, h)
            keep = scores > self.score_thresh
            boxes, scores, labels = boxes[keep], scores[keep], labels[keep]
            ws = boxes[:, 2] - boxes[:, 0]
            hs = boxes[:, 3] - boxes[:, 1]
            keep = (ws > 1e-2) & (hs > 1e-2)
            boxes, scores, labels = boxes[keep], scores[keep], labels[keep]
            keep = batched_nms(boxes, scores, labels, self.nms_thresh)
            keep = keep[:self.detections_per_img]
            results.append({"boxes": boxes[keep], "scores": scores[keep],
                            "labels": labels[keep]})
        return results

    def forward(self, features, proposals, image_sizes, targets=None):
        if self.training:
            proposals, labels, regression_targets = \
                self.select_training_samples(proposals, targets)
        box_features = self.box_roi_pool(features, proposals, image_sizes)
        box_features = self.box_head(box_features)
        class_logits, box_regression = self.box_predictor(box_features)
        if self.training:
            loss_classifier, loss_box_reg = fastrcnn_loss(
                class_logits, box_regression, labels, regression_targets)
            return [], {"loss_classifier": loss_classifier,
                        "loss_box_reg": loss_box_reg}
        return self.postprocess_detections(class_logits, box_regression,
                                           proposals, image_sizes), {}
