"""Box ops: IoU/GIoU/CIoU/DIoU, NMS, batched NMS, box coding helpers.

Reference parity: box_iou (RetinaNet network_files/boxes.py:38-154), bbox_iou
CIoU (yolov5 utils/metrics.py:239), GIoU (FCOS models/loss.py:388),
torchvision.ops.nms / batched_nms call sites (yolov5 utils/general.py:694,
fasterRcnn models/roi_head.py:326). GPU path: csrc/boxes.hip.
"""
from __future__ import annotations

import math

import torch

from ._ext import ext, use_hip


def box_area(boxes: torch.Tensor) -> torch.Tensor:
    return (boxes[:, 2] - boxes[:, 0]).clamp(min=0) * (boxes[:, 3] - boxes[:, 1]).clamp(min=0)


def _box_iou_eager(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    area1, area2 = box_area(a), box_area(b)
    lt = torch.max(a[:, None, :2], b[None, :, :2])
    rb = torch.min(a[:, None, 2:], b[None, :, 2:])
    wh = (rb - lt).clamp(min=0)
    inter = wh[..., 0] * wh[..., 1]
    union = area1[:, None] + area2[None, :] - inter
    return torch.where(union > 0, inter / union, torch.zeros_like(inter))


def box_iou(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Pairwise IoU matrix (N, M), xyxy boxes."""
    if use_hip(a, b):
        return ext().box_iou(a, b, False).to(a.dtype)
    return _box_iou_eager(a, b)


def generalized_box_iou(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    if use_hip(a, b):
        return ext().box_iou(a, b, True).to(a.dtype)
    iou = _box_iou_eager(a, b)
    lt = torch.min(a[:, None, :2], b[None, :, :2])
    rb = torch.max(a[:, None, 2:], b[None, :, 2:])
    wh = (rb - lt).clamp(min=0)
    carea = wh[..., 0] * wh[..., 1]
    area1, area2 = box_area(a), box_area(b)
    lt2 = torch.max(a[:, None, :2], b[None, :, :2])
    rb2 = torch.min(a[:, None, 2:], b[None, :, 2:])
    wh2 = (rb2 - lt2).clamp(min=0)
    union = area1[:, None] + area2[None, :] - wh2[..., 0] * wh2[..., 1]
    return torch.where(carea > 0, iou - (carea - union) / carea, iou)


def bbox_iou_aligned(box1: torch.Tensor, box2: torch.Tensor, xywh: bool = True,
                     GIoU: bool = False, DIoU: bool = False, CIoU: bool = False,
                     eps: float = 1e-7) -> torch.Tensor:
    """Element-aligned IoU of box1 vs box2 (yolov5 utils/metrics.py:bbox_iou
    semantics; differentiable, used inside losses)."""
    if xywh:
        (x1, y1, w1, h1), (x2, y2, w2, h2) = box1.chunk(4, -1), box2.chunk(4, -1)
        b1x1, b1x2, b1y1, b1y2 = x1 - w1 / 2, x1 + w1 / 2, y1 - h1 / 2, y1 + h1 / 2
        b2x1, b2x2, b2y1, b2y2 = x2 - w2 / 2, x2 + w2 / 2, y2 - h2 / 2, y2 + h2 / 2
    else:
        b1x1, b1y1, b1x2, b1y2 = box1.chunk(4, -1)
        b2x1, b2y1, b2x2, b2y2 = box2.chunk(4, -1)
        w1, h1 = b1x2 - b1x1, (b1y2 - b1y1).clamp(eps)
        w2, h2 = b2x2 - b2x1, (b2y2 - b2y1).clamp(eps)
    inter = (b1x2.minimum(b2x2) - b1x1.maximum(b2x1)).clamp(0) * \
            (b1y2.minimum(b2y2) - b1y1.maximum(b2y1)).clamp(0)
    union = w1 * h1 + w2 * h2 - inter + eps
    iou = inter / union
    if CIoU or DIoU or GIoU:
        cw = b1x2.maximum(b2x2) - b1x1.minimum(b2x1)
        ch = b1y2.maximum(b2y2) - b1y1.minimum(b2y1)
        if CIoU or DIoU:
            c2 = cw**2 + ch**2 + eps
            rho2 = ((b2x1 + b2x2 - b1x1 - b1x2) ** 2 +
                    (b2y1 + b2y2 - b1y1 - b1y2) ** 2) / 4
            if CIoU:
                v = (4 / math.pi**2) * (torch.atan(w2 / h2) - torch.atan(w1 / h1)).pow(2)
                with torch.no_grad():
                    alpha = v / (v - iou + (1 + eps))
                return iou - (rho2 / c2 + v * alpha)
            return iou - rho2 / c2
        c_area = cw * ch + eps
        return iou - (c_area - union) / c_area
    return iou


def _nms_eager(boxes: torch.Tensor, scores: torch.Tensor, iou_threshold: float) -> torch.Tensor:
    order = scores.argsort(descending=True)
    keep = []
    while order.numel() > 0:
        i = order[0]
        keep.append(i.item())
        if order.numel() == 1:
            break
        ious = _box_iou_eager(boxes[i].unsqueeze(0), boxes[order[1:]]).squeeze(0)
        order = order[1:][ious <= iou_threshold]
    return torch.tensor(keep, dtype=torch.long, device=boxes.device)


def nms(boxes: torch.Tensor, scores: torch.Tensor, iou_threshold: float) -> torch.Tensor:
    """torchvision.ops.nms semantics: returns kept indices, score-descending."""
    if boxes.numel() == 0:
        return torch.empty((0,), dtype=torch.long, device=boxes.device)
    if use_hip(boxes, scores):
        order = scores.argsort(descending=True)
        keep_sorted = ext().nms(boxes[order].contiguous(), iou_threshold)
        return order[keep_sorted]
    return _nms_eager(boxes, scores, iou_threshold)


def batched_nms(boxes: torch.Tensor, scores: torch.Tensor, idxs: torch.Tensor,
                iou_threshold: float) -> torch.Tensor:
    """Class-aware NMS via the coordinate-offset trick
    (RetinaNet network_files/boxes.py:38-75)."""
    if boxes.numel() == 0:
        return torch.empty((0,), dtype=torch.long, device=boxes.device)
    max_coord = boxes.max()
    offsets = idxs.to(boxes) * (max_coord + 1)
    return nms(boxes + offsets[:, None], scores, iou_threshold)


def clip_boxes_to_image(boxes: torch.Tensor, size) -> torch.Tensor:
    h, w = size
    boxes = boxes.clone()
    boxes[..., 0::2] = boxes[..., 0::2].clamp(min=0, max=w)
    boxes[..., 1::2] = boxes[..., 1::2].clamp(min=0, max=h)
    return boxes


def remove_small_boxes(boxes: torch.Tensor, min_size: float) -> torch.Tensor:
    ws = boxes[:, 2] - boxes[:, 0]
    hs = boxes[:, 3] - boxes[:, 1]
    return torch.where((ws >= min_size) & (hs >= min_size))[0]
