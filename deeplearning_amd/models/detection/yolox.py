"""YOLOX: anchor-free YOLO with decoupled head and SimOTA assignment.

Reference parity: detection/YOLOX (yolox/models/darknet.py CSPDarknet,
yolo_pafpn.py, yolo_head.py YOLOXHead.get_losses:254, SimOTA
get_assignments:426 + dynamic_k_matching:608, decode_outputs:237) —
re-designed: BN+SiLU via the framework's HIP kernels; the SimOTA cost matrix
is built in one batched pass per image on-device.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import batched_nms, box_iou
from ..registry import register_model
from .yolov5 import Conv, SPPF


class Focus(nn.Module):
    """Space-to-depth stem (ref darknet.py Focus)."""

    def __init__(self, c1, c2, k=3):
        super().__init__()
        self.conv = Conv(c1 * 4, c2, k, 1)

    def forward(self, x):
        return self.conv(torch.cat([
            x[..., ::2, ::2], x[..., 1::2, ::2],
            x[..., ::2, 1::2], x[..., 1::2, 1::2]], dim=1))


class CSPBottleneck(nn.Module):
    def __init__(self, c, shortcut=True, e=0.5):
        super().__init__()
        c_ = int(c * e)
        self.cv1 = Conv(c, c_, 1)
        self.cv2 = Conv(c_, c, 3)
        self.add = shortcut

    def forward(self, x):
        y = self.cv2(self.cv1(x))
        return x + y if self.add else y


class CSPLayer(nn.Module):
    def __init__(self, c1, c2, n=1, shortcut=True, e=0.5):
        super().__init__()
        c_ = int(c2 * e)
        self.cv1 = Conv(c1, c_, 1)
        self.cv2 = Conv(c1, c_, 1)
        self.cv3 = Conv(2 * c_, c2, 1)
        self.m = nn.Sequential(*[CSPBottleneck(c_, shortcut)
                                 for _ in range(n)])

    def forward(self, x):
        return self.cv3(torch.cat([self.m(self.cv1(x)), self.cv2(x)], 1))


class CSPDarknet(nn.Module):
    def __init__(self, dep_mul=0.33, wid_mul=0.50):
        super().__init__()
        base_c = int(wid_mul * 64)
        base_d = max(round(dep_mul * 3), 1)
        self.stem = Focus(3, base_c)
        self.dark2 = nn.Sequential(
            Conv(base_c, base_c * 2, 3, 2),
            CSPLayer(base_c * 2, base_c * 2, base_d))
        self.dark3 = nn.Sequential(
            Conv(base_c * 2, base_c * 4, 3, 2),
            CSPLayer(base_c * 4, base_c * 4, base_d * 3))
        self.dark4 = nn.Sequential(
            Conv(base_c * 4, base_c * 8, 3, 2),
            CSPLayer(base_c * 8, base_c * 8, base_d * 3))
        self.dark5 = nn.Sequential(
            Conv(base_c * 8, base_c * 16, 3, 2),
            SPPF(base_c * 16, base_c * 16),
            CSPLayer(base_c * 16, base_c * 16, base_d, shortcut=False))

    def forward(self, x):
        x = self.dark2(self.stem(x))
        c3 = self.dark3(x)
        c4 = self.dark4(c3)
        c5 = self.dark5(c4)
        return c3, c4, c5


class YOLOPAFPN(nn.Module):
    def __init__(self, dep_mul=0.33, wid_mul=0.50):
        super().__init__()
        self.backbone = CSPDarknet(dep_mul, wid_mul)
        c = int(wid_mul * 64)
        d = max(round(dep_mul * 3), 1)
        self.lateral5 = Conv(c * 16, c * 8, 1)
        self.c3_p4 = CSPLayer(c * 16, c * 8, d, shortcut=False)
        self.lateral4 = Conv(c * 8, c * 4, 1)
        self.c3_p3 = CSPLayer(c * 8, c * 4, d, shortcut=False)
        self.down3 = Conv(c * 4, c * 4, 3, 2)
        self.c3_n3 = CSPLayer(c * 8, c * 8, d, shortcut=False)
        self.down4 = Conv(c * 8, c * 8, 3, 2)
        self.c3_n4 = CSPLayer(c * 16, c * 16, d, shortcut=False)
        self.out_channels = (c * 4, c * 8, c * 16)

    def forward(self, x):
        c3, c4, c5 = self.backbone(x)
        p5 = self.lateral5(c5)
        p4 = self.c3_p4(torch.cat(
            [F.interpolate(p5, scale_factor=2, mode="nearest"), c4], 1))
        p4l = self.lateral4(p4)
        p3 = self.c3_p3(torch.cat(
            [F.interpolate(p4l, scale_factor=2, mode="nearest"), c3], 1))
        n3 = self.c3_n3(torch.cat([self.down3(p3), p4l], 1))
        n4 = self.c3_n4(torch.cat([self.down4(n3), p5], 1))
        return p3, n3, n4


class YOLOXHead(nn.Module):
    strides = (8, 16, 32)

    def __init__(self, num_classes, in_channels, width=1.0):
        super().__init__()
        self.nc = num_classes
        feat_c = int(256 * width)
        self.stems = nn.ModuleList()
        self.cls_convs = nn.ModuleList()
        self.reg_convs = nn.ModuleList()
        self.cls_preds = nn.ModuleList()
        self.reg_preds = nn.ModuleList()
        self.obj_preds = nn.ModuleList()
        for c in in_channels:
            self.stems.append(Conv(c, feat_c, 1))
            self.cls_convs.append(nn.Sequential(
                Conv(feat_c, feat_c, 3), Conv(feat_c, feat_c, 3)))
            self.reg_convs.append(nn.Sequential(
                Conv(feat_c, feat_c, 3), Conv(feat_c, feat_c, 3)))
            self.cls_preds.append(nn.Conv2d(feat_c, num_classes, 1))
            self.reg_preds.append(nn.Conv2d(feat_c, 4, 1))
            self.obj_preds.append(nn.Conv2d(feat_c, 1, 1))

    def forward(self, feats):
        outputs = []
        for i, f in enumerate(feats):
            s = self.stems[i](f)
            cls_feat = self.cls_convs[i](s)
            reg_feat = self.reg_convs[i](s)
            out = torch.cat([self.reg_preds[i](reg_feat),
                             self.obj_preds[i](reg_feat),
                             self.cls_preds[i](cls_feat)], dim=1)
            outputs.append(out)
        return outputs  # per level: [B, 5+nc, H, W]


def _grids_strides(outputs, strides, device):
    grids, stride_t = [], []
    for out, s in zip(outputs, strides):
        H, W = out.shape[-2:]
        gy, gx = torch.meshgrid(torch.arange(H, device=device),
                                torch.arange(W, device=device), indexing="ij")
        grids.append(torch.stack([gx, gy], -1).reshape(-1, 2))
        stride_t.append(torch.full((H * W, 1), s, device=device,
                                   dtype=torch.float32))
    return torch.cat(grids).float(), torch.cat(stride_t)


def decode_outputs(outputs, strides):
    """[B, 5+nc, H, W] per level -> [B, P, 5+nc] with xy/wh in pixels
    (ref yolo_head.py decode_outputs:237)."""
    device = outputs[0].device
    flat = [o.flatten(2).permute(0, 2, 1) for o in outputs]
    out = torch.cat(flat, dim=1)
    grids, stride_t = _grids_strides(outputs, strides, device)
    xy = (out[..., :2] + grids) * stride_t
    wh = torch.exp(out[..., 2:4]) * stride_t
    return torch.cat([xy, wh, out[..., 4:]], dim=-1)


class YOLOX(nn.Module):
    def __init__(self, num_classes=80, dep_mul=0.33, wid_mul=0.50):
        super().__init__()
        self.backbone = YOLOPAFPN(dep_mul, wid_mul)
        self.head = YOLOXHead(num_classes, self.backbone.out_channels,
                              wid_mul)
        self.nc = num_classes

    def forward(self, x, targets=None):
        outputs = self.head(self.backbone(x))
        if self.training:
            assert targets is not None
            return yolox_loss(outputs, targets, self.nc,
                              YOLOXHead.strides,
                              use_l1=getattr(self, "use_l1", False))
        return decode_outputs(outputs, YOLOXHead.strides)


# ------------------------------------------------------------------ SimOTA
@torch.no_grad()
def simota_assign(pred_boxes, pred_cls, pred_obj, gt_boxes, gt_labels,
                  grids, stride_t, num_classes, center_radius=2.5):
    """SimOTA label assignment for one image
    (ref yolo_head.py get_assignments:426 + dynamic_k_matching:608).

    pred_boxes: [P,4] cxcywh pixels; gt_boxes: [G,4] xyxy pixels.
    Returns fg_mask [P] bool, matched_gt [num_fg], ious [num_fg].
    """
    P = pred_boxes.shape[0]
    G = gt_boxes.shape[0]
    stride_t = stride_t.reshape(-1, 1)
    centers = grids * stride_t + stride_t * 0.5  # anchor-point centers
    # candidate filter: center inside gt box or inside center_radius cells
    x, y = centers[:, 0], centers[:, 1]
    in_box = (x[:, None] > gt_boxes[None, :, 0]) & \
        (x[:, None] < gt_boxes[None, :, 2]) & \
        (y[:, None] > gt_boxes[None, :, 1]) & \
        (y[:, None] < gt_boxes[None, :, 3])
    gcx = (gt_boxes[:, 0] + gt_boxes[:, 2]) / 2
    gcy = (gt_boxes[:, 1] + gt_boxes[:, 3]) / 2
    r = center_radius * stride_t
    in_center = ((x[:, None] - gcx[None]).abs() < r) & \
        ((y[:, None] - gcy[None]).abs() < r)
    candidate = in_box.any(1) | in_center.any(1)
    both = in_box[candidate] & in_center[candidate]

    if candidate.sum() == 0 or G == 0:
        return torch.zeros(P, dtype=torch.bool, device=pred_boxes.device), \
            torch.zeros(0, dtype=torch.long, device=pred_boxes.device), \
            torch.zeros(0, device=pred_boxes.device)

    cand_boxes = pred_boxes[candidate]
    xyxy = torch.cat([cand_boxes[:, :2] - cand_boxes[:, 2:] / 2,
                      cand_boxes[:, :2] + cand_boxes[:, 2:] / 2], dim=1)
    ious = box_iou(gt_boxes, xyxy)  # G, C
    iou_cost = -torch.log(ious + 1e-8)

    # fp32 + autocast off: BCE-on-probabilities is rejected inside autocast
    # (the reference computes this cost block with amp disabled too,
    # yolo_head.py get_assignments)
    with torch.autocast(pred_cls.device.type, enabled=False):
        cls_prob = (pred_cls[candidate].float().sigmoid() *
                    pred_obj[candidate].float().sigmoid()[:, None]).sqrt()
        gt_onehot = F.one_hot(gt_labels, num_classes).float()  # G, nc
        cls_cost = F.binary_cross_entropy(
            cls_prob[None].expand(G, -1, -1),
            gt_onehot[:, None].expand(-1, cls_prob.shape[0], -1),
            reduction="none").sum(-1)
    cost = cls_cost + 3.0 * iou_cost + 100000.0 * (~both.T)

    # dynamic-k: top-10 IoU sum per gt
    k = min(10, ious.shape[1])
    topk_ious, _ = ious.topk(k, dim=1)
    dynamic_ks = topk_ious.sum(1).int().clamp(min=1)
    matching = torch.zeros_like(cost, dtype=torch.bool)
    for g in range(G):
        _, pos = cost[g].topk(int(dynamic_ks[g]), largest=False)
        matching[g, pos] = True
    # resolve anchors claimed by >1 gt: keep min-cost
    multi = matching.sum(0) > 1
    if multi.any():
        best = cost[:, multi].argmin(0)
        matching[:, multi] = False
        matching[best, multi] = True
    fg_cand = matching.any(0)
    matched_gt = matching[:, fg_cand].float().argmax(0)
    fg_mask = torch.zeros(P, dtype=torch.bool, device=pred_boxes.device)
    idx = torch.where(candidate)[0][fg_cand]
    fg_mask[idx] = True
    matched_ious = ious[matched_gt, torch.arange(matched_gt.shape[0],
                                                 device=ious.device)]
    return fg_mask, matched_gt, matched_ious


def yolox_loss(outputs, targets, num_classes, strides, use_l1=False):
    """targets: list of dicts with 'boxes' (xyxy pixels) and 'labels'.
    use_l1 adds the raw-output L1 regression term the reference enables for
    the no-aug tail epochs (yolo_head.py get_losses/get_l1_target)."""
    device = outputs[0].device
    flat = [o.flatten(2).permute(0, 2, 1) for o in outputs]
    out = torch.cat(flat, dim=1)  # B, P, 5+nc
    grids, stride_t = _grids_strides(outputs, strides, device)
    xy = (out[..., :2] + grids) * stride_t
    wh = torch.exp(out[..., 2:4].clamp(max=8)) * stride_t
    pred_boxes = torch.cat([xy, wh], dim=-1)  # cxcywh
    pred_obj = out[..., 4]
    pred_cls = out[..., 5:]

    B = out.shape[0]
    num_fg_total = 0
    loss_iou = out.new_zeros(())
    loss_obj = out.new_zeros(())
    loss_cls = out.new_zeros(())
    loss_l1 = out.new_zeros(())
    for b in range(B):
        t = targets[b]
        gt, labels = t["boxes"], t["labels"]
        fg_mask, matched_gt, _ = simota_assign(
            pred_boxes[b], pred_cls[b], pred_obj[b], gt, labels,
            grids, stride_t.squeeze(-1), num_classes)
        obj_target = fg_mask.float()
        loss_obj = loss_obj + F.binary_cross_entropy_with_logits(
            pred_obj[b], obj_target, reduction="sum")
        num_fg = int(fg_mask.sum())
        num_fg_total += num_fg
        if num_fg == 0:
            continue
        pb = pred_boxes[b][fg_mask]
        gb = gt[matched_gt]
        pb_xyxy = torch.cat([pb[:, :2] - pb[:, 2:] / 2,
                             pb[:, :2] + pb[:, 2:] / 2], 1)
        # IoU loss (aligned)
        lt = torch.max(pb_xyxy[:, :2], gb[:, :2])
        rb = torch.min(pb_xyxy[:, 2:], gb[:, 2:])
        inter = (rb - lt).clamp(min=0).prod(1)
        area_p = (pb_xyxy[:, 2:] - pb_xyxy[:, :2]).clamp(min=0).prod(1)
        area_g = (gb[:, 2:] - gb[:, :2]).clamp(min=0).prod(1)
        iou = inter / (area_p + area_g - inter + 1e-8)
        loss_iou = loss_iou + (1 - iou ** 2).sum()
        cls_target = F.one_hot(labels[matched_gt], num_classes).float() * \
            iou.detach()[:, None]
        loss_cls = loss_cls + F.binary_cross_entropy_with_logits(
            pred_cls[b][fg_mask], cls_target, reduction="sum")
        if use_l1:
            # raw-output-space target (ref get_l1_target): t_xy = gt_c/stride
            # - grid, t_wh = log(gt_wh/stride)
            st = stride_t.squeeze(-1)[fg_mask, None]
            g_c = (gb[:, :2] + gb[:, 2:]) / 2
            g_wh = (gb[:, 2:] - gb[:, :2]).clamp(min=1e-8)
            l1_t = torch.cat([g_c / st - grids[fg_mask],
                              torch.log(g_wh / st)], 1)
            loss_l1 = loss_l1 + F.l1_loss(out[b][fg_mask][:, :4], l1_t,
                                          reduction="sum")
    n = max(num_fg_total, 1)
    losses = {"iou_loss": 5.0 * loss_iou / n, "obj_loss": loss_obj / n,
              "cls_loss": loss_cls / n}
    if use_l1:
        losses["l1_loss"] = loss_l1 / n
    return losses


def yolox_postprocess(decoded, num_classes, conf_thre=0.25, nms_thre=0.45):
    """[B,P,5+nc] -> per-image detections dict (ref yolox/utils/boxes.py)."""
    box_corner = decoded.new_empty(decoded.shape[0], decoded.shape[1], 4)
    box_corner[..., 0] = decoded[..., 0] - decoded[..., 2] / 2
    box_corner[..., 1] = decoded[..., 1] - decoded[..., 3] / 2
    box_corner[..., 2] = decoded[..., 0] + decoded[..., 2] / 2
    box_corner[..., 3] = decoded[..., 1] + decoded[..., 3] / 2
    results = []
    for i in range(decoded.shape[0]):
        obj = decoded[i, :, 4].sigmoid()
        cls_prob = decoded[i, :, 5:].sigmoid()
        score, label = (cls_prob * obj[:, None]).max(1)
        keep = score > conf_thre
        boxes, score, label = box_corner[i][keep], score[keep], label[keep]
        k = batched_nms(boxes, score, label, nms_thre)
        results.append({"boxes": boxes[k], "scores": score[k],
                        "labels": label[k]})
    return results


@register_model
def yolox_s(num_classes=80, **kw):
    return YOLOX(num_classes, 0.33, 0.50)


@register_model
def yolox_m(num_classes=80, **kw):
    return YOLOX(num_classes, 0.67, 0.75)


@register_model
def yolox_l(num_classes=80, **kw):
    return YOLOX(num_classes, 1.0, 1.0)


@register_model
def yolox_x(num_classes=80, **kw):
    return YOLOX(num_classes, 1.33, 1.25)
