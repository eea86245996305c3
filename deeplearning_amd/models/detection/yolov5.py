"""YOLOv5: yaml-defined CSP network, Detect head, ComputeLoss with
anchor-ratio target building.

Reference parity: detection/yolov5 (models/yolo.py Model/parse_model:121,297,
Detect:39; models/common.py Conv/C3/SPPF:36-200; utils/loss.py
ComputeLoss.build_targets:91-182) — re-designed: Conv is conv + framework HIP
BN + fused HIP SiLU; model graph still yaml-driven for config parity, with
the v5 s/m/l/x gd/gw scaling.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ...ops import BatchNorm2d, SiLU, bbox_iou_aligned
from ..registry import register_model


def autopad(k, p=None):
    return k // 2 if p is None else p


class Conv(nn.Module):
    def __init__(self, c1, c2, k=1, s=1, p=None, g=1, act=True):
        super().__init__()
        self.conv = nn.Conv2d(c1, c2, k, s, autopad(k, p), groups=g, bias=False)
        self.bn = BatchNorm2d(c2)
        self.act = SiLU() if act else nn.Identity()

    def forward(self, x):
        return self.act(self.bn(self.conv(x)))


class Bottleneck(nn.Module):
    def __init__(self, c1, c2, shortcut=True, g=1, e=0.5):
        super().__init__()
        c_ = int(c2 * e)
        self.cv1 = Conv(c1, c_, 1)
        self.cv2 = Conv(c_, c2, 3, g=g)
        self.add = shortcut and c1 == c2

    def forward(self, x):
        y = self.cv2(self.cv1(x))
        return x + y if self.add else y


class C3(nn.Module):
    def __init__(self, c1, c2, n=1, shortcut=True, g=1, e=0.5):
        super().__init__()
        c_ = int(c2 * e)
        self.cv1 = Conv(c1, c_, 1)
        self.cv2 = Conv(c1, c_, 1)
        self.cv3 = Conv(2 * c_, c2, 1)
        self.m = nn.Sequential(*[Bottleneck(c_, c_, shortcut, g, 1.0)
                                 for _ in range(n)])

    def forward(self, x):
        return self.cv3(torch.cat([self.m(self.cv1(x)), self.cv2(x)], dim=1))


class SPPF(nn.Module):
    def __init__(self, c1, c2, k=5):
        super().__init__()
        c_ = c1 // 2
        self.cv1 = Conv(c1, c_, 1)
        self.cv2 = Conv(c_ * 4, c2, 1)
        self.m = nn.MaxPool2d(k, 1, k // 2)

    def forward(self, x):
        x = self.cv1(x)
        y1 = self.m(x)
        y2 = self.m(y1)
        return self.cv2(torch.cat([x, y1, y2, self.m(y2)], 1))


class Concat(nn.Module):
    def __init__(self, dim=1):
        super().__init__()
        self.d = dim

    def forward(self, xs):
        return torch.cat(xs, self.d)


class Detect(nn.Module):
    """Per-level 1x1 conv -> [B, A, H, W, 5+nc] (ref models/yolo.py:39)."""

    stride = None

    def __init__(self, nc=80, anchors=(), ch=()):
        super().__init__()
        self.nc = nc
        self.no = nc + 5
        self.nl = len(anchors)
        self.na = len(anchors[0]) // 2
        a = torch.tensor(anchors).float().view(self.nl, -1, 2)
        self.register_buffer("anchors", a)  # in input pixels
        self.m = nn.ModuleList(nn.Conv2d(c, self.no * self.na, 1) for c in ch)
        self.grid = [torch.zeros(1)] * self.nl

    def forward(self, xs):
        out = []
        for i, x in enumerate(xs):
            x = self.m[i](x)
            B, _, H, W = x.shape
            x = x.view(B, self.na, self.no, H, W).permute(0, 1, 3, 4, 2)
            out.append(x.contiguous())
        if self.training:
            return out
        # decode
        decoded = []
        for i, x in enumerate(out):
            B, na, H, W, _ = x.shape
            if self.grid[i].shape[-3:-1] != (H, W) or \
                    self.grid[i].device != x.device:
                gy, gx = torch.meshgrid(
                    torch.arange(H, device=x.device),
                    torch.arange(W, device=x.device), indexing="ij")
                self.grid[i] = torch.stack([gx, gy], -1).view(
                    1, 1, H, W, 2).float()
            y = x.sigmoid()
            stride = self.stride[i]
            xy = (y[..., 0:2] * 2 - 0.5 + self.grid[i]) * stride
            wh = (y[..., 2:4] * 2) ** 2 * \
                self.anchors[i].view(1, na, 1, 1, 2)
            decoded.append(torch.cat(
                [xy, wh, y[..., 4:]], -1).view(B, -1, self.no))
        return torch.cat(decoded, 1), out


# yolov5 network graph (models/yolov5s.yaml structure; gd/gw scale variants)
YOLOV5_CFG = {
    "anchors": [[10, 13, 16, 30, 33, 23],
                [30, 61, 62, 45, 59, 119],
                [116, 90, 156, 198, 373, 326]],
    "backbone": [
        # [from, number, module, args]
        [-1, 1, "Conv", [64, 6, 2, 2]],     # 0 P1/2
        [-1, 1, "Conv", [128, 3, 2]],       # 1 P2/4
        [-1, 3, "C3", [128]],
        [-1, 1, "Conv", [256, 3, 2]],       # 3 P3/8
        [-1, 6, "C3", [256]],
        [-1, 1, "Conv", [512, 3, 2]],       # 5 P4/16
        [-1, 9, "C3", [512]],
        [-1, 1, "Conv", [1024, 3, 2]],      # 7 P5/32
        [-1, 3, "C3", [1024]],
        [-1, 1, "SPPF", [1024, 5]],         # 9
    ],
    "head": [
        [-1, 1, "Conv", [512, 1, 1]],
        [-1, 1, "Upsample", [None, 2, "nearest"]],
        [[-1, 6], 1, "Concat", [1]],
        [-1, 3, "C3", [512, False]],        # 13

        [-1, 1, "Conv", [256, 1, 1]],
        [-1, 1, "Upsample", [None, 2, "nearest"]],
        [[-1, 4], 1, "Concat", [1]],
        [-1, 3, "C3", [256, False]],        # 17 (P3/8)

        [-1, 1, "Conv", [256, 3, 2]],
        [[-1, 14], 1, "Concat", [1]],
        [-1, 3, "C3", [512, False]],        # 20 (P4/16)

        [-1, 1, "Conv", [512, 3, 2]],
        [[-1, 10], 1, "Concat", [1]],
        [-1, 3, "C3", [1024, False]],       # 23 (P5/32)

        [[17, 20, 23], 1, "Detect", ["nc", "anchors"]],
    ],
}

_MODULES = {"Conv": Conv, "C3": C3, "SPPF": SPPF, "Concat": Concat,
            "Detect": Detect, "Upsample": nn.Upsample}


def parse_model(cfg: dict, nc: int, gd: float, gw: float, ch=3):
    """Build the layer list from the yaml-style dict (ref yolo.py:297)."""
    anchors = cfg["anchors"]
    layers, save, c2 = [], [], ch
    out_ch = []  # out_ch[j] = output channels of layer j

    def ch_of(j):
        return (out_ch[-1] if out_ch else ch) if j == -1 else out_ch[j]

    for i, (f, n, m, args) in enumerate(cfg["backbone"] + cfg["head"]):
        mod = _MODULES[m]
        n = max(round(n * gd), 1) if n > 1 else n
        args = list(args)
        if m in ("Conv", "C3", "SPPF"):
            c1 = ch_of(f if isinstance(f, int) else f[0])
            c2 = max(round(args[0] * gw / 8) * 8, 8)
            args = [c1, c2, *args[1:]]
            if m == "C3":
                args.insert(2, n)
                n = 1
        elif m == "Concat":
            c2 = sum(ch_of(x) for x in f)
        elif m == "Detect":
            args = [nc, anchors, [ch_of(x) for x in f]]
        elif m == "Upsample":
            c2 = ch_of(f)
        block = mod(*args) if n == 1 else nn.Sequential(
            *[mod(*args) for _ in range(n)])
        block.f, block.i = f, i
        layers.append(block)
        out_ch.append(c2)
        if isinstance(f, list):
            save.extend(x % i for x in f if x != -1)
        elif f != -1:
            save.append(f % i)
    return nn.ModuleList(layers), sorted(set(save))


class YoloV5(nn.Module):
    def __init__(self, nc=80, gd=0.33, gw=0.50, cfg=None):
        super().__init__()
        self.nc = nc
        self.model, self.save = parse_model(cfg or YOLOV5_CFG, nc, gd, gw)
        detect = self.model[-1]
        # infer strides with a dry run
        s = 256
        detect.stride = torch.tensor(
            [s / x.shape[-2] for x in self._forward_once(
                torch.zeros(1, 3, s, s))])
        self.stride = detect.stride
        self._initialize_biases()

    def _forward_once(self, x):
        y = []
        for m in self.model:
            if m.f != -1:
                x = y[m.f] if isinstance(m.f, int) else \
                    [x if j == -1 else y[j] for j in m.f]
            x = m(x)
            y.append(x if m.i in self.save or isinstance(m, Detect) else None)
        return x

    def _initialize_biases(self):
        detect = self.model[-1]
        for mi, s in zip(detect.m, detect.stride):
            b = mi.bias.view(detect.na, -1)
            with torch.no_grad():
                b[:, 4] += math.log(8 / (640 / s) ** 2)
                b[:, 5:] += math.log(0.6 / (detect.nc - 0.999999))
                mi.bias.copy_(b.view(-1))

    def forward(self, x):
        return self._forward_once(x)


class ComputeLoss:
    """YOLOv5 loss: BCE cls/obj + CIoU box, anchor-ratio matching with
    neighbor-cell expansion (ref utils/loss.py:91-182)."""

    def __init__(self, model, box_gain=0.05, cls_gain=0.5, obj_gain=1.0,
                 anchor_t=4.0):
        detect = model.model[-1]
        self.na, self.nc, self.nl = detect.na, detect.nc, detect.nl
        self.anchors = detect.anchors  # input-pixel units
        self.stride = detect.stride
        self.gains = (box_gain, cls_gain, obj_gain)
        self.anchor_t = anchor_t
        self.balance = [4.0, 1.0, 0.4]
        self.bce = nn.BCEWithLogitsLoss()

    def build_targets(self, preds, targets):
        """targets: [N, 6] = (img_idx, cls, x, y, w, h) normalized 0-1."""
        na, nt = self.na, targets.shape[0]
        tcls, tbox, indices, anch = [], [], [], []
        gain = torch.ones(7, device=targets.device)
        ai = torch.arange(na, device=targets.device).float().view(
            na, 1).repeat(1, nt)
        targets = torch.cat(
            (targets.repeat(na, 1, 1), ai[..., None]), 2)  # na,nt,7
        g = 0.5
        off = torch.tensor([[0, 0], [1, 0], [0, 1], [-1, 0], [0, -1]],
                           device=targets.device).float() * g

        for i in range(self.nl):
            anchors_grid = self.anchors[i] / self.stride[i]
            shape = preds[i].shape  # B,na,H,W,no
            gain[2:6] = torch.tensor(shape)[[3, 2, 3, 2]]
            t = targets * gain
            if nt:
                r = t[..., 4:6] / anchors_grid[:, None]
                keep = torch.max(r, 1 / r).max(2).values < self.anchor_t
                t = t[keep]
                gxy = t[:, 2:4]
                gxi = gain[[2, 3]] - gxy
                j, k = ((gxy % 1 < g) & (gxy > 1)).T
                l, m = ((gxi % 1 < g) & (gxi > 1)).T
                j = torch.stack((torch.ones_like(j), j, k, l, m))
                t = t.repeat((5, 1, 1))[j]
                offsets = (torch.zeros_like(gxy)[None] + off[:, None])[j]
            else:
                t = targets[0]
                offsets = 0
            bc = t[:, 0].long()
            cls = t[:, 1].long()
            gxy = t[:, 2:4]
            gwh = t[:, 4:6]
            gij = (gxy - offsets).long()
            gi, gj = gij.T
            a = t[:, 6].long()
            gj = gj.clamp_(0, shape[2] - 1)
            gi = gi.clamp_(0, shape[3] - 1)
            indices.append((bc, a, gj, gi))
            tbox.append(torch.cat((gxy - gij, gwh), 1))
            anch.append(anchors_grid[a])
            tcls.append(cls)
        return tcls, tbox, indices, anch

    def __call__(self, preds, targets):
        device = targets.device
        lcls = torch.zeros(1, device=device)
        lbox = torch.zeros(1, device=device)
        lobj = torch.zeros(1, device=device)
        tcls, tbox, indices, anchors = self.build_targets(preds, targets)
        for i, pi in enumerate(preds):
            b, a, gj, gi = indices[i]
            tobj = torch.zeros_like(pi[..., 0])
            if b.shape[0]:
                ps = pi[b, a, gj, gi]
                pxy = ps[:, :2].sigmoid() * 2 - 0.5
                pwh = (ps[:, 2:4].sigmoid() * 2) ** 2 * anchors[i]
                pbox = torch.cat((pxy, pwh), 1)
                iou = bbox_iou_aligned(pbox, tbox[i], xywh=True,
                                       CIoU=True).squeeze(-1)
                lbox += (1.0 - iou).mean()
                tobj[b, a, gj, gi] = iou.detach().clamp(0)
                if self.nc > 1:
                    t = torch.zeros_like(ps[:, 5:])
                    t[range(b.shape[0]), tcls[i]] = 1.0
                    lcls += self.bce(ps[:, 5:], t)
            lobj += self.bce(pi[..., 4], tobj) * self.balance[i]
        box_g, cls_g, obj_g = self.gains
        bs = preds[0].shape[0]
        loss = lbox * box_g + lobj * obj_g + lcls * cls_g
        return loss * bs, {"box": lbox.detach(), "obj": lobj.detach(),
                           "cls": lcls.detach()}


@register_model
def yolov5s(num_classes=80, **kw):
    return YoloV5(nc=num_classes, gd=0.33, gw=0.50)


@register_model
def yolov5m(num_classes=80, **kw):
    return YoloV5(nc=num_classes, gd=0.67, gw=0.75)


@register_model
def yolov5l(num_classes=80, **kw):
    return YoloV5(nc=num_classes, gd=1.0, gw=1.0)


@register_model
def yolov5x(num_classes=80, **kw):
    return YoloV5(nc=num_classes, gd=1.33, gw=1.25)


def yolov5_from_yaml(cfg_path, nc=80, gd=None, gw=None):
    """Build a YoloV5 from a yaml model file (ref models/yolo.py Model(cfg)).

    The yaml uses the reference schema: nc, depth_multiple, width_multiple,
    anchors, backbone, head.
    """
    import yaml

    with open(cfg_path) as f:
        d = yaml.safe_load(f)
    cfg = {"anchors": d["anchors"], "backbone": d["backbone"],
           "head": d["head"]}
    return YoloV5(nc=d.get("nc", nc) if nc == 80 else nc,
                  gd=gd if gd is not None else d.get("depth_multiple", 0.33),
                  gw=gw if gw is not None else d.get("width_multiple", 0.5),
                  cfg=cfg)
