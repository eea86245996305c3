"""HRNet-W18/W48: parallel multi-resolution trunk with fusion, plus a
segmentation head. The same trunk backs the pose-estimation keypoint model
(models/pose/hrnet_pose.py).

Reference parity: Image_segmentation/HR-Net-Seg/models/seg_hrnet.py (482 LoC)
and pose_estimation/Insulator/models/hrnet.py:5-299 — re-designed once as a
shared trunk on fused HIP BN+ReLU (the reference carries two near-copies).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import BatchNorm2d
from ..classification.resnet import BasicBlock, Bottleneck
from ..registry import register_model


class HRModule(nn.Module):
    """N parallel branches of BasicBlocks + full pairwise fusion."""

    def __init__(self, num_branches, num_blocks, channels):
        super().__init__()
        self.num_branches = num_branches
        self.branches = nn.ModuleList([
            nn.Sequential(*[BasicBlock(channels[i], channels[i])
                            for _ in range(num_blocks)])
            for i in range(num_branches)])
        fuse = []
        for i in range(num_branches):
            row = []
            for j in range(num_branches):
                if j > i:  # upsample j -> i
                    row.append(nn.Sequential(
                        nn.Conv2d(channels[j], channels[i], 1, bias=False),
                        BatchNorm2d(channels[i])))
                elif j == i:
                    row.append(nn.Identity())
                else:  # downsample j -> i with strided 3x3 chain
                    convs = []
                    for k in range(i - j):
                        cout = channels[i] if k == i - j - 1 else channels[j]
                        convs += [nn.Conv2d(channels[j], cout, 3, 2, 1,
                                            bias=False),
                                  BatchNorm2d(cout, relu=(k != i - j - 1))]
                    row.append(nn.Sequential(*convs))
            fuse.append(nn.ModuleList(row))
        self.fuse_layers = nn.ModuleList(fuse)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, xs):
        xs = [b(x) for b, x in zip(self.branches, xs)]
        outs = []
        for i in range(self.num_branches):
            y = None
            for j in range(self.num_branches):
                z = self.fuse_layers[i][j](xs[j])
                if j > i:
                    z = F.interpolate(z, size=xs[i].shape[-2:], mode="bilinear",
                                      align_corners=False)
                y = z if y is None else y + z
            outs.append(self.relu(y))
        return outs


class HRNetTrunk(nn.Module):
    def __init__(self, width=18):
        super().__init__()
        c = width
        self.channels = [c, c * 2, c * 4, c * 8]
        self.stem = nn.Sequential(
            nn.Conv2d(3, 64, 3, 2, 1, bias=False), BatchNorm2d(64, relu=True),
            nn.Conv2d(64, 64, 3, 2, 1, bias=False), BatchNorm2d(64, relu=True))
        # stage1: 4 bottlenecks at 1/4 res
        downsample = nn.Sequential(nn.Conv2d(64, 256, 1, bias=False),
                                   BatchNorm2d(256))
        self.layer1 = nn.Sequential(
            Bottleneck(64, 64, downsample=downsample),
            *[Bottleneck(256, 64) for _ in range(3)])
        # transitions + stages 2-4
        self.transition1 = nn.ModuleList([
            nn.Sequential(nn.Conv2d(256, c, 3, 1, 1, bias=False),
                          BatchNorm2d(c, relu=True)),
            nn.Sequential(nn.Conv2d(256, c * 2, 3, 2, 1, bias=False),
                          BatchNorm2d(c * 2, relu=True))])
        self.stage2 = nn.Sequential(HRModule(2, 4, self.channels[:2]))
        self.transition2 = nn.Sequential(
            nn.Conv2d(c * 2, c * 4, 3, 2, 1, bias=False),
            BatchNorm2d(c * 4, relu=True))
        self.stage3 = nn.Sequential(*[
            _Seq3(HRModule(3, 4, self.channels[:3])) for _ in range(4)])
        self.transition3 = nn.Sequential(
            nn.Conv2d(c * 4, c * 8, 3, 2, 1, bias=False),
            BatchNorm2d(c * 8, relu=True))
        self.stage4 = nn.Sequential(*[
            _Seq3(HRModule(4, 4, self.channels)) for _ in range(3)])

    def forward(self, x):
        x = self.layer1(self.stem(x))
        xs = [t(x) for t in self.transition1]
        xs = self.stage2[0](xs)
        xs = xs + [self.transition2(xs[-1])]
        for m in self.stage3:
            xs = m(xs)
        xs = xs + [self.transition3(xs[-1])]
        for m in self.stage4:
            xs = m(xs)
        return xs  # list of 4 feature maps, 1/4 .. 1/32


class _Seq3(nn.Module):
    """Wrap an HRModule so nn.Sequential-style stacking passes lists."""

    def __init__(self, mod):
        super().__init__()
        self.mod = mod

    def forward(self, xs):
        return self.mod(xs)


class HRNetSeg(nn.Module):
    def __init__(self, width=18, num_classes=19):
        super().__init__()
        self.trunk = HRNetTrunk(width)
        total = sum(self.trunk.channels)
        self.head = nn.Sequential(
            nn.Conv2d(total, total, 1, bias=False),
            BatchNorm2d(total, relu=True),
            nn.Conv2d(total, num_classes, 1))

    def forward(self, x):
        size = x.shape[-2:]
        xs = self.trunk(x)
        h, w = xs[0].shape[-2:]
        up = [xs[0]] + [F.interpolate(t, size=(h, w), mode="bilinear",
                                      align_corners=False) for t in xs[1:]]
        out = self.head(torch.cat(up, dim=1))
        return {"out": F.interpolate(out, size=size, mode="bilinear",
                                     align_corners=False)}


class OhemCrossEntropy(nn.Module):
    """Online hard example mining CE: keep the hardest pixels
    (ref Image_segmentation/HR-Net-Seg/loss/OhemCrossEntropy.py:6-49)."""

    def __init__(self, ignore_label=255, thres=0.7, min_kept=100000):
        super().__init__()
        self.ignore_label = ignore_label
        self.thresh = thres
        self.min_kept = max(1, min_kept)

    def forward(self, logits, target):
        pred = F.softmax(logits, dim=1)
        pixel_losses = F.cross_entropy(
            logits, target, ignore_index=self.ignore_label,
            reduction="none").contiguous().view(-1)
        mask = target.contiguous().view(-1) != self.ignore_label

        tmp_target = target.clone()
        tmp_target[tmp_target == self.ignore_label] = 0
        pred = pred.gather(1, tmp_target.unsqueeze(1)).squeeze(1)
        pred, ind = pred.contiguous().view(-1)[mask].contiguous().sort()
        if pred.numel() == 0:
            return pixel_losses.sum() * 0
        min_value = pred[min(self.min_kept, pred.numel() - 1)]
        threshold = max(min_value, self.thresh)
        pixel_losses = pixel_losses[mask][ind]
        keep = pred < threshold
        if not bool(keep.any()):
            # all predictions equally confident (e.g. near-uniform logits):
            # strict < selects nothing and mean() would be NaN — keep the
            # min_kept hardest instead
            k = min(self.min_kept, pixel_losses.numel())
            return pixel_losses[:k].mean()
        return pixel_losses[keep].mean()


@register_model
def hrnet_w18_seg(num_classes=19, **kw):
    return HRNetSeg(width=18, num_classes=num_classes)


@register_model
def hrnet_w48_seg(num_classes=19, **kw):
    return HRNetSeg(width=48, num_classes=num_classes)
