"""SE-ResNet and SK-Net (selective kernel) on the shared ResNet skeleton.

Reference parity: classification/seNet/models/se_resnet.py and
classification/skNet/models/sknet.py — re-designed: both are channel-attention
bottlenecks dropped into this repo's ResNet trunk (fused HIP BN+ReLU and
fused add+relu residual joins).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import BatchNorm2d, add_relu
from ..registry import register_model
from .resnet import ResNet, conv1x1, conv3x3, make_norm


class SELayer(nn.Module):
    def __init__(self, channel, reduction=16):
        super().__init__()
        self.avg_pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Sequential(
            nn.Linear(channel, channel // reduction, bias=False),
            nn.ReLU(inplace=True),
            nn.Linear(channel // reduction, channel, bias=False),
            nn.Sigmoid())

    def forward(self, x):
        b, c = x.shape[:2]
        y = self.avg_pool(x).view(b, c)
        y = self.fc(y).view(b, c, 1, 1)
        return x * y


class SEBottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, groups=1,
                 base_width=64, dilation=1, norm_layer=None, reduction=16):
        super().__init__()
        norm_layer = norm_layer or BatchNorm2d
        width = int(planes * (base_width / 64.0)) * groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = make_norm(norm_layer, width, relu=True)
        self.conv2 = conv3x3(width, width, stride, groups, dilation)
        self.bn2 = make_norm(norm_layer, width, relu=True)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = norm_layer(planes * self.expansion)
        self.se = SELayer(planes * self.expansion, reduction)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        out = self.se(self.bn3(self.conv3(out)))
        if self.downsample is not None:
            identity = self.downsample(x)
        return add_relu(out, identity)


class SKConv(nn.Module):
    """Selective-kernel conv: M parallel branches, softmax channel selection."""

    def __init__(self, channels, branches=2, groups=32, reduce=16, stride=1,
                 min_width=32):
        super().__init__()
        d = max(channels // reduce, min_width)
        self.branches = nn.ModuleList([
            nn.Sequential(
                nn.Conv2d(channels, channels, 3, stride=stride, padding=1 + i,
                          dilation=1 + i, groups=groups, bias=False),
                BatchNorm2d(channels, relu=True))
            for i in range(branches)])
        self.gap = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Sequential(
            nn.Conv2d(channels, d, 1, bias=False),
            BatchNorm2d(d, relu=True))
        self.fcs = nn.ModuleList([nn.Conv2d(d, channels, 1) for _ in range(branches)])
        self.softmax = nn.Softmax(dim=1)

    def forward(self, x):
        feats = torch.stack([b(x) for b in self.branches], dim=1)  # B,M,C,H,W
        attn = self.fc(self.gap(feats.sum(1)))
        attn = torch.stack([f(attn) for f in self.fcs], dim=1)  # B,M,C,1,1
        attn = self.softmax(attn)
        return (feats * attn).sum(1)


class SKBottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, groups=1,
                 base_width=64, dilation=1, norm_layer=None):
        super().__init__()
        norm_layer = norm_layer or BatchNorm2d
        width = planes * 2  # SK-Net bottleneck runs at 2x planes with 32 groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = make_norm(norm_layer, width, relu=True)
        self.sk = SKConv(width, stride=stride)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = norm_layer(planes * self.expansion)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.sk(out)
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return add_relu(out, identity)


@register_model
def se_resnet50(num_classes=1000, **kw):
    return ResNet(SEBottleneck, [3, 4, 6, 3], num_classes=num_classes, **kw)


@register_model
def se_resnet101(num_classes=1000, **kw):
    return ResNet(SEBottleneck, [3, 4, 23, 3], num_classes=num_classes, **kw)


@register_model
def sk_resnet50(num_classes=1000, **kw):
    return ResNet(SKBottleneck, [3, 4, 6, 3], num_classes=num_classes, **kw)


@register_model
def sk_resnet101(num_classes=1000, **kw):
    return ResNet(SKBottleneck, [3, 4, 23, 3], num_classes=num_classes, **kw)
