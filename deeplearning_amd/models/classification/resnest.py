"""ResNeSt (split-attention networks).

Reference parity: classification/resnest/models/resnest.py — re-designed on the
shared ResNet trunk with fused HIP BN+ReLU; SplitAttn implements radix-major
split attention (rSoftMax over radix groups).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import BatchNorm2d, add_relu
from ..registry import register_model
from .resnet import conv1x1, make_norm


class RSoftMax(nn.Module):
    def __init__(self, radix, cardinality):
        super().__init__()
        self.radix = radix
        self.cardinality = cardinality

    def forward(self, x):
        b = x.size(0)
        if self.radix > 1:
            x = x.view(b, self.cardinality, self.radix, -1).transpose(1, 2)
            x = F.softmax(x, dim=1)
            return x.reshape(b, -1)
        return torch.sigmoid(x)


class SplitAttnConv(nn.Module):
    def __init__(self, cin, channels, kernel_size=3, stride=1, padding=1,
                 groups=1, radix=2, reduction_factor=4):
        super().__init__()
        inter = max(32, cin * radix // reduction_factor)
        self.radix = radix
        self.cardinality = groups
        self.channels = channels
        self.conv = nn.Conv2d(cin, channels * radix, kernel_size, stride,
                              padding, groups=groups * radix, bias=False)
        self.bn0 = BatchNorm2d(channels * radix, relu=True)
        self.fc1 = nn.Conv2d(channels, inter, 1, groups=groups)
        self.bn1 = BatchNorm2d(inter, relu=True)
        self.fc2 = nn.Conv2d(inter, channels * radix, 1, groups=groups)
        self.rsoftmax = RSoftMax(radix, groups)

    def forward(self, x):
        x = self.bn0(self.conv(x))
        B, RC, H, W = x.shape
        if self.radix > 1:
            splits = x.view(B, self.radix, self.channels, H, W)
            gap = splits.sum(1)
        else:
            gap = x
        gap = gap.mean((2, 3), keepdim=True)
        gap = self.bn1(self.fc1(gap))
        attn = self.rsoftmax(self.fc2(gap)).view(B, -1, 1, 1)
        if self.radix > 1:
            attn = attn.view(B, self.radix, self.channels, 1, 1)
            return (splits * attn).sum(1)
        return x * attn


class ResNeStBottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, radix=2,
                 cardinality=1, bottleneck_width=64, avd=True, is_first=False,
                 norm_layer=None):
        super().__init__()
        norm_layer = norm_layer or BatchNorm2d
        width = int(planes * (bottleneck_width / 64.0)) * cardinality
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = make_norm(norm_layer, width, relu=True)
        self.avd = avd and (stride > 1 or is_first)
        if self.avd:
            self.avd_layer = nn.AvgPool2d(3, stride, padding=1)
            stride = 1
        self.conv2 = SplitAttnConv(width, width, 3, stride, 1,
                                   groups=cardinality, radix=radix)
        self.conv3 = conv1x1(width, planes * 4)
        self.bn3 = norm_layer(planes * 4)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.conv2(out)
        if self.avd:
            out = self.avd_layer(out)
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return add_relu(out, identity)


class ResNeSt(nn.Module):
    def __init__(self, layers, radix=2, cardinality=1, bottleneck_width=64,
                 num_classes=1000, stem_width=32, norm_layer=None):
        super().__init__()
        self._norm = norm_layer or BatchNorm2d
        self.radix, self.cardinality = radix, cardinality
        self.bottleneck_width = bottleneck_width
        self.inplanes = stem_width * 2
        # deep stem: three 3x3 convs
        self.stem = nn.Sequential(
            nn.Conv2d(3, stem_width, 3, 2, 1, bias=False),
            make_norm(self._norm, stem_width, relu=True),
            nn.Conv2d(stem_width, stem_width, 3, 1, 1, bias=False),
            make_norm(self._norm, stem_width, relu=True),
            nn.Conv2d(stem_width, stem_width * 2, 3, 1, 1, bias=False),
            make_norm(self._norm, stem_width * 2, relu=True))
        self.maxpool = nn.MaxPool2d(3, 2, 1)
        self.layer1 = self._make_layer(64, layers[0], 1, is_first=False)
        self.layer2 = self._make_layer(128, layers[1], 2)
        self.layer3 = self._make_layer(256, layers[2], 2)
        self.layer4 = self._make_layer(512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * 4, num_classes)

    def _make_layer(self, planes, blocks, stride, is_first=True):
        downsample = None
        if stride != 1 or self.inplanes != planes * 4:
            # ResNeSt-D style: avgpool + 1x1 conv downsample
            down = []
            if stride != 1:
                down.append(nn.AvgPool2d(stride, stride, ceil_mode=True))
            down += [conv1x1(self.inplanes, planes * 4), self._norm(planes * 4)]
            downsample = nn.Sequential(*down)
        layers = [ResNeStBottleneck(
            self.inplanes, planes, stride, downsample, self.radix,
            self.cardinality, self.bottleneck_width, is_first=is_first,
            norm_layer=self._norm)]
        self.inplanes = planes * 4
        for _ in range(1, blocks):
            layers.append(ResNeStBottleneck(
                self.inplanes, planes, radix=self.radix,
                cardinality=self.cardinality,
                bottleneck_width=self.bottleneck_width, norm_layer=self._norm))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.stem(x))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.fc(self.avgpool(x).flatten(1))


@register_model
def resnest50(num_classes=1000, **kw):
    return ResNeSt([3, 4, 6, 3], num_classes=num_classes, **kw)


@register_model
def resnest101(num_classes=1000, **kw):
    return ResNeSt([3, 4, 23, 3], stem_width=64, num_classes=num_classes, **kw)
