"""EfficientNet B0-B7 (MBConv + SE, SiLU activation).

Reference parity: classification/efficientNet/models/network.py
(MBConvConfig:149, SELayer:126) — re-designed: conv->BN->SiLU chains use the
framework's HIP BN kernel + fused HIP SiLU.
"""
from __future__ import annotations

import math
from functools import partial

import torch
import torch.nn as nn

from ...ops import BatchNorm2d, DropPath, SiLU
from ..registry import register_model


def _round_channels(c, width_mult, divisor=8):
    c *= width_mult
    new_c = max(divisor, int(c + divisor / 2) // divisor * divisor)
    if new_c < 0.9 * c:
        new_c += divisor
    return int(new_c)


class ConvBNAct(nn.Sequential):
    def __init__(self, cin, cout, k=3, stride=1, groups=1, act=True):
        layers = [nn.Conv2d(cin, cout, k, stride, (k - 1) // 2, groups=groups,
                            bias=False),
                  BatchNorm2d(cout, eps=1e-3, momentum=0.1)]
        if act:
            layers.append(SiLU())
        super().__init__(*layers)


class SqueezeExcite(nn.Module):
    def __init__(self, cin, expand_c, ratio=0.25):
        super().__init__()
        squeeze_c = int(cin * ratio)
        self.fc1 = nn.Conv2d(expand_c, squeeze_c, 1)
        self.act1 = SiLU()
        self.fc2 = nn.Conv2d(squeeze_c, expand_c, 1)
        self.act2 = nn.Sigmoid()

    def forward(self, x):
        s = x.mean((2, 3), keepdim=True)
        s = self.act2(self.fc2(self.act1(self.fc1(s))))
        return x * s


class MBConv(nn.Module):
    def __init__(self, cin, cout, k, stride, expand_ratio, drop_path=0.0):
        super().__init__()
        self.use_res = stride == 1 and cin == cout
        mid = cin * expand_ratio
        layers = []
        if expand_ratio != 1:
            layers.append(ConvBNAct(cin, mid, k=1))
        layers += [ConvBNAct(mid, mid, k=k, stride=stride, groups=mid),
                   SqueezeExcite(cin, mid),
                   ConvBNAct(mid, cout, k=1, act=False)]
        self.block = nn.Sequential(*layers)
        self.drop_path = DropPath(drop_path) if drop_path > 0 else nn.Identity()

    def forward(self, x):
        out = self.block(x)
        if self.use_res:
            out = x + self.drop_path(out)
        return out


# (expand, k, stride, channels, repeats) per stage — EfficientNet-B0
_B0_STAGES = [
    (1, 3, 1, 16, 1), (6, 3, 2, 24, 2), (6, 5, 2, 40, 2), (6, 3, 2, 80, 3),
    (6, 5, 1, 112, 3), (6, 5, 2, 192, 4), (6, 3, 1, 320, 1),
]


class EfficientNet(nn.Module):
    def __init__(self, width_mult=1.0, depth_mult=1.0, dropout=0.2,
                 drop_path_rate=0.2, num_classes=1000):
        super().__init__()
        round_c = partial(_round_channels, width_mult=width_mult)
        stem_c = round_c(32)
        self.stem = ConvBNAct(3, stem_c, stride=2)

        total_blocks = sum(math.ceil(r * depth_mult) for *_, r in _B0_STAGES)
        blocks, cin, bidx = [], stem_c, 0
        for expand, k, stride, c, repeats in _B0_STAGES:
            cout = round_c(c)
            for i in range(math.ceil(repeats * depth_mult)):
                blocks.append(MBConv(cin, cout, k, stride if i == 0 else 1,
                                     expand,
                                     drop_path_rate * bidx / total_blocks))
                cin = cout
                bidx += 1
        self.blocks = nn.Sequential(*blocks)

        head_c = round_c(1280)
        self.head_conv = ConvBNAct(cin, head_c, k=1)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.classifier = nn.Sequential(
            nn.Dropout(dropout), nn.Linear(head_c, num_classes))

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out")
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.head_conv(self.blocks(self.stem(x)))
        return self.classifier(self.avgpool(x).flatten(1))


_PARAMS = {  # width, depth, resolution (unused at build), dropout
    "b0": (1.0, 1.0, 224, 0.2), "b1": (1.0, 1.1, 240, 0.2),
    "b2": (1.1, 1.2, 260, 0.3), "b3": (1.2, 1.4, 300, 0.3),
    "b4": (1.4, 1.8, 380, 0.4), "b5": (1.6, 2.2, 456, 0.4),
    "b6": (1.8, 2.6, 528, 0.5), "b7": (2.0, 3.1, 600, 0.5),
}


def _factory(v):
    w, d, _, p = _PARAMS[v]

    def f(num_classes=1000, **kw):
        return EfficientNet(w, d, dropout=p, num_classes=num_classes, **kw)
    f.__name__ = f"efficientnet_{v}"
    return f


for _v in _PARAMS:
    register_model(_factory(_v))
