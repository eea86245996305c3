"""Extra backbones from the Happy-Whale retrieval model zoo: DPN (dual-path
networks) and Inception-v4.

Reference parity: metric_learning/Happy-Whale/retrieval/models/modelZoo/
(dpn 381 LoC, inceptionV4 335 LoC; nasnet/polyNet are documented as out of
scope in SURVEY-parity notes) — re-designed on fused HIP BN+ReLU.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import BatchNorm2d
from ..registry import register_model


class BnActConv(nn.Sequential):
    def __init__(self, cin, cout, k, stride=1, padding=0, groups=1):
        super().__init__(BatchNorm2d(cin, relu=True),
                         nn.Conv2d(cin, cout, k, stride, padding,
                                   groups=groups, bias=False))


class DualPathBlock(nn.Module):
    """1x1 -> 3x3(grouped) -> 1x1 with a residual part + a dense part."""

    def __init__(self, cin, num_1x1_a, num_3x3_b, num_1x1_c, inc, groups,
                 block_type="normal"):
        super().__init__()
        self.num_1x1_c = num_1x1_c
        self.has_proj = block_type in ("proj", "down")
        stride = 2 if block_type == "down" else 1
        if self.has_proj:
            self.c1x1_w = BnActConv(cin, num_1x1_c + 2 * inc, 1, stride)
        self.c1x1_a = BnActConv(cin, num_1x1_a, 1)
        self.c3x3_b = BnActConv(num_1x1_a, num_3x3_b, 3, stride, 1, groups)
        self.c1x1_c = BnActConv(num_3x3_b, num_1x1_c + inc, 1)

    def forward(self, x):
        if isinstance(x, tuple):
            res, dense = x
            inp = torch.cat([res, dense], dim=1)
        else:
            inp = x
        if self.has_proj:
            proj = self.c1x1_w(inp)
            res = proj[:, :self.num_1x1_c]
            dense = proj[:, self.num_1x1_c:]
        out = self.c1x1_c(self.c3x3_b(self.c1x1_a(inp)))
        res = res + out[:, :self.num_1x1_c]
        dense = torch.cat([dense, out[:, self.num_1x1_c:]], dim=1)
        return res, dense


class DPN(nn.Module):
    def __init__(self, num_init=64, k_r=96, groups=32,
                 k_sec=(3, 4, 20, 3), inc_sec=(16, 32, 24, 128),
                 num_classes=1000):
        super().__init__()
        self.stem = nn.Sequential(
            nn.Conv2d(3, num_init, 7, 2, 3, bias=False),
            BatchNorm2d(num_init, relu=True),
            nn.MaxPool2d(3, 2, 1))
        blocks = []
        cin = num_init
        bw = 256
        for i, (k, inc) in enumerate(zip(k_sec, inc_sec)):
            r = (k_r * bw) // 256
            btype = "proj" if i == 0 else "down"
            blocks.append(DualPathBlock(cin, r, r, bw, inc, groups, btype))
            cin = bw + 3 * inc
            for _ in range(1, k):
                blocks.append(DualPathBlock(cin, r, r, bw, inc, groups))
                cin += inc
            bw *= 2
        self.features = nn.Sequential(*blocks)
        self.final_bn = BatchNorm2d(cin, relu=True)
        self.classifier = nn.Linear(cin, num_classes)

    def forward(self, x):
        x = self.stem(x)
        for block in self.features:
            x = block(x)
        x = torch.cat(x, dim=1)
        x = self.final_bn(x)
        x = x.mean((2, 3))
        return self.classifier(x)


# ------------------------------------------------------------ inception-v4 --
class Conv2dBn(nn.Sequential):
    def __init__(self, cin, cout, k, stride=1, padding=0):
        super().__init__(nn.Conv2d(cin, cout, k, stride, padding, bias=False),
                         BatchNorm2d(cout, relu=True, eps=1e-3))


class InceptionA(nn.Module):
    def __init__(self, cin=384):
        super().__init__()
        self.b0 = Conv2dBn(cin, 96, 1)
        self.b1 = nn.Sequential(Conv2dBn(cin, 64, 1), Conv2dBn(64, 96, 3, 1, 1))
        self.b2 = nn.Sequential(Conv2dBn(cin, 64, 1), Conv2dBn(64, 96, 3, 1, 1),
                                Conv2dBn(96, 96, 3, 1, 1))
        self.b3 = nn.Sequential(nn.AvgPool2d(3, 1, 1), Conv2dBn(cin, 96, 1))

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.b2(x), self.b3(x)], 1)


class ReductionA(nn.Module):
    def __init__(self, cin=384):
        super().__init__()
        self.b0 = Conv2dBn(cin, 384, 3, 2)
        self.b1 = nn.Sequential(Conv2dBn(cin, 192, 1),
                                Conv2dBn(192, 224, 3, 1, 1),
                                Conv2dBn(224, 256, 3, 2))
        self.b2 = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.b2(x)], 1)


class InceptionB(nn.Module):
    def __init__(self, cin=1024):
        super().__init__()
        self.b0 = Conv2dBn(cin, 384, 1)
        self.b1 = nn.Sequential(Conv2dBn(cin, 192, 1),
                                Conv2dBn(192, 224, (1, 7), 1, (0, 3)),
                                Conv2dBn(224, 256, (7, 1), 1, (3, 0)))
        self.b2 = nn.Sequential(Conv2dBn(cin, 192, 1),
                                Conv2dBn(192, 192, (7, 1), 1, (3, 0)),
                                Conv2dBn(192, 224, (1, 7), 1, (0, 3)),
                                Conv2dBn(224, 224, (7, 1), 1, (3, 0)),
                                Conv2dBn(224, 256, (1, 7), 1, (0, 3)))
        self.b3 = nn.Sequential(nn.AvgPool2d(3, 1, 1), Conv2dBn(cin, 128, 1))

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.b2(x), self.b3(x)], 1)


class ReductionB(nn.Module):
    def __init__(self, cin=1024):
        super().__init__()
        self.b0 = nn.Sequential(Conv2dBn(cin, 192, 1), Conv2dBn(192, 192, 3, 2))
        self.b1 = nn.Sequential(Conv2dBn(cin, 256, 1),
                                Conv2dBn(256, 256, (1, 7), 1, (0, 3)),
                                Conv2dBn(256, 320, (7, 1), 1, (3, 0)),
                                Conv2dBn(320, 320, 3, 2))
        self.b2 = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.b2(x)], 1)


class InceptionC(nn.Module):
    def __init__(self, cin=1536):
        super().__init__()
        self.b0 = Conv2dBn(cin, 256, 1)
        self.b1_stem = Conv2dBn(cin, 384, 1)
        self.b1a = Conv2dBn(384, 256, (1, 3), 1, (0, 1))
        self.b1b = Conv2dBn(384, 256, (3, 1), 1, (1, 0))
        self.b2_stem = nn.Sequential(Conv2dBn(cin, 384, 1),
                                     Conv2dBn(384, 448, (3, 1), 1, (1, 0)),
                                     Conv2dBn(448, 512, (1, 3), 1, (0, 1)))
        self.b2a = Conv2dBn(512, 256, (1, 3), 1, (0, 1))
        self.b2b = Conv2dBn(512, 256, (3, 1), 1, (1, 0))
        self.b3 = nn.Sequential(nn.AvgPool2d(3, 1, 1), Conv2dBn(cin, 256, 1))

    def forward(self, x):
        b1 = self.b1_stem(x)
        b2 = self.b2_stem(x)
        return torch.cat([self.b0(x), self.b1a(b1), self.b1b(b1),
                          self.b2a(b2), self.b2b(b2), self.b3(x)], 1)


class InceptionV4(nn.Module):
    def __init__(self, num_classes=1000, dropout=0.2):
        super().__init__()
        self.stem = nn.Sequential(
            Conv2dBn(3, 32, 3, 2), Conv2dBn(32, 32, 3),
            Conv2dBn(32, 64, 3, 1, 1), nn.MaxPool2d(3, 2),
            Conv2dBn(64, 80, 1), Conv2dBn(80, 192, 3), nn.MaxPool2d(3, 2),
            Conv2dBn(192, 384, 1))
        self.features = nn.Sequential(
            *[InceptionA() for _ in range(4)], ReductionA(),
            *[InceptionB() for _ in range(7)], ReductionB(),
            *[InceptionC() for _ in range(3)])
        self.dropout = nn.Dropout(dropout)
        self.classifier = nn.Linear(1536, num_classes)

    def forward(self, x):
        x = self.features(self.stem(x))
        x = x.mean((2, 3))
        return self.classifier(self.dropout(x))


@register_model
def dpn92(num_classes=1000, **kw):
    return DPN(num_init=64, k_r=96, groups=32, k_sec=(3, 4, 20, 3),
               inc_sec=(16, 32, 24, 128), num_classes=num_classes)


@register_model
def dpn68(num_classes=1000, **kw):
    return DPN(num_init=10, k_r=128, groups=32, k_sec=(3, 4, 12, 3),
               inc_sec=(16, 32, 32, 64), num_classes=num_classes)


@register_model
def inception_v4(num_classes=1000, **kw):
    return InceptionV4(num_classes=num_classes, **kw)
