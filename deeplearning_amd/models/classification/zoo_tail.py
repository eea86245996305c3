"""Remaining Happy-Whale retrieval model-zoo backbones: Xception, SENet-154,
PolyNet and NASNet-A-Large.

Reference parity: metric_learning/Happy-Whale/retrieval/models/modelZoo/
{xception.py:1 (194 LoC), senet.py:1 (449), ployNet.py:1 (490),
nasnet.py:1 (643)} — independent implementations of the published
architectures (Chollet'17; Hu'18 SENet-154; Zhang'17 PolyNet; Zoph'18
NASNet-A), built on the framework's fused HIP BatchNorm2d(+ReLU), not
translations of the vendored zoo files. The mxnet/TF weight converters are
out of scope (no offline source weights in this image).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import BatchNorm2d
from ..registry import register_model


def conv_bn(cin, cout, k, stride=1, padding=0, relu=True, groups=1):
    return nn.Sequential(
        nn.Conv2d(cin, cout, k, stride, padding, groups=groups, bias=False),
        BatchNorm2d(cout, relu=relu))


# ------------------------------------------------------------- Xception ----
class SeparableConv2d(nn.Module):
    def __init__(self, cin, cout, k=3, stride=1, padding=1):
        super().__init__()
        self.depthwise = nn.Conv2d(cin, cin, k, stride, padding, groups=cin,
                                   bias=False)
        self.pointwise = nn.Conv2d(cin, cout, 1, bias=False)

    def forward(self, x):
        return self.pointwise(self.depthwise(x))


class XceptionBlock(nn.Module):
    def __init__(self, cin, cout, reps, stride=1, start_with_relu=True,
                 grow_first=True):
        super().__init__()
        self.skip = None
        if cout != cin or stride != 1:
            self.skip = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride, bias=False),
                BatchNorm2d(cout))
        layers = []
        c = cin
        for i in range(reps):
            if i > 0 or start_with_relu:
                layers.append(nn.ReLU(inplace=False))
            co = cout if (grow_first or i == reps - 1) else cin
            layers.append(SeparableConv2d(c, co))
            layers.append(BatchNorm2d(co))
            c = co
        if stride != 1:
            layers.append(nn.MaxPool2d(3, stride, 1))
        self.rep = nn.Sequential(*layers)

    def forward(self, x):
        out = self.rep(x)
        skip = self.skip(x) if self.skip is not None else x
        return out + skip


class Xception(nn.Module):
    """Chollet'17 Xception: entry/middle/exit flow of separable convs."""

    def __init__(self, num_classes=1000):
        super().__init__()
        self.conv1 = conv_bn(3, 32, 3, 2, 1)
        self.conv2 = conv_bn(32, 64, 3, 1, 1)
        self.block1 = XceptionBlock(64, 128, 2, 2, start_with_relu=False)
        self.block2 = XceptionBlock(128, 256, 2, 2)
        self.block3 = XceptionBlock(256, 728, 2, 2)
        self.middle = nn.Sequential(*[
            XceptionBlock(728, 728, 3) for _ in range(8)])
        self.block12 = XceptionBlock(728, 1024, 2, 2, grow_first=False)
        self.conv3 = nn.Sequential(SeparableConv2d(1024, 1536),
                                   BatchNorm2d(1536, relu=True))
        self.conv4 = nn.Sequential(SeparableConv2d(1536, 2048),
                                   BatchNorm2d(2048, relu=True))
        self.num_features = 2048
        self.fc = nn.Linear(2048, num_classes)

    def forward_features(self, x):
        x = self.conv2(self.conv1(x))
        x = self.block3(self.block2(self.block1(x)))
        x = self.middle(x)
        x = self.block12(x)
        return self.conv4(self.conv3(x))

    def forward(self, x):
        x = self.forward_features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


# ------------------------------------------------------------- SENet-154 ---
class SEModule(nn.Module):
    def __init__(self, channels, reduction=16):
        super().__init__()
        self.fc1 = nn.Conv2d(channels, channels // reduction, 1)
        self.fc2 = nn.Conv2d(channels // reduction, channels, 1)

    def forward(self, x):
        s = F.adaptive_avg_pool2d(x, 1)
        s = torch.sigmoid(self.fc2(F.relu(self.fc1(s), inplace=True)))
        return x * s


class SEBottleneck154(nn.Module):
    """SENet-154 bottleneck: grouped 3x3 with doubled 1x1-in width."""
    expansion = 4

    def __init__(self, cin, planes, stride=1, downsample=None, groups=64,
                 reduction=16):
        super().__init__()
        self.conv1 = conv_bn(cin, planes * 2, 1)
        self.conv2 = conv_bn(planes * 2, planes * 4, 3, stride, 1,
                             groups=groups)
        self.conv3 = conv_bn(planes * 4, planes * 4, 1, relu=False)
        self.se = SEModule(planes * 4, reduction)
        self.downsample = downsample

    def forward(self, x):
        identity = self.downsample(x) if self.downsample is not None else x
        out = self.conv3(self.conv2(self.conv1(x)))
        return F.relu(self.se(out) + identity, inplace=True)


class SENet154(nn.Module):
    """Hu'18 SENet-154 (the zoo's senet.py flagship): 3-conv stem,
    grouped SE bottlenecks [3, 8, 36, 3]."""

    def __init__(self, num_classes=1000, layers=(3, 8, 36, 3), dropout=0.2):
        super().__init__()
        self.stem = nn.Sequential(conv_bn(3, 64, 3, 2, 1),
                                  conv_bn(64, 64, 3, 1, 1),
                                  conv_bn(64, 128, 3, 1, 1),
                                  nn.MaxPool2d(3, 2, ceil_mode=True))
        self.inplanes = 128
        self.layer1 = self._make_layer(64, layers[0], stride=1)
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.dropout = nn.Dropout(dropout)
        self.num_features = 2048
        self.fc = nn.Linear(2048, num_classes)

    def _make_layer(self, planes, blocks, stride):
        downsample = None
        if stride != 1 or self.inplanes != planes * 4:
            downsample = conv_bn(self.inplanes, planes * 4, 1, stride,
                                 relu=False)
        layers = [SEBottleneck154(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * 4
        layers += [SEBottleneck154(self.inplanes, planes)
                   for _ in range(blocks - 1)]
        return nn.Sequential(*layers)

    def forward_features(self, x):
        x = self.stem(x)
        return self.layer4(self.layer3(self.layer2(self.layer1(x))))

    def forward(self, x):
        x = self.forward_features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(self.dropout(x))


# --------------------------------------------------------------- PolyNet ---
class PolyConv(nn.Module):
    """Conv shared across poly paths with per-path BN (PolyNet's trick)."""

    def __init__(self, cin, cout, k, stride=1, padding=0, num_blocks=3):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, k, stride, padding, bias=False)
        self.bns = nn.ModuleList(
            [BatchNorm2d(cout, relu=True) for _ in range(num_blocks)])

    def forward(self, x, idx):
        return self.bns[idx](self.conv(x))


class InceptionResA(nn.Module):
    """Inception-ResNet-A unit (PolyNet stage A building block)."""

    def __init__(self, c=384, scale=1.0):
        super().__init__()
        self.scale = scale
        self.b0 = conv_bn(c, 32, 1)
        self.b1 = nn.Sequential(conv_bn(c, 32, 1), conv_bn(32, 32, 3, 1, 1))
        self.b2 = nn.Sequential(conv_bn(c, 32, 1), conv_bn(32, 48, 3, 1, 1),
                                conv_bn(48, 64, 3, 1, 1))
        self.proj = nn.Conv2d(32 + 32 + 64, c, 1)

    def forward(self, x):
        u = torch.cat([self.b0(x), self.b1(x), self.b2(x)], 1)
        return F.relu(x + self.scale * self.proj(u), inplace=True)


class PolyStageA(nn.Module):
    """poly-3: three residual compositions sharing one InceptionResA."""

    def __init__(self, c=384, scale=0.3):
        super().__init__()
        self.unit = InceptionResA(c, scale=1.0)
        self.scale = scale

    def forward(self, x):
        # x + s*F(x) + s*F(F(x)) + s*F(F(F(x))) (published poly-3 form)
        f1 = self.unit(x)
        f2 = self.unit(f1)
        f3 = self.unit(f2)
        return F.relu(x + self.scale * (f1 + f2 + f3), inplace=True)


class InceptionResB(nn.Module):
    def __init__(self, c=1152, scale=1.0):
        super().__init__()
        self.scale = scale
        self.b0 = conv_bn(c, 192, 1)
        self.b1 = nn.Sequential(conv_bn(c, 128, 1),
                                conv_bn(128, 160, (1, 7), 1, (0, 3)),
                                conv_bn(160, 192, (7, 1), 1, (3, 0)))
        self.proj = nn.Conv2d(192 + 192, c, 1)

    def forward(self, x):
        u = torch.cat([self.b0(x), self.b1(x)], 1)
        return F.relu(x + self.scale * self.proj(u), inplace=True)


class TwoWayB(nn.Module):
    """2-way: sum of two first-order units (PolyNet stage B mixing)."""

    def __init__(self, c=1152, scale=0.3):
        super().__init__()
        self.f = InceptionResB(c, scale=1.0)
        self.g = InceptionResB(c, scale=1.0)
        self.scale = scale

    def forward(self, x):
        return F.relu(x + self.scale * (self.f(x) + self.g(x) - 2 * x),
                      inplace=True)


class InceptionResC(nn.Module):
    def __init__(self, c=2048, scale=1.0):
        super().__init__()
        self.scale = scale
        self.b0 = conv_bn(c, 192, 1)
        self.b1 = nn.Sequential(conv_bn(c, 192, 1),
                                conv_bn(192, 224, (1, 3), 1, (0, 1)),
                                conv_bn(224, 256, (3, 1), 1, (1, 0)))
        self.proj = nn.Conv2d(192 + 256, c, 1)

    def forward(self, x):
        u = torch.cat([self.b0(x), self.b1(x)], 1)
        return F.relu(x + self.scale * self.proj(u), inplace=True)


class ReductionA(nn.Module):
    def __init__(self, cin=384):
        super().__init__()
        self.b0 = conv_bn(cin, 384, 3, 2)
        self.b1 = nn.Sequential(conv_bn(cin, 256, 1),
                                conv_bn(256, 256, 3, 1, 1),
                                conv_bn(256, 384, 3, 2))
        self.pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.pool(x)], 1)


class ReductionB(nn.Module):
    def __init__(self, cin=1152):
        super().__init__()
        self.b0 = nn.Sequential(conv_bn(cin, 256, 1), conv_bn(256, 384, 3, 2))
        self.b1 = nn.Sequential(conv_bn(cin, 256, 1), conv_bn(256, 288, 3, 2))
        self.b2 = nn.Sequential(conv_bn(cin, 256, 1),
                                conv_bn(256, 288, 3, 1, 1),
                                conv_bn(288, 320, 3, 2))
        self.pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.b2(x), self.pool(x)],
                         1)


class PolyNet(nn.Module):
    """Zhang'17 PolyNet: Inception-ResNet-v2 skeleton with poly-3 / 2-way
    polynomial compositions in stages A/B/C."""

    def __init__(self, num_classes=1000, blocks=(10, 10, 5)):
        super().__init__()
        # IR-v2 stem (to 384 channels)
        self.stem = nn.Sequential(
            conv_bn(3, 32, 3, 2), conv_bn(32, 32, 3, 1),
            conv_bn(32, 64, 3, 1, 1), nn.MaxPool2d(3, 2),
            conv_bn(64, 80, 1), conv_bn(80, 192, 3, 1),
            nn.MaxPool2d(3, 2), conv_bn(192, 384, 1))
        self.stage_a = nn.Sequential(*[PolyStageA(384)
                                       for _ in range(blocks[0])])
        self.red_a = ReductionA(384)  # -> 1152
        self.stage_b = nn.Sequential(*[TwoWayB(1152)
                                       for _ in range(blocks[1])])
        self.red_b = ReductionB(1152)  # -> 1152+384+288+320 = 2144
        self.conv_c = conv_bn(2144, 2048, 1)
        self.stage_c = nn.Sequential(*[InceptionResC(2048, scale=0.2)
                                       for _ in range(blocks[2])])
        self.num_features = 2048
        self.dropout = nn.Dropout(0.2)
        self.fc = nn.Linear(2048, num_classes)

    def forward_features(self, x):
        x = self.stem(x)
        x = self.red_a(self.stage_a(x))
        x = self.red_b(self.stage_b(x))
        return self.stage_c(self.conv_c(x))

    def forward(self, x):
        x = self.forward_features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(self.dropout(x))


# -------------------------------------------------------------- NASNet-A ---
class BranchSeparables(nn.Module):
    """NASNet separable: relu -> sep(dw+pw) -> bn -> relu -> sep -> bn."""

    def __init__(self, cin, cout, k, stride=1):
        super().__init__()
        pad = k // 2
        self.op = nn.Sequential(
            nn.ReLU(inplace=False),
            SeparableConv2d(cin, cin, k, stride, pad), BatchNorm2d(cin),
            nn.ReLU(inplace=False),
            SeparableConv2d(cin, cout, k, 1, pad), BatchNorm2d(cout))

    def forward(self, x):
        return self.op(x)


class FitReduce(nn.Module):
    """Path-split 1x1 reduction used to align the previous-previous state."""

    def __init__(self, cin, cout):
        super().__init__()
        self.relu = nn.ReLU(inplace=False)
        self.p1 = nn.Sequential(nn.AvgPool2d(1, 2),
                                nn.Conv2d(cin, cout // 2, 1, bias=False))
        self.p2 = nn.Sequential(nn.AvgPool2d(1, 2),
                                nn.Conv2d(cin, cout - cout // 2, 1,
                                          bias=False))
        self.bn = BatchNorm2d(cout)

    def forward(self, x):
        x = self.relu(x)
        p1 = self.p1(x)
        p2 = self.p2(F.pad(x, (0, 1, 0, 1))[:, :, 1:, 1:])
        return self.bn(torch.cat([p1, p2], 1))


def _fit(cin, cout, reduce_spatial):
    if reduce_spatial:
        return FitReduce(cin, cout)
    return nn.Sequential(nn.ReLU(inplace=False),
                         nn.Conv2d(cin, cout, 1, bias=False),
                         BatchNorm2d(cout))


class NormalCell(nn.Module):
    """NASNet-A normal cell (the published 5-pair DAG)."""

    def __init__(self, c_prev_prev, c_prev, c, prev_reduced=False):
        super().__init__()
        self.fit_pp = _fit(c_prev_prev, c, prev_reduced)
        self.fit_p = _fit(c_prev, c, False)
        self.b1_l = BranchSeparables(c, c, 5)
        self.b1_r = BranchSeparables(c, c, 3)
        self.b2_l = BranchSeparables(c, c, 5)
        self.b2_r = BranchSeparables(c, c, 3)
        self.b3_l = nn.AvgPool2d(3, 1, 1)
        self.b4_l = nn.AvgPool2d(3, 1, 1)
        self.b4_r = nn.AvgPool2d(3, 1, 1)
        self.b5_l = BranchSeparables(c, c, 3)
        self.out_channels = 6 * c

    def forward(self, x_pp, x_p):
        h_pp = self.fit_pp(x_pp)
        h_p = self.fit_p(x_p)
        y1 = self.b1_l(h_p) + self.b1_r(h_pp)
        y2 = self.b2_l(h_pp) + self.b2_r(h_pp)
        y3 = self.b3_l(h_p) + h_pp
        y4 = self.b4_l(h_pp) + self.b4_r(h_pp)
        y5 = self.b5_l(h_p) + h_p
        return torch.cat([h_pp, y1, y2, y3, y4, y5], 1)


class ReductionCell(nn.Module):
    """NASNet-A reduction cell (stride-2 DAG)."""

    def __init__(self, c_prev_prev, c_prev, c, prev_reduced=False):
        super().__init__()
        self.fit_pp = _fit(c_prev_prev, c, prev_reduced)
        self.fit_p = _fit(c_prev, c, False)
        self.b1_l = BranchSeparables(c, c, 5, 2)
        self.b1_r = BranchSeparables(c, c, 7, 2)
        self.b2_l = nn.MaxPool2d(3, 2, 1)
        self.b2_r = BranchSeparables(c, c, 7, 2)
        self.b3_l = nn.AvgPool2d(3, 2, 1)
        self.b3_r = BranchSeparables(c, c, 5, 2)
        self.b4_r = nn.AvgPool2d(3, 1, 1)
        self.b5_l = BranchSeparables(c, c, 3)
        self.b5_r = nn.MaxPool2d(3, 2, 1)
        self.out_channels = 4 * c

    def forward(self, x_pp, x_p):
        h_pp = self.fit_pp(x_pp)
        h_p = self.fit_p(x_p)
        y1 = self.b1_l(h_p) + self.b1_r(h_pp)
        y2 = self.b2_l(h_p) + self.b2_r(h_pp)
        y3 = self.b3_l(h_p) + self.b3_r(h_pp)
        y4 = self.b4_r(y1) + y2
        y5 = self.b5_l(y1) + self.b5_r(h_p)
        return torch.cat([y2, y3, y4, y5], 1)


class NASNetA(nn.Module):
    """Zoph'18 NASNet-A: stem -> 2 reductions -> 3x(N normal + reduction)
    stacks. `num_cells`/`filters` scale the family (4@1056 ... 6@4032)."""

    def __init__(self, num_classes=1000, num_cells=4, filters=44,
                 stem_filters=32):
        super().__init__()
        self.stem = conv_bn(3, stem_filters, 3, 2, 0, relu=False)
        f = filters
        # stem cells: two reduction cells bringing channels up
        self.cell_s1 = ReductionCell(stem_filters, stem_filters, f // 4)
        self.cell_s2 = ReductionCell(stem_filters, self.cell_s1.out_channels,
                                     f // 2, prev_reduced=True)
        cells = []
        c_pp, c_p = self.cell_s1.out_channels, self.cell_s2.out_channels
        prev_reduced = True
        for stage in range(3):
            for i in range(num_cells):
                cell = NormalCell(c_pp, c_p, f, prev_reduced and i == 0)
                cells.append(cell)
                c_pp, c_p = c_p, cell.out_channels
            if stage < 2:
                f *= 2
                red = ReductionCell(c_pp, c_p, f)
                cells.append(red)
                c_pp, c_p = c_p, red.out_channels
                prev_reduced = True
        self.cells = nn.ModuleList(cells)
        self.num_cells = num_cells
        self.num_features = c_p
        self.dropout = nn.Dropout(0.5)
        self.fc = nn.Linear(c_p, num_classes)

    def forward_features(self, x):
        x = self.stem(x)
        s1 = self.cell_s1(x, x)
        s2 = self.cell_s2(x, s1)
        x_pp, x_p = s1, s2
        for cell in self.cells:
            out = cell(x_pp, x_p)
            x_pp, x_p = x_p, out
        return F.relu(x_p, inplace=True)

    def forward(self, x):
        x = self.forward_features(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(self.dropout(x))


@register_model
def xception(num_classes=1000, **kw):
    return Xception(num_classes=num_classes, **kw)


@register_model
def senet154(num_classes=1000, **kw):
    return SENet154(num_classes=num_classes, **kw)


@register_model
def polynet(num_classes=1000, **kw):
    return PolyNet(num_classes=num_classes, **kw)


@register_model
def nasnet_a_large(num_classes=1000, **kw):
    kw.setdefault("num_cells", 6)
    kw.setdefault("filters", 168)
    kw.setdefault("stem_filters", 96)
    return NASNetA(num_classes=num_classes, **kw)


@register_model
def nasnet_a_mobile(num_classes=1000, **kw):
    kw.setdefault("num_cells", 4)
    kw.setdefault("filters", 44)
    kw.setdefault("stem_filters", 32)
    return NASNetA(num_classes=num_classes, **kw)
