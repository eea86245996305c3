"""ShuffleNet v1 / v2 (channel shuffle networks).

Reference parity: classification/ShuffleNet/models/{shufflenetv1,shufflenetv2}.py
— re-designed on fused HIP BN+ReLU; channel_shuffle is a view/transpose (a
dedicated kernel buys nothing: it is pure data movement the following
grouped conv re-reads anyway).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import BatchNorm2d
from ..registry import register_model


def channel_shuffle(x: torch.Tensor, groups: int) -> torch.Tensor:
    b, c, h, w = x.shape
    x = x.view(b, groups, c // groups, h, w).transpose(1, 2).contiguous()
    return x.view(b, c, h, w)


# ---------------------------------------------------------------- v1 ----
class ShuffleUnitV1(nn.Module):
    def __init__(self, cin, cout, groups, stride, first_group_conv=True):
        super().__init__()
        self.stride = stride
        mid = cout // 4
        if stride == 2:
            cout -= cin
        g = groups if first_group_conv else 1
        self.gconv1 = nn.Sequential(
            nn.Conv2d(cin, mid, 1, groups=g, bias=False),
            BatchNorm2d(mid, relu=True))
        self.groups = groups
        self.dwconv = nn.Sequential(
            nn.Conv2d(mid, mid, 3, stride, 1, groups=mid, bias=False),
            BatchNorm2d(mid))
        self.gconv2 = nn.Sequential(
            nn.Conv2d(mid, cout, 1, groups=groups, bias=False),
            BatchNorm2d(cout))
        self.shortcut = nn.AvgPool2d(3, 2, 1) if stride == 2 else nn.Identity()
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        out = self.gconv1(x)
        out = channel_shuffle(out, self.groups)
        out = self.gconv2(self.dwconv(out))
        if self.stride == 2:
            return self.relu(torch.cat([self.shortcut(x), out], 1))
        return self.relu(x + out)


class ShuffleNetV1(nn.Module):
    _stage_out = {1: [144, 288, 576], 2: [200, 400, 800], 3: [240, 480, 960],
                  4: [272, 544, 1088], 8: [384, 768, 1536]}

    def __init__(self, groups=3, num_classes=1000):
        super().__init__()
        out = self._stage_out[groups]
        self.conv1 = nn.Sequential(
            nn.Conv2d(3, 24, 3, 2, 1, bias=False), BatchNorm2d(24, relu=True))
        self.maxpool = nn.MaxPool2d(3, 2, 1)
        cin = 24
        stages = []
        for i, (reps, cout) in enumerate(zip([4, 8, 4], out)):
            units = [ShuffleUnitV1(cin, cout, groups, 2,
                                   first_group_conv=(i != 0))]
            cin = cout
            units += [ShuffleUnitV1(cin, cout, groups, 1) for _ in range(reps - 1)]
            stages.append(nn.Sequential(*units))
        self.stage2, self.stage3, self.stage4 = stages
        self.gap = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(out[-1], num_classes)

    def forward(self, x):
        x = self.maxpool(self.conv1(x))
        x = self.stage4(self.stage3(self.stage2(x)))
        return self.fc(self.gap(x).flatten(1))


# ---------------------------------------------------------------- v2 ----
class InvertedResidualV2(nn.Module):
    def __init__(self, cin, cout, stride):
        super().__init__()
        self.stride = stride
        branch_c = cout // 2
        if stride > 1:
            self.branch1 = nn.Sequential(
                nn.Conv2d(cin, cin, 3, stride, 1, groups=cin, bias=False),
                BatchNorm2d(cin),
                nn.Conv2d(cin, branch_c, 1, bias=False),
                BatchNorm2d(branch_c, relu=True))
        else:
            self.branch1 = nn.Identity()
        in2 = cin if stride > 1 else branch_c
        self.branch2 = nn.Sequential(
            nn.Conv2d(in2, branch_c, 1, bias=False),
            BatchNorm2d(branch_c, relu=True),
            nn.Conv2d(branch_c, branch_c, 3, stride, 1, groups=branch_c,
                      bias=False),
            BatchNorm2d(branch_c),
            nn.Conv2d(branch_c, branch_c, 1, bias=False),
            BatchNorm2d(branch_c, relu=True))

    def forward(self, x):
        if self.stride == 1:
            x1, x2 = x.chunk(2, dim=1)
            out = torch.cat([x1, self.branch2(x2)], 1)
        else:
            out = torch.cat([self.branch1(x), self.branch2(x)], 1)
        return channel_shuffle(out, 2)


class ShuffleNetV2(nn.Module):
    def __init__(self, stages_repeats, stages_out_channels, num_classes=1000):
        super().__init__()
        self.conv1 = nn.Sequential(
            nn.Conv2d(3, stages_out_channels[0], 3, 2, 1, bias=False),
            BatchNorm2d(stages_out_channels[0], relu=True))
        self.maxpool = nn.MaxPool2d(3, 2, 1)
        cin = stages_out_channels[0]
        for name, reps, cout in zip(["stage2", "stage3", "stage4"],
                                    stages_repeats, stages_out_channels[1:]):
            seq = [InvertedResidualV2(cin, cout, 2)]
            seq += [InvertedResidualV2(cout, cout, 1) for _ in range(reps - 1)]
            setattr(self, name, nn.Sequential(*seq))
            cin = cout
        self.conv5 = nn.Sequential(
            nn.Conv2d(cin, stages_out_channels[-1], 1, bias=False),
            BatchNorm2d(stages_out_channels[-1], relu=True))
        self.gap = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(stages_out_channels[-1], num_classes)

    def forward(self, x):
        x = self.maxpool(self.conv1(x))
        x = self.stage4(self.stage3(self.stage2(x)))
        x = self.conv5(x)
        return self.fc(self.gap(x).flatten(1))


@register_model
def shufflenet_v1_g3(num_classes=1000, **kw):
    return ShuffleNetV1(groups=3, num_classes=num_classes, **kw)


@register_model
def shufflenet_v2_x0_5(num_classes=1000, **kw):
    return ShuffleNetV2([4, 8, 4], [24, 48, 96, 192, 1024],
                        num_classes=num_classes, **kw)


@register_model
def shufflenet_v2_x1_0(num_classes=1000, **kw):
    return ShuffleNetV2([4, 8, 4], [24, 116, 232, 464, 1024],
                        num_classes=num_classes, **kw)


@register_model
def shufflenet_v2_x2_0(num_classes=1000, **kw):
    return ShuffleNetV2([4, 8, 4], [24, 244, 488, 976, 2048],
                        num_classes=num_classes, **kw)
