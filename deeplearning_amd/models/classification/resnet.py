"""ResNet / ResNeXt / WideResNet family, built on the framework's fused ops:
BatchNorm2d(relu=True) (HIP fused BN+ReLU) and add_relu (fused residual join).

Reference parity: classification/resnet/models/networks.py (BasicBlock:38,
Bottleneck:78, ResNet:127, factories :235-341) — re-designed, not translated:
every 1x1 conv (bottleneck conv1/conv3, downsample) runs the hand-written
implicit-GEMM MFMA kernel with the BatchNorm stats pass fused into the conv
epilogue (csrc/conv1x1.hip); 3x3 convs run MIOpen by default with a fully
hand-written TAPS=9 route available (DLA_CONV3X3=1, see ROADMAP); each
residual join is one fused add+relu kernel.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import BatchNorm2d, add_relu
from ...ops.conv1x1 import conv_bn
from ..registry import register_model


def _downsample_fwd(downsample, x):
    """Route a (conv1x1, bn) downsample Sequential through the fused
    implicit-GEMM conv+BN path when it applies."""
    if (isinstance(downsample, nn.Sequential) and len(downsample) == 2 and
            isinstance(downsample[0], nn.Conv2d)):
        return conv_bn(x, downsample[0], downsample[1])
    return downsample(x)


class _NormRelu(nn.Sequential):
    """norm + ReLU for norm layers without a fused relu flag (e.g. FrozenBatchNorm2d)."""

    def __init__(self, norm, c):
        super().__init__(norm(c), nn.ReLU(inplace=True))


def make_norm(norm_layer, c, relu=False):
    """Build norm_layer(c) with fused relu when supported, else norm+ReLU."""
    if not relu:
        return norm_layer(c)
    try:
        return norm_layer(c, relu=True)
    except TypeError:
        return _NormRelu(norm_layer, c)


def conv3x3(cin, cout, stride=1, groups=1, dilation=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=dilation, groups=groups,
                     bias=False, dilation=dilation)


def conv1x1(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None, groups=1,
                 base_width=64, dilation=1, norm_layer=None):
        super().__init__()
        norm_layer = norm_layer or BatchNorm2d
        if groups != 1 or base_width != 64:
            raise ValueError("BasicBlock only supports groups=1, base_width=64")
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = make_norm(norm_layer, planes, relu=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = norm_layer(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        out = conv_bn(x, self.conv1, self.bn1)
        out = conv_bn(out, self.conv2, self.bn2)
        if self.downsample is not None:
            identity = _downsample_fwd(self.downsample, x)
        return add_relu(out, identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None, groups=1,
                 base_width=64, dilation=1, norm_layer=None):
        super().__init__()
        norm_layer = norm_layer or BatchNorm2d
        width = int(planes * (base_width / 64.0)) * groups
        self.conv1 = conv1x1(inplanes, width)
        self.bn1 = make_norm(norm_layer, width, relu=True)
        self.conv2 = conv3x3(width, width, stride, groups, dilation)
        self.bn2 = make_norm(norm_layer, width, relu=True)
        self.conv3 = conv1x1(width, planes * self.expansion)
        self.bn3 = norm_layer(planes * self.expansion)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        # conv1/conv3 (1x1) ride the hand-written MFMA implicit-GEMM kernel
        # with the BN stats pass fused into the conv epilogue (conv1x1.hip);
        # conv2 (3x3) uses MIOpen by default (the TAPS=9 route is opt-in).
        out = conv_bn(x, self.conv1, self.bn1)
        out = conv_bn(out, self.conv2, self.bn2)  # 3x3 s1 -> TAPS=9 kernel
        out = conv_bn(out, self.conv3, self.bn3)
        if self.downsample is not None:
            identity = _downsample_fwd(self.downsample, x)
        return add_relu(out, identity)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, groups=1, width_per_group=64,
                 replace_stride_with_dilation=None, norm_layer=None,
                 zero_init_residual=False, include_top=True):
        super().__init__()
        self._norm_layer = norm_layer or BatchNorm2d
        self.inplanes = 64
        self.dilation = 1
        replace_stride_with_dilation = replace_stride_with_dilation or [False] * 3
        self.groups = groups
        self.base_width = width_per_group
        self.include_top = include_top

        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = make_norm(self._norm_layer, 64, relu=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2,
                                       dilate=replace_stride_with_dilation[0])
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2,
                                       dilate=replace_stride_with_dilation[1])
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2,
                                       dilate=replace_stride_with_dilation[2])
        if include_top:
            self.avgpool = nn.AdaptiveAvgPool2d(1)
            self.fc = nn.Linear(512 * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, (nn.BatchNorm2d, nn.GroupNorm)):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.zeros_(m.bn3.weight)
                elif isinstance(m, BasicBlock):
                    nn.init.zeros_(m.bn2.weight)

    def _make_layer(self, block, planes, blocks, stride=1, dilate=False):
        norm_layer = self._norm_layer
        downsample = None
        prev_dilation = self.dilation
        if dilate:
            self.dilation *= stride
            stride = 1
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                norm_layer(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample, self.groups,
                        self.base_width, prev_dilation, norm_layer)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes, groups=self.groups,
                                base_width=self.base_width, dilation=self.dilation,
                                norm_layer=norm_layer))
        return nn.Sequential(*layers)

    def forward_features(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return x

    def forward(self, x):
        x = self.forward_features(x)
        if self.include_top:
            x = self.avgpool(x)
            x = torch.flatten(x, 1)
            x = self.fc(x)
        return x


@register_model
def resnet18(num_classes=1000, **kw):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes=num_classes, **kw)


@register_model
def resnet34(num_classes=1000, **kw):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes=num_classes, **kw)


@register_model
def resnet50(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes=num_classes, **kw)


@register_model
def resnet101(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes=num_classes, **kw)


@register_model
def resnet152(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes=num_classes, **kw)


@register_model
def resnext50_32x4d(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes=num_classes, groups=32,
                  width_per_group=4, **kw)


@register_model
def resnext101_32x8d(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes=num_classes, groups=32,
                  width_per_group=8, **kw)


@register_model
def wide_resnet50_2(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes=num_classes,
                  width_per_group=128, **kw)


@register_model
def wide_resnet101_2(num_classes=1000, **kw):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes=num_classes,
                  width_per_group=128, **kw)
