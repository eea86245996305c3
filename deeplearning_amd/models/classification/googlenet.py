"""GoogLeNet (Inception v1) with optional aux classifiers.

Reference parity: classification/GoogleNet/models/googlenet.py:282 — re-designed
on fused HIP BN+ReLU (BasicConv2d = conv -> one fused BN+ReLU kernel).
"""
from __future__ import annotations

from collections import namedtuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import BatchNorm2d
from ..registry import register_model

GoogLeNetOutputs = namedtuple("GoogLeNetOutputs", ["logits", "aux_logits2",
                                                   "aux_logits1"])


class BasicConv2d(nn.Module):
    def __init__(self, cin, cout, **kw):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, bias=False, **kw)
        self.bn = BatchNorm2d(cout, relu=True, eps=0.001)

    def forward(self, x):
        return self.bn(self.conv(x))


class Inception(nn.Module):
    def __init__(self, cin, ch1x1, ch3x3red, ch3x3, ch5x5red, ch5x5, pool_proj):
        super().__init__()
        self.branch1 = BasicConv2d(cin, ch1x1, kernel_size=1)
        self.branch2 = nn.Sequential(
            BasicConv2d(cin, ch3x3red, kernel_size=1),
            BasicConv2d(ch3x3red, ch3x3, kernel_size=3, padding=1))
        self.branch3 = nn.Sequential(
            BasicConv2d(cin, ch5x5red, kernel_size=1),
            BasicConv2d(ch5x5red, ch5x5, kernel_size=3, padding=1))
        self.branch4 = nn.Sequential(
            nn.MaxPool2d(3, stride=1, padding=1, ceil_mode=True),
            BasicConv2d(cin, pool_proj, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.branch1(x), self.branch2(x), self.branch3(x),
                          self.branch4(x)], 1)


class InceptionAux(nn.Module):
    def __init__(self, cin, num_classes, dropout=0.7):
        super().__init__()
        self.averagePool = nn.AvgPool2d(5, stride=3)
        self.conv = BasicConv2d(cin, 128, kernel_size=1)
        self.fc1 = nn.Linear(2048, 1024)
        self.fc2 = nn.Linear(1024, num_classes)
        self.dropout = dropout

    def forward(self, x):
        x = self.conv(self.averagePool(x)).flatten(1)
        x = F.dropout(x, self.dropout, training=self.training)
        x = F.relu(self.fc1(x), inplace=True)
        x = F.dropout(x, self.dropout, training=self.training)
        return self.fc2(x)


class GoogLeNet(nn.Module):
    def __init__(self, num_classes=1000, aux_logits=True, init_weights=True):
        super().__init__()
        self.aux_logits = aux_logits
        self.conv1 = BasicConv2d(3, 64, kernel_size=7, stride=2, padding=3)
        self.maxpool1 = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.conv2 = BasicConv2d(64, 64, kernel_size=1)
        self.conv3 = BasicConv2d(64, 192, kernel_size=3, padding=1)
        self.maxpool2 = nn.MaxPool2d(3, stride=2, ceil_mode=True)

        self.inception3a = Inception(192, 64, 96, 128, 16, 32, 32)
        self.inception3b = Inception(256, 128, 128, 192, 32, 96, 64)
        self.maxpool3 = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.inception4a = Inception(480, 192, 96, 208, 16, 48, 64)
        self.inception4b = Inception(512, 160, 112, 224, 24, 64, 64)
        self.inception4c = Inception(512, 128, 128, 256, 24, 64, 64)
        self.inception4d = Inception(512, 112, 144, 288, 32, 64, 64)
        self.inception4e = Inception(528, 256, 160, 320, 32, 128, 128)
        self.maxpool4 = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.inception5a = Inception(832, 256, 160, 320, 32, 128, 128)
        self.inception5b = Inception(832, 384, 192, 384, 48, 128, 128)

        if aux_logits:
            self.aux1 = InceptionAux(512, num_classes)
            self.aux2 = InceptionAux(528, num_classes)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.dropout = nn.Dropout(0.4)
        self.fc = nn.Linear(1024, num_classes)

        if init_weights:
            for m in self.modules():
                if isinstance(m, (nn.Conv2d, nn.Linear)):
                    nn.init.trunc_normal_(m.weight, mean=0.0, std=0.01, a=-2, b=2)
                    if m.bias is not None:
                        nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.maxpool1(self.conv1(x))
        x = self.maxpool2(self.conv3(self.conv2(x)))
        x = self.inception3b(self.inception3a(x))
        x = self.maxpool3(x)
        x = self.inception4a(x)
        aux1 = self.aux1(x) if self.aux_logits and self.training else None
        x = self.inception4d(self.inception4c(self.inception4b(x)))
        aux2 = self.aux2(x) if self.aux_logits and self.training else None
        x = self.maxpool4(self.inception4e(x))
        x = self.inception5b(self.inception5a(x))
        x = self.dropout(self.avgpool(x).flatten(1))
        x = self.fc(x)
        if self.aux_logits and self.training:
            return GoogLeNetOutputs(x, aux2, aux1)
        return x


@register_model
def googlenet(num_classes=1000, **kw):
    return GoogLeNet(num_classes=num_classes, **kw)
