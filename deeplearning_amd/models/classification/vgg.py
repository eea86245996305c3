"""VGG-11/13/16/19 (+BN variants with the framework's fused HIP BN+ReLU).

Reference parity: classification/vggNet/models/network.py — re-designed: BN
variants use BatchNorm2d(relu=True) so each conv is followed by ONE fused
kernel instead of separate BN and ReLU launches.
"""
from __future__ import annotations

import torch.nn as nn

from ...ops import BatchNorm2d
from ..registry import register_model

_CFGS = {
    "A": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "B": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "D": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512, "M",
          512, 512, 512, "M"],
    "E": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M", 512, 512, 512, 512,
          "M", 512, 512, 512, 512, "M"],
}


def _make_features(cfg, batch_norm):
    layers, cin = [], 3
    for v in cfg:
        if v == "M":
            layers.append(nn.MaxPool2d(2, 2))
        else:
            if batch_norm:
                layers += [nn.Conv2d(cin, v, 3, padding=1, bias=False),
                           BatchNorm2d(v, relu=True)]
            else:
                layers += [nn.Conv2d(cin, v, 3, padding=1),
                           nn.ReLU(inplace=True)]
            cin = v
    return nn.Sequential(*layers)


class VGG(nn.Module):
    def __init__(self, cfg, num_classes=1000, batch_norm=False, dropout=0.5,
                 init_weights=True):
        super().__init__()
        self.features = _make_features(_CFGS[cfg], batch_norm)
        self.avgpool = nn.AdaptiveAvgPool2d(7)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(True), nn.Dropout(dropout),
            nn.Linear(4096, 4096), nn.ReLU(True), nn.Dropout(dropout),
            nn.Linear(4096, num_classes))
        if init_weights:
            for m in self.modules():
                if isinstance(m, nn.Conv2d):
                    nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                            nonlinearity="relu")
                    if m.bias is not None:
                        nn.init.zeros_(m.bias)
                elif isinstance(m, nn.Linear):
                    nn.init.normal_(m.weight, 0, 0.01)
                    nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.features(x)
        x = self.avgpool(x).flatten(1)
        return self.classifier(x)


@register_model
def vgg11(num_classes=1000, **kw):
    return VGG("A", num_classes=num_classes, **kw)


@register_model
def vgg13(num_classes=1000, **kw):
    return VGG("B", num_classes=num_classes, **kw)


@register_model
def vgg16(num_classes=1000, **kw):
    return VGG("D", num_classes=num_classes, **kw)


@register_model
def vgg19(num_classes=1000, **kw):
    return VGG("E", num_classes=num_classes, **kw)


@register_model
def vgg16_bn(num_classes=1000, **kw):
    return VGG("D", num_classes=num_classes, batch_norm=True, **kw)


@register_model
def vgg19_bn(num_classes=1000, **kw):
    return VGG("E", num_classes=num_classes, batch_norm=True, **kw)
