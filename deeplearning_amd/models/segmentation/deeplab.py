"""DeepLabV3 and DeepLabV3+ (ASPP, dilated ResNet backbone).

Reference parity: Image_segmentation/DeepLabV3/models/deeplabv3.py
(ASPP:143-199, deeplabv3_resnet50:201) and DeepLabV3Plus/models/deeplabv3plus.py
— re-designed on this repo's dilated ResNet + fused HIP BN+ReLU.
"""
from __future__ import annotations

from collections import OrderedDict

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import BatchNorm2d
from ..classification.resnet import ResNet, Bottleneck
from ..registry import register_model
from .fcn import FCNHead, IntermediateLayerGetter


class ASPPConv(nn.Sequential):
    def __init__(self, cin, cout, dilation):
        super().__init__(
            nn.Conv2d(cin, cout, 3, padding=dilation, dilation=dilation,
                      bias=False),
            BatchNorm2d(cout, relu=True))


class ASPPPooling(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.pool = nn.Sequential(
            nn.AdaptiveAvgPool2d(1),
            nn.Conv2d(cin, cout, 1, bias=False))
        self.bn = BatchNorm2d(cout, relu=True)

    def forward(self, x):
        size = x.shape[-2:]
        y = self.bn(self.pool(x))
        return F.interpolate(y, size=size, mode="bilinear", align_corners=False)


class ASPP(nn.Module):
    def __init__(self, cin, atrous_rates=(12, 24, 36), cout=256):
        super().__init__()
        mods = [nn.Sequential(nn.Conv2d(cin, cout, 1, bias=False),
                              BatchNorm2d(cout, relu=True))]
        mods += [ASPPConv(cin, cout, r) for r in atrous_rates]
        mods.append(ASPPPooling(cin, cout))
        self.convs = nn.ModuleList(mods)
        self.project = nn.Sequential(
            nn.Conv2d(len(mods) * cout, cout, 1, bias=False),
            BatchNorm2d(cout, relu=True), nn.Dropout(0.5))

    def forward(self, x):
        return self.project(torch.cat([m(x) for m in self.convs], dim=1))


class DeepLabHead(nn.Sequential):
    def __init__(self, cin, num_classes):
        super().__init__(
            ASPP(cin),
            nn.Conv2d(256, 256, 3, padding=1, bias=False),
            BatchNorm2d(256, relu=True),
            nn.Conv2d(256, num_classes, 1))


class DeepLabV3(nn.Module):
    def __init__(self, backbone, classifier, aux_classifier=None):
        super().__init__()
        self.backbone = backbone
        self.classifier = classifier
        self.aux_classifier = aux_classifier

    def forward(self, x):
        size = x.shape[-2:]
        feats = self.backbone(x)
        out = F.interpolate(self.classifier(feats["out"]), size=size,
                            mode="bilinear", align_corners=False)
        result = OrderedDict(out=out)
        if self.aux_classifier is not None:
            result["aux"] = F.interpolate(
                self.aux_classifier(feats["aux"]), size=size,
                mode="bilinear", align_corners=False)
        return result


class DeepLabV3Plus(nn.Module):
    """ASPP + low-level-feature decoder branch
    (ref Image_segmentation/DeepLabV3Plus/models/deeplabv3plus.py)."""

    def __init__(self, backbone, cin=2048, low_c=256, num_classes=21):
        super().__init__()
        self.backbone = backbone
        self.aspp = ASPP(cin, (6, 12, 18))
        self.low_proj = nn.Sequential(
            nn.Conv2d(low_c, 48, 1, bias=False), BatchNorm2d(48, relu=True))
        self.decoder = nn.Sequential(
            nn.Conv2d(256 + 48, 256, 3, padding=1, bias=False),
            BatchNorm2d(256, relu=True),
            nn.Conv2d(256, 256, 3, padding=1, bias=False),
            BatchNorm2d(256, relu=True),
            nn.Conv2d(256, num_classes, 1))

    def forward(self, x):
        size = x.shape[-2:]
        feats = self.backbone(x)
        low = self.low_proj(feats["low"])
        out = self.aspp(feats["out"])
        out = F.interpolate(out, size=low.shape[-2:], mode="bilinear",
                            align_corners=False)
        out = self.decoder(torch.cat([out, low], dim=1))
        out = F.interpolate(out, size=size, mode="bilinear", align_corners=False)
        return {"out": out}


def _dilated_backbone(layers, return_layers):
    net = ResNet(Bottleneck, layers, include_top=False,
                 replace_stride_with_dilation=[False, True, True])
    return IntermediateLayerGetter(net, return_layers)




@register_model
def deeplabv3_resnet50(num_classes=21, aux_loss=True, **kw):
    rl = {"layer4": "out"}
    if aux_loss:
        rl["layer3"] = "aux"
    backbone = _dilated_backbone([3, 4, 6, 3], rl)
    aux = FCNHead(1024, num_classes) if aux_loss else None
    return DeepLabV3(backbone, DeepLabHead(2048, num_classes), aux)


@register_model
def deeplabv3_resnet101(num_classes=21, aux_loss=True, **kw):
    rl = {"layer4": "out"}
    if aux_loss:
        rl["layer3"] = "aux"
    backbone = _dilated_backbone([3, 4, 23, 3], rl)
    aux = FCNHead(1024, num_classes) if aux_loss else None
    return DeepLabV3(backbone, DeepLabHead(2048, num_classes), aux)


@register_model
def deeplabv3plus_resnet50(num_classes=21, **kw):
    backbone = _dilated_backbone([3, 4, 6, 3],
                                 {"layer1": "low", "layer4": "out"})
    return DeepLabV3Plus(backbone, num_classes=num_classes)
