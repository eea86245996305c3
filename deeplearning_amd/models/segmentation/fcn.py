"""FCN-ResNet50/101 with optional aux head, on the dilated ResNet backbone.

Reference parity: Image_segmentation/FCN/models/networks.py — re-designed on
this repo's ResNet (replace_stride_with_dilation) + IntermediateLayerGetter.
"""
from __future__ import annotations

from collections import OrderedDict

import torch.nn as nn
import torch.nn.functional as F

from ...ops import BatchNorm2d
from ..classification.resnet import ResNet, Bottleneck
from ..registry import register_model


class IntermediateLayerGetter(nn.ModuleDict):
    """Run a backbone, returning named intermediate features
    (ref Image_segmentation/DeepLabV3/models/deeplabv3.py:15)."""

    def __init__(self, model: nn.Module, return_layers: dict):
        orig = dict(return_layers)
        layers = OrderedDict()
        for name, module in model.named_children():
            layers[name] = module
            if name in return_layers:
                del return_layers[name]
            if not return_layers:
                break
        super().__init__(layers)
        self.return_layers = orig

    def forward(self, x):
        out = OrderedDict()
        for name, module in self.items():
            x = module(x)
            if name in self.return_layers:
                out[self.return_layers[name]] = x
        return out


class FCNHead(nn.Sequential):
    def __init__(self, cin, num_classes):
        inter = cin // 4
        super().__init__(
            nn.Conv2d(cin, inter, 3, padding=1, bias=False),
            BatchNorm2d(inter, relu=True),
            nn.Dropout(0.1),
            nn.Conv2d(inter, num_classes, 1))


class FCN(nn.Module):
    def __init__(self, backbone, classifier, aux_classifier=None):
        super().__init__()
        self.backbone = backbone
        self.classifier = classifier
        self.aux_classifier = aux_classifier

    def forward(self, x):
        size = x.shape[-2:]
        feats = self.backbone(x)
        out = self.classifier(feats["out"])
        out = F.interpolate(out, size=size, mode="bilinear", align_corners=False)
        result = OrderedDict(out=out)
        if self.aux_classifier is not None:
            aux = self.aux_classifier(feats["aux"])
            result["aux"] = F.interpolate(aux, size=size, mode="bilinear",
                                          align_corners=False)
        return result


def _dilated_resnet(layers):
    return ResNet(Bottleneck, layers, include_top=False,
                  replace_stride_with_dilation=[False, True, True])


def _build_fcn(layers, num_classes, aux):
    backbone = _dilated_resnet(layers)
    return_layers = {"layer4": "out"}
    if aux:
        return_layers["layer3"] = "aux"
    body = IntermediateLayerGetter(backbone, return_layers)
    aux_head = FCNHead(1024, num_classes) if aux else None
    return FCN(body, FCNHead(2048, num_classes), aux_head)


@register_model
def fcn_resnet50(num_classes=21, aux_loss=True, **kw):
    return _build_fcn([3, 4, 6, 3], num_classes, aux_loss)


@register_model
def fcn_resnet101(num_classes=21, aux_loss=True, **kw):
    return _build_fcn([3, 4, 23, 3], num_classes, aux_loss)
