"""Feature Pyramid Network + ResNet-FPN backbone builder.

Reference parity: detection/FPN/fpn_model.py:1-268 and
detection/fasterRcnn/models/backbone/{resnet50_fpn.py:195,
feature_pyramid_network.py} — re-designed: lateral 1x1 + top-down nearest
upsample + 3x3 smooth, with an optional extra max-pool (P6) or P6/P7 conv
levels (RetinaNet).
"""
from __future__ import annotations

from collections import OrderedDict

import torch.nn.functional as F
from torch import nn

from ...ops import FrozenBatchNorm2d
from ..classification.resnet import ResNet, Bottleneck
from ..segmentation.fcn import IntermediateLayerGetter


class FeaturePyramidNetwork(nn.Module):
    def __init__(self, in_channels_list, out_channels=256,
                 extra_blocks="maxpool"):
        super().__init__()
        self.inner_blocks = nn.ModuleList(
            [nn.Conv2d(c, out_channels, 1) for c in in_channels_list])
        self.layer_blocks = nn.ModuleList(
            [nn.Conv2d(out_channels, out_channels, 3, padding=1)
             for _ in in_channels_list])
        self.extra_blocks = extra_blocks
        if extra_blocks == "p6p7":
            self.p6 = nn.Conv2d(out_channels, out_channels, 3, 2, 1)
            self.p7 = nn.Conv2d(out_channels, out_channels, 3, 2, 1)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_uniform_(m.weight, a=1)
                nn.init.zeros_(m.bias)

    def forward(self, x: "OrderedDict[str, any]"):
        names = list(x.keys())
        feats = list(x.values())
        last_inner = self.inner_blocks[-1](feats[-1])
        results = [self.layer_blocks[-1](last_inner)]
        for idx in range(len(feats) - 2, -1, -1):
            inner = self.inner_blocks[idx](feats[idx])
            last_inner = inner + F.interpolate(
                last_inner, size=inner.shape[-2:], mode="nearest")
            results.insert(0, self.layer_blocks[idx](last_inner))
        if self.extra_blocks == "maxpool":
            names = names + ["pool"]
            results.append(F.max_pool2d(results[-1], 1, 2, 0))
        elif self.extra_blocks == "p6p7":
            p6 = self.p6(results[-1])
            p7 = self.p7(F.relu(p6))
            names = names + ["p6", "p7"]
            results.extend([p6, p7])
        return OrderedDict(zip(names, results))


class BackboneWithFPN(nn.Module):
    def __init__(self, body, fpn, out_channels=256):
        super().__init__()
        self.body = body
        self.fpn = fpn
        self.out_channels = out_channels

    def forward(self, x):
        return self.fpn(self.body(x))


def resnet_fpn_backbone(layers=(3, 4, 6, 3), trainable_layers=3,
                        returned_layers=(1, 2, 3, 4), extra_blocks="maxpool",
                        norm_layer=FrozenBatchNorm2d):
    """ResNet-50 style backbone with frozen BN + FPN
    (ref resnet50_fpn.py:195)."""
    backbone = ResNet(Bottleneck, list(layers), include_top=False,
                      norm_layer=norm_layer)
    # freeze all but the last `trainable_layers` stages
    layers_to_train = ["layer4", "layer3", "layer2", "layer1",
                       "conv1"][:trainable_layers]
    for name, p in backbone.named_parameters():
        if not any(name.startswith(l) for l in layers_to_train):
            p.requires_grad_(False)
    return_layers = {f"layer{k}": str(i)
                     for i, k in enumerate(returned_layers)}
    in_channels = [64 * 2 ** (k - 1) * Bottleneck.expansion
                   for k in returned_layers]
    body = IntermediateLayerGetter(backbone, return_layers)
    fpn = FeaturePyramidNetwork(in_channels, 256, extra_blocks=extra_blocks)
    return BackboneWithFPN(body, fpn, 256)
