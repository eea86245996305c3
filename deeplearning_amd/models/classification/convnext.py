"""ConvNeXt-T/S/B/L with layer scale + stochastic depth.

Reference parity: classification/convNext/models/networks.py (233 LoC) —
re-designed: channels-first LayerNorm runs the framework's HIP LayerNorm2d
kernel; the block's LN+GELU run HIP kernels.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import DropPath, LayerNorm, LayerNorm2d, gelu
from ..registry import register_model


class ConvNeXtBlock(nn.Module):
    """dwconv7x7 -> LN -> pwconv(4x) -> GELU -> pwconv -> layer-scale -> droppath."""

    def __init__(self, dim, drop_path=0.0, layer_scale_init=1e-6):
        super().__init__()
        self.dwconv = nn.Conv2d(dim, dim, 7, padding=3, groups=dim)
        self.norm = LayerNorm(dim, eps=1e-6)
        self.pwconv1 = nn.Linear(dim, 4 * dim)
        self.pwconv2 = nn.Linear(4 * dim, dim)
        self.gamma = nn.Parameter(layer_scale_init * torch.ones(dim)) \
            if layer_scale_init > 0 else None
        self.drop_path = DropPath(drop_path) if drop_path > 0 else nn.Identity()

    def forward(self, x):
        shortcut = x
        x = self.dwconv(x)
        x = x.permute(0, 2, 3, 1)  # NHWC
        x = self.pwconv2(gelu(self.pwconv1(self.norm(x))))
        if self.gamma is not None:
            x = self.gamma * x
        x = x.permute(0, 3, 1, 2)
        return shortcut + self.drop_path(x)


class ConvNeXt(nn.Module):
    def __init__(self, in_chans=3, num_classes=1000, depths=(3, 3, 9, 3),
                 dims=(96, 192, 384, 768), drop_path_rate=0.0,
                 layer_scale_init=1e-6, head_init_scale=1.0):
        super().__init__()
        self.downsample_layers = nn.ModuleList()
        stem = nn.Sequential(
            nn.Conv2d(in_chans, dims[0], 4, stride=4),
            LayerNorm2d(dims[0], eps=1e-6))
        self.downsample_layers.append(stem)
        for i in range(3):
            self.downsample_layers.append(nn.Sequential(
                LayerNorm2d(dims[i], eps=1e-6),
                nn.Conv2d(dims[i], dims[i + 1], 2, stride=2)))

        self.stages = nn.ModuleList()
        dpr = [x.item() for x in torch.linspace(0, drop_path_rate, sum(depths))]
        cur = 0
        for i in range(4):
            self.stages.append(nn.Sequential(*[
                ConvNeXtBlock(dims[i], dpr[cur + j], layer_scale_init)
                for j in range(depths[i])]))
            cur += depths[i]

        self.norm = LayerNorm(dims[-1], eps=1e-6)
        self.head = nn.Linear(dims[-1], num_classes)
        self.apply(self._init_weights)
        with torch.no_grad():
            self.head.weight.mul_(head_init_scale)
            self.head.bias.mul_(head_init_scale)

    @staticmethod
    def _init_weights(m):
        if isinstance(m, (nn.Conv2d, nn.Linear)):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward_features(self, x):
        for down, stage in zip(self.downsample_layers, self.stages):
            x = stage(down(x))
        return self.norm(x.mean([-2, -1]))  # global avg pool then LN

    def forward(self, x):
        return self.head(self.forward_features(x))


@register_model
def convnext_tiny(num_classes=1000, **kw):
    return ConvNeXt(depths=(3, 3, 9, 3), dims=(96, 192, 384, 768),
                    num_classes=num_classes, **kw)


@register_model
def convnext_small(num_classes=1000, **kw):
    return ConvNeXt(depths=(3, 3, 27, 3), dims=(96, 192, 384, 768),
                    num_classes=num_classes, **kw)


@register_model
def convnext_base(num_classes=1000, **kw):
    return ConvNeXt(depths=(3, 3, 27, 3), dims=(128, 256, 512, 1024),
                    num_classes=num_classes, **kw)


@register_model
def convnext_large(num_classes=1000, **kw):
    return ConvNeXt(depths=(3, 3, 27, 3), dims=(192, 384, 768, 1536),
                    num_classes=num_classes, **kw)
