"""U-Net encoder/decoder segmenter.

Reference parity: Image_segmentation/U-Net/models/networks.py (121 LoC) —
re-designed: DoubleConv uses the framework's fused HIP BN+ReLU.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import BatchNorm2d
from ..registry import register_model


class DoubleConv(nn.Sequential):
    def __init__(self, cin, cout, mid=None):
        mid = mid or cout
        super().__init__(
            nn.Conv2d(cin, mid, 3, padding=1, bias=False),
            BatchNorm2d(mid, relu=True),
            nn.Conv2d(mid, cout, 3, padding=1, bias=False),
            BatchNorm2d(cout, relu=True))


class Down(nn.Sequential):
    def __init__(self, cin, cout):
        super().__init__(nn.MaxPool2d(2), DoubleConv(cin, cout))


class Up(nn.Module):
    def __init__(self, cin, cout, bilinear=True):
        super().__init__()
        if bilinear:
            self.up = nn.Upsample(scale_factor=2, mode="bilinear",
                                  align_corners=True)
            self.conv = DoubleConv(cin, cout, cin // 2)
        else:
            self.up = nn.ConvTranspose2d(cin, cin // 2, 2, stride=2)
            self.conv = DoubleConv(cin, cout)

    def forward(self, x1, x2):
        x1 = self.up(x1)
        dy = x2.size(2) - x1.size(2)
        dx = x2.size(3) - x1.size(3)
        x1 = F.pad(x1, [dx // 2, dx - dx // 2, dy // 2, dy - dy // 2])
        return self.conv(torch.cat([x2, x1], dim=1))


class UNet(nn.Module):
    def __init__(self, in_channels=3, num_classes=2, bilinear=True,
                 base_c=64):
        super().__init__()
        self.in_conv = DoubleConv(in_channels, base_c)
        self.down1 = Down(base_c, base_c * 2)
        self.down2 = Down(base_c * 2, base_c * 4)
        self.down3 = Down(base_c * 4, base_c * 8)
        factor = 2 if bilinear else 1
        self.down4 = Down(base_c * 8, base_c * 16 // factor)
        self.up1 = Up(base_c * 16, base_c * 8 // factor, bilinear)
        self.up2 = Up(base_c * 8, base_c * 4 // factor, bilinear)
        self.up3 = Up(base_c * 4, base_c * 2 // factor, bilinear)
        self.up4 = Up(base_c * 2, base_c, bilinear)
        self.out_conv = nn.Conv2d(base_c, num_classes, 1)

    def forward(self, x):
        x1 = self.in_conv(x)
        x2 = self.down1(x1)
        x3 = self.down2(x2)
        x4 = self.down3(x3)
        x5 = self.down4(x4)
        x = self.up1(x5, x4)
        x = self.up2(x, x3)
        x = self.up3(x, x2)
        x = self.up4(x, x1)
        return {"out": self.out_conv(x)}


@register_model
def unet(num_classes=2, **kw):
    return UNet(num_classes=num_classes, **kw)
