"""CoAtNet: conv (MBConv) + transformer hybrid with relative attention.

Reference parity: classification/coatNet/models/networks.py (346 LoC) —
re-designed on the framework's HIP ops (BN, GELU, LayerNorm).
Stages: S0 conv stem, S1-S2 MBConv, S3-S4 relative-attention transformer.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import GELU, BatchNorm2d, LayerNorm
from ..registry import register_model
from .vit import Mlp


class MBConvBlock(nn.Module):
    def __init__(self, cin, cout, stride=1, expand=4):
        super().__init__()
        self.use_res = stride == 1 and cin == cout
        mid = cin * expand
        self.pre_norm = BatchNorm2d(cin)
        self.conv = nn.Sequential(
            nn.Conv2d(cin, mid, 1, bias=False), BatchNorm2d(mid, relu=True),
            nn.Conv2d(mid, mid, 3, stride, 1, groups=mid, bias=False),
            BatchNorm2d(mid, relu=True),
            nn.Conv2d(mid, cout, 1, bias=False), BatchNorm2d(cout))
        self.proj = None
        if not self.use_res:
            self.proj = nn.Sequential(
                nn.MaxPool2d(3, stride, 1) if stride > 1 else nn.Identity(),
                nn.Conv2d(cin, cout, 1))

    def forward(self, x):
        out = self.conv(self.pre_norm(x))
        shortcut = x if self.proj is None else self.proj(x)
        return shortcut + out


class RelAttention(nn.Module):
    """MHSA with learned relative position bias over an HxW grid."""

    def __init__(self, dim, heads, grid):
        super().__init__()
        self.heads = heads
        self.head_dim = dim // heads
        self.scale = self.head_dim ** -0.5
        self.qkv = nn.Linear(dim, dim * 3, bias=True)
        self.proj = nn.Linear(dim, dim)
        H, W = grid
        self.rel_bias = nn.Parameter(torch.zeros((2 * H - 1) * (2 * W - 1), heads))
        coords = torch.stack(torch.meshgrid(
            torch.arange(H), torch.arange(W), indexing="ij")).flatten(1)
        rel = coords[:, :, None] - coords[:, None, :]
        rel = rel.permute(1, 2, 0).contiguous()
        rel[:, :, 0] += H - 1
        rel[:, :, 1] += W - 1
        rel[:, :, 0] *= 2 * W - 1
        self.register_buffer("rel_index", rel.sum(-1), persistent=False)
        nn.init.trunc_normal_(self.rel_bias, std=0.02)

    def forward(self, x):
        B, N, C = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.heads, self.head_dim)
        q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)
        attn = (q @ k.transpose(-2, -1)) * self.scale
        bias = self.rel_bias[self.rel_index.view(-1)].view(N, N, -1)
        attn = attn + bias.permute(2, 0, 1).unsqueeze(0)
        attn = attn.softmax(dim=-1)
        return self.proj((attn @ v).transpose(1, 2).reshape(B, N, C))


class TransformerBlock(nn.Module):
    def __init__(self, cin, cout, grid, heads=8, downsample=False):
        super().__init__()
        self.downsample = downsample
        if downsample:
            self.pool = nn.MaxPool2d(2)
            self.proj = nn.Conv2d(cin, cout, 1)
        self.norm1 = LayerNorm(cout)
        self.attn = RelAttention(cout, heads, grid)
        self.norm2 = LayerNorm(cout)
        self.mlp = Mlp(cout, cout * 4, act_layer=GELU)

    def forward(self, x):
        # x: B,C,H,W
        if self.downsample:
            x = self.proj(self.pool(x))
        B, C, H, W = x.shape
        t = x.flatten(2).transpose(1, 2)  # B,N,C
        t = t + self.attn(self.norm1(t))
        t = t + self.mlp(self.norm2(t))
        return t.transpose(1, 2).reshape(B, C, H, W)


class CoAtNet(nn.Module):
    def __init__(self, img_size=224, num_blocks=(2, 2, 3, 5, 2),
                 channels=(64, 96, 192, 384, 768), num_classes=1000,
                 heads=(8, 16)):
        super().__init__()
        s = img_size
        self.s0 = nn.Sequential(*(
            [nn.Conv2d(3, channels[0], 3, 2, 1, bias=False),
             BatchNorm2d(channels[0], relu=True)] +
            [m for _ in range(num_blocks[0] - 1)
             for m in (nn.Conv2d(channels[0], channels[0], 3, 1, 1, bias=False),
                       BatchNorm2d(channels[0], relu=True))]))
        s //= 2
        self.s1 = nn.Sequential(*[
            MBConvBlock(channels[0] if i == 0 else channels[1], channels[1],
                        stride=2 if i == 0 else 1)
            for i in range(num_blocks[1])])
        s //= 2
        self.s2 = nn.Sequential(*[
            MBConvBlock(channels[1] if i == 0 else channels[2], channels[2],
                        stride=2 if i == 0 else 1)
            for i in range(num_blocks[2])])
        s //= 2
        s3_grid = (s // 2, s // 2)
        self.s3 = nn.Sequential(*[
            TransformerBlock(channels[2] if i == 0 else channels[3], channels[3],
                             s3_grid, heads[0], downsample=(i == 0))
            for i in range(num_blocks[3])])
        s //= 2
        s4_grid = (s // 2, s // 2)
        self.s4 = nn.Sequential(*[
            TransformerBlock(channels[3] if i == 0 else channels[4], channels[4],
                             s4_grid, heads[1], downsample=(i == 0))
            for i in range(num_blocks[4])])
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(channels[4], num_classes)

    def forward(self, x):
        x = self.s4(self.s3(self.s2(self.s1(self.s0(x)))))
        return self.fc(self.pool(x).flatten(1))


@register_model
def coatnet_0(num_classes=1000, **kw):
    return CoAtNet(num_blocks=(2, 2, 3, 5, 2), channels=(64, 96, 192, 384, 768),
                   num_classes=num_classes, **kw)


@register_model
def coatnet_1(num_classes=1000, **kw):
    return CoAtNet(num_blocks=(2, 2, 6, 14, 2),
                   channels=(64, 96, 192, 384, 768), num_classes=num_classes, **kw)
