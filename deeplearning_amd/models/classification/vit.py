"""Vision Transformer (ViT-B/L/H) on the framework's fused ops: HIP LayerNorm,
HIP GELU, DropPath; attention is explicit QKV GEMM + softmax(QK^T)V (hipBLASLt
GEMMs) with a hand-written fused-attention HIP kernel slot (ops.attention) to
swap in.

Reference parity: classification/vision_transformer/vit_model.py (PatchEmbed:43,
Attention:71, Block:136, VisionTransformer:164, factories :290-358).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import GELU, DropPath, LayerNorm
from ..registry import register_model


def patch_embed_gemm(x: torch.Tensor, weight: torch.Tensor,
                     bias: torch.Tensor | None, patch: int) -> torch.Tensor:
    """Non-overlapping patch embed (conv k=s=p) as reshape + one hipBLASLt
    GEMM. MIOpen routes this conv shape to its naive fallback + im2col on
    gfx950 (rocprof: 35% of ViT step time); the GEMM form runs on MFMA.
    weight: [D, C, p, p] conv weight. Returns [B, N, D]."""
    B, C, H, W = x.shape
    x = x.reshape(B, C, H // patch, patch, W // patch, patch)
    x = x.permute(0, 2, 4, 1, 3, 5).reshape(B, -1, C * patch * patch)
    return torch.nn.functional.linear(x, weight.reshape(weight.shape[0], -1),
                                      bias)


class PatchEmbed(nn.Module):
    def __init__(self, img_size=224, patch_size=16, in_chans=3, embed_dim=768,
                 norm_layer=None):
        super().__init__()
        self.img_size = (img_size, img_size)
        self.patch_size = (patch_size, patch_size)
        self.grid_size = (img_size // patch_size, img_size // patch_size)
        self.num_patches = self.grid_size[0] * self.grid_size[1]
        self.proj = nn.Conv2d(in_chans, embed_dim, kernel_size=patch_size,
                              stride=patch_size)
        self.norm = norm_layer(embed_dim) if norm_layer else nn.Identity()

    def forward(self, x):
        B, C, H, W = x.shape
        assert H == self.img_size[0] and W == self.img_size[1], \
            f"input {H}x{W} doesn't match model {self.img_size}"
        x = patch_embed_gemm(x, self.proj.weight, self.proj.bias,
                             self.patch_size[0])  # B, N, C
        return self.norm(x)


class Attention(nn.Module):
    def __init__(self, dim, num_heads=8, qkv_bias=False, attn_drop=0.0, proj_drop=0.0):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.scale = self.head_dim ** -0.5
        self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Linear(dim, dim)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x):
        from ...ops.attention import fused_attention
        B, N, C = x.shape
        qkv = self.qkv(x)
        if self.attn_drop.p == 0.0 or not self.training:
            # one fused HIP MFMA kernel (csrc/attention.hip)
            out = fused_attention(qkv, self.num_heads, self.scale)
            return self.proj_drop(self.proj(out))
        qkv = qkv.reshape(B, N, 3, self.num_heads, self.head_dim)
        qkv = qkv.permute(2, 0, 3, 1, 4)  # 3, B, H, N, d
        q, k, v = qkv.unbind(0)
        attn = (q @ k.transpose(-2, -1)) * self.scale
        attn = attn.softmax(dim=-1)
        attn = self.attn_drop(attn)
        x = (attn @ v).transpose(1, 2).reshape(B, N, C)
        return self.proj_drop(self.proj(x))


class Mlp(nn.Module):
    def __init__(self, in_features, hidden_features=None, out_features=None,
                 act_layer=GELU, drop=0.0):
        super().__init__()
        out_features = out_features or in_features
        hidden_features = hidden_features or in_features
        self.fc1 = nn.Linear(in_features, hidden_features)
        self.act = act_layer()
        self.fc2 = nn.Linear(hidden_features, out_features)
        self.drop = nn.Dropout(drop)

    def forward(self, x):
        return self.drop(self.fc2(self.drop(self.act(self.fc1(x)))))


class Block(nn.Module):
    def __init__(self, dim, num_heads, mlp_ratio=4.0, qkv_bias=True, drop=0.0,
                 attn_drop=0.0, drop_path=0.0, norm_layer=LayerNorm):
        super().__init__()
        self.norm1 = norm_layer(dim)
        self.attn = Attention(dim, num_heads, qkv_bias, attn_drop, drop)
        self.drop_path = DropPath(drop_path) if drop_path > 0 else nn.Identity()
        self.norm2 = norm_layer(dim)
        self.mlp = Mlp(dim, int(dim * mlp_ratio), drop=drop)

    def forward(self, x):
        x = x + self.drop_path(self.attn(self.norm1(x)))
        x = x + self.drop_path(self.mlp(self.norm2(x)))
        return x


class VisionTransformer(nn.Module):
    def __init__(self, img_size=224, patch_size=16, in_chans=3, num_classes=1000,
                 embed_dim=768, depth=12, num_heads=12, mlp_ratio=4.0,
                 qkv_bias=True, representation_size=None, drop_rate=0.0,
                 attn_drop_rate=0.0, drop_path_rate=0.0, norm_layer=LayerNorm):
        super().__init__()
        self.num_classes = num_classes
        self.num_features = self.embed_dim = embed_dim
        self.patch_embed = PatchEmbed(img_size, patch_size, in_chans, embed_dim)
        num_patches = self.patch_embed.num_patches

        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.pos_embed = nn.Parameter(torch.zeros(1, num_patches + 1, embed_dim))
        self.pos_drop = nn.Dropout(drop_rate)

        dpr = [x.item() for x in torch.linspace(0, drop_path_rate, depth)]
        self.blocks = nn.Sequential(*[
            Block(embed_dim, num_heads, mlp_ratio, qkv_bias, drop_rate,
                  attn_drop_rate, dpr[i], norm_layer) for i in range(depth)
        ])
        self.norm = norm_layer(embed_dim)

        if representation_size:
            self.pre_logits = nn.Sequential(
                nn.Linear(embed_dim, representation_size), nn.Tanh())
            head_in = representation_size
        else:
            self.pre_logits = nn.Identity()
            head_in = embed_dim
        self.head = nn.Linear(head_in, num_classes) if num_classes > 0 else nn.Identity()

        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)
        self.apply(self._init_weights)

    @staticmethod
    def _init_weights(m):
        if isinstance(m, nn.Linear):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)

    @torch.jit.ignore
    def no_weight_decay(self):
        return {"pos_embed", "cls_token", "dist_token"}

    def forward_features(self, x):
        x = self.patch_embed(x)
        cls = self.cls_token.expand(x.shape[0], -1, -1)
        x = torch.cat((cls, x), dim=1)
        x = self.pos_drop(x + self.pos_embed)
        x = self.blocks(x)
        x = self.norm(x)
        return self.pre_logits(x[:, 0])

    def forward(self, x):
        return self.head(self.forward_features(x))


@register_model
def vit_b16(num_classes=1000, **kw):
    return VisionTransformer(patch_size=16, embed_dim=768, depth=12, num_heads=12,
                             num_classes=num_classes, **kw)


@register_model
def vit_b32(num_classes=1000, **kw):
    return VisionTransformer(patch_size=32, embed_dim=768, depth=12, num_heads=12,
                             num_classes=num_classes, **kw)


@register_model
def vit_l16(num_classes=1000, **kw):
    return VisionTransformer(patch_size=16, embed_dim=1024, depth=24, num_heads=16,
                             num_classes=num_classes, **kw)


@register_model
def vit_h14(num_classes=1000, **kw):
    return VisionTransformer(patch_size=14, embed_dim=1280, depth=32, num_heads=16,
                             num_classes=num_classes, **kw)
