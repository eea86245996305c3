"""Swin Transformer v1 & v2 on the framework's fused ops.

MI355X design notes:
- shifted-window roll+partition / merge+roll run as ONE fused HIP gather kernel
  each way (ops.window -> csrc/window.hip) instead of roll + 6-d permute chains;
- LayerNorm / GELU are the framework's HIP kernels;
- window attention stays as hipBLASLt batched GEMMs over [B*nW, 49, C] tiles
  (49-token windows are too small for a standalone flash kernel to win).

Reference parity (structure/behavior only, re-designed):
  classification/swin_transformer/models/swin_transformer.py
    (WindowAttention:70, SwinTransformerBlock:168, PatchMerging:308,
     BasicLayer:357, SwinTransformer:423)
  classification/swin_transformer/models/swin_transformer_v2.py (cosine attention)
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import GELU, DropPath, LayerNorm
from ...ops.window import (roll_and_window_partition, window_merge_and_roll,
                           window_partition_eager)
from ..registry import register_model
from .vit import Mlp


def _rel_pos_index(window_size: int) -> torch.Tensor:
    coords = torch.stack(torch.meshgrid(
        torch.arange(window_size), torch.arange(window_size), indexing="ij"))
    coords_flat = torch.flatten(coords, 1)  # 2, ws*ws
    rel = coords_flat[:, :, None] - coords_flat[:, None, :]  # 2, N, N
    rel = rel.permute(1, 2, 0).contiguous()
    rel[:, :, 0] += window_size - 1
    rel[:, :, 1] += window_size - 1
    rel[:, :, 0] *= 2 * window_size - 1
    return rel.sum(-1)  # N, N


class WindowAttention(nn.Module):
    """W-MSA with relative position bias (v1) or log-CPB cosine attention (v2)."""

    def __init__(self, dim, window_size, num_heads, qkv_bias=True, attn_drop=0.0,
                 proj_drop=0.0, v2=False, pretrained_window_size=0):
        super().__init__()
        self.dim = dim
        self.window_size = window_size
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.v2 = v2

        if v2:
            self.logit_scale = nn.Parameter(
                torch.log(10 * torch.ones(num_heads, 1, 1)))
            # continuous relative position bias MLP (log-spaced coords)
            self.cpb_mlp = nn.Sequential(
                nn.Linear(2, 512, bias=True), nn.ReLU(inplace=True),
                nn.Linear(512, num_heads, bias=False))
            coords = torch.arange(-(window_size - 1), window_size, dtype=torch.float32)
            table = torch.stack(torch.meshgrid(coords, coords, indexing="ij"))
            table = table.permute(1, 2, 0).contiguous().unsqueeze(0)  # 1,2w-1,2w-1,2
            denom = pretrained_window_size - 1 if pretrained_window_size > 0 else window_size - 1
            table = table / max(denom, 1) * 8
            table = torch.sign(table) * torch.log2(table.abs() + 1.0) / 3.0  # log2(8)
            self.register_buffer("rel_coords_table", table, persistent=False)
            self.qkv = nn.Linear(dim, dim * 3, bias=False)
            if qkv_bias:
                self.q_bias = nn.Parameter(torch.zeros(dim))
                self.v_bias = nn.Parameter(torch.zeros(dim))
            else:
                self.q_bias = self.v_bias = None
        else:
            self.scale = self.head_dim ** -0.5
            self.relative_position_bias_table = nn.Parameter(
                torch.zeros((2 * window_size - 1) ** 2, num_heads))
            nn.init.trunc_normal_(self.relative_position_bias_table, std=0.02)
            self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias)

        self.register_buffer("relative_position_index",
                             _rel_pos_index(window_size), persistent=False)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Linear(dim, dim)
        self.proj_drop = nn.Dropout(proj_drop)
        self.softmax = nn.Softmax(dim=-1)

    def _bias(self):
        N = self.window_size * self.window_size
        if self.v2:
            table = self.cpb_mlp(self.rel_coords_table).view(-1, self.num_heads)
            bias = table[self.relative_position_index.view(-1)].view(N, N, -1)
            bias = 16 * torch.sigmoid(bias)
        else:
            bias = self.relative_position_bias_table[
                self.relative_position_index.view(-1)].view(N, N, -1)
        return bias.permute(2, 0, 1).contiguous().unsqueeze(0)  # 1, nH, N, N

    def forward(self, x, mask=None):
        from ...ops.attention import fused_attention
        B_, N, C = x.shape
        if self.v2 and (self.attn_drop.p == 0.0 or not self.training):
            # fused cosine path, train + eval (in-kernel q/k norms +
            # per-head logit scale; backward from kernel-saved P with
            # grads for logit_scale and the CPB bias)
            from ...ops.attention import fused_attention_cosine
            qkv_bias = None
            if self.q_bias is not None:
                qkv_bias = torch.cat((
                    self.q_bias, torch.zeros_like(self.v_bias), self.v_bias))
            qkv = torch.nn.functional.linear(x, self.qkv.weight, qkv_bias)
            lscale = torch.clamp(
                self.logit_scale,
                max=torch.log(torch.tensor(100.0,
                                           device=x.device))).exp().flatten()
            out = fused_attention_cosine(qkv, self.num_heads, lscale,
                                         bias=self._bias().squeeze(0),
                                         mask=mask)
            return self.proj_drop(self.proj(out))
        if not self.v2 and (self.attn_drop.p == 0.0 or not self.training):
            # one fused HIP MFMA kernel: scale*QK^T + rel-pos bias (+window
            # mask) + softmax + V, straight from the packed qkv projection
            qkv = self.qkv(x)
            out = fused_attention(qkv, self.num_heads, self.scale,
                                  bias=self._bias().squeeze(0), mask=mask)
            return self.proj_drop(self.proj(out))
        if self.v2:
            qkv_bias = None
            if self.q_bias is not None:
                qkv_bias = torch.cat((
                    self.q_bias,
                    torch.zeros_like(self.v_bias, requires_grad=False),
                    self.v_bias))
            qkv = torch.nn.functional.linear(x, self.qkv.weight, qkv_bias)
        else:
            qkv = self.qkv(x)
        qkv = qkv.reshape(B_, N, 3, self.num_heads, self.head_dim).permute(2, 0, 3, 1, 4)
        q, k, v = qkv.unbind(0)

        if self.v2:
            attn = torch.nn.functional.normalize(q, dim=-1) @ \
                torch.nn.functional.normalize(k, dim=-1).transpose(-2, -1)
            logit_scale = torch.clamp(
                self.logit_scale,
                max=torch.log(torch.tensor(100.0, device=x.device))).exp()
            attn = attn * logit_scale
        else:
            attn = (q * self.scale) @ k.transpose(-2, -1)
        attn = attn + self._bias()

        if mask is not None:
            nW = mask.shape[0]
            attn = attn.view(B_ // nW, nW, self.num_heads, N, N) + \
                mask.unsqueeze(1).unsqueeze(0)
            attn = attn.view(-1, self.num_heads, N, N)
        attn = self.attn_drop(self.softmax(attn))
        x = (attn @ v).transpose(1, 2).reshape(B_, N, C)
        return self.proj_drop(self.proj(x))


class SwinTransformerBlock(nn.Module):
    def __init__(self, dim, input_resolution, num_heads, window_size=7, shift_size=0,
                 mlp_ratio=4.0, qkv_bias=True, drop=0.0, attn_drop=0.0, drop_path=0.0,
                 norm_layer=LayerNorm, v2=False, pretrained_window_size=0):
        super().__init__()
        self.dim = dim
        self.input_resolution = input_resolution
        self.num_heads = num_heads
        self.window_size = window_size
        self.shift_size = shift_size
        self.mlp_ratio = mlp_ratio
        self.v2 = v2
        if min(input_resolution) <= window_size:
            self.shift_size = 0
            self.window_size = min(input_resolution)

        self.norm1 = norm_layer(dim)
        self.attn = WindowAttention(dim, self.window_size, num_heads, qkv_bias,
                                    attn_drop, drop, v2=v2,
                                    pretrained_window_size=pretrained_window_size)
        self.drop_path = DropPath(drop_path) if drop_path > 0 else nn.Identity()
        self.norm2 = norm_layer(dim)
        self.mlp = Mlp(dim, int(dim * mlp_ratio), drop=drop)

        if self.shift_size > 0:
            H, W = input_resolution
            img_mask = torch.zeros((1, H, W, 1))
            slices = (slice(0, -self.window_size),
                      slice(-self.window_size, -self.shift_size),
                      slice(-self.shift_size, None))
            cnt = 0
            for h in slices:
                for w in slices:
                    img_mask[:, h, w, :] = cnt
                    cnt += 1
            mask_windows = window_partition_eager(img_mask, self.window_size)
            mask_windows = mask_windows.view(-1, self.window_size * self.window_size)
            attn_mask = mask_windows.unsqueeze(1) - mask_windows.unsqueeze(2)
            attn_mask = attn_mask.masked_fill(attn_mask != 0, -100.0)
        else:
            attn_mask = None
        self.register_buffer("attn_mask", attn_mask, persistent=False)

    def forward(self, x):
        H, W = self.input_resolution
        B, L, C = x.shape
        shortcut = x
        if not self.v2:
            x = self.norm1(x)
        x = x.view(B, H, W, C)

        # fused HIP roll(-s)+partition  -> [B*nW, ws, ws, C]
        ws = self.window_size
        x_windows = roll_and_window_partition(x, ws, self.shift_size)
        x_windows = x_windows.view(-1, ws * ws, C)
        attn_windows = self.attn(
            x_windows, mask=self.attn_mask.to(x.dtype) if self.attn_mask is not None else None)
        attn_windows = attn_windows.view(-1, ws, ws, C)
        # fused HIP merge+roll(+s)  -> [B, H, W, C]
        x = window_merge_and_roll(attn_windows, B, H, W, ws, self.shift_size)
        x = x.view(B, H * W, C)

        if self.v2:  # post-norm residual (v2)
            x = shortcut + self.drop_path(self.norm1(x))
            x = x + self.drop_path(self.norm2(self.mlp(x)))
        else:
            x = shortcut + self.drop_path(x)
            x = x + self.drop_path(self.mlp(self.norm2(x)))
        return x


class PatchMerging(nn.Module):
    """2x2 spatial concat -> linear 4C->2C (ref swin_transformer.py:308)."""

    def __init__(self, input_resolution, dim, norm_layer=LayerNorm, v2=False):
        super().__init__()
        self.input_resolution = input_resolution
        self.dim = dim
        self.v2 = v2
        self.reduction = nn.Linear(4 * dim, 2 * dim, bias=False)
        self.norm = norm_layer(2 * dim if v2 else 4 * dim)

    def forward(self, x):
        H, W = self.input_resolution
        B, L, C = x.shape
        x = x.view(B, H, W, C)
        x = torch.cat([x[:, 0::2, 0::2], x[:, 1::2, 0::2],
                       x[:, 0::2, 1::2], x[:, 1::2, 1::2]], -1)
        x = x.view(B, -1, 4 * C)
        if self.v2:
            return self.norm(self.reduction(x))
        return self.reduction(self.norm(x))


class BasicLayer(nn.Module):
    def __init__(self, dim, input_resolution, depth, num_heads, window_size,
                 mlp_ratio=4.0, qkv_bias=True, drop=0.0, attn_drop=0.0,
                 drop_path=0.0, norm_layer=LayerNorm, downsample=None, v2=False,
                 pretrained_window_size=0):
        super().__init__()
        self.blocks = nn.ModuleList([
            SwinTransformerBlock(
                dim, input_resolution, num_heads, window_size,
                0 if i % 2 == 0 else window_size // 2, mlp_ratio, qkv_bias,
                drop, attn_drop,
                drop_path[i] if isinstance(drop_path, (list, tuple)) else drop_path,
                norm_layer, v2=v2, pretrained_window_size=pretrained_window_size)
            for i in range(depth)])
        self.downsample = downsample(input_resolution, dim, norm_layer, v2=v2) \
            if downsample is not None else None

    def forward(self, x):
        for blk in self.blocks:
            x = blk(x)
        if self.downsample is not None:
            x = self.downsample(x)
        return x


class SwinTransformer(nn.Module):
    def __init__(self, img_size=224, patch_size=4, in_chans=3, num_classes=1000,
                 embed_dim=96, depths=(2, 2, 6, 2), num_heads=(3, 6, 12, 24),
                 window_size=7, mlp_ratio=4.0, qkv_bias=True, drop_rate=0.0,
                 attn_drop_rate=0.0, drop_path_rate=0.1, norm_layer=LayerNorm,
                 ape=False, patch_norm=True, v2=False,
                 pretrained_window_sizes=(0, 0, 0, 0)):
        super().__init__()
        self.num_classes = num_classes
        self.num_layers = len(depths)
        self.embed_dim = embed_dim
        self.num_features = int(embed_dim * 2 ** (self.num_layers - 1))

        self.patch_embed = nn.Conv2d(in_chans, embed_dim, patch_size, patch_size)
        patches_resolution = (img_size // patch_size, img_size // patch_size)
        self.patches_resolution = patches_resolution
        num_patches = patches_resolution[0] * patches_resolution[1]
        self.patch_norm = norm_layer(embed_dim) if patch_norm else nn.Identity()

        self.ape = ape
        if ape:
            self.absolute_pos_embed = nn.Parameter(
                torch.zeros(1, num_patches, embed_dim))
            nn.init.trunc_normal_(self.absolute_pos_embed, std=0.02)
        self.pos_drop = nn.Dropout(drop_rate)

        dpr = [x.item() for x in torch.linspace(0, drop_path_rate, sum(depths))]
        self.layers = nn.ModuleList()
        for i in range(self.num_layers):
            layer = BasicLayer(
                dim=int(embed_dim * 2 ** i),
                input_resolution=(patches_resolution[0] // (2 ** i),
                                  patches_resolution[1] // (2 ** i)),
                depth=depths[i], num_heads=num_heads[i], window_size=window_size,
                mlp_ratio=mlp_ratio, qkv_bias=qkv_bias, drop=drop_rate,
                attn_drop=attn_drop_rate,
                drop_path=dpr[sum(depths[:i]):sum(depths[:i + 1])],
                norm_layer=norm_layer,
                downsample=PatchMerging if i < self.num_layers - 1 else None,
                v2=v2, pretrained_window_size=pretrained_window_sizes[i])
            self.layers.append(layer)

        self.norm = norm_layer(self.num_features)
        self.avgpool = nn.AdaptiveAvgPool1d(1)
        self.head = nn.Linear(self.num_features, num_classes) \
            if num_classes > 0 else nn.Identity()
        self.apply(self._init_weights)

    @staticmethod
    def _init_weights(m):
        if isinstance(m, nn.Linear):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)

    @torch.jit.ignore
    def no_weight_decay(self):
        # ref swin_transformer.py no_weight_decay{_keywords}
        return {"absolute_pos_embed"}

    @torch.jit.ignore
    def no_weight_decay_keywords(self):
        return {"relative_position_bias_table", "logit_scale", "cpb_mlp"}

    def forward_features(self, x):
        from .vit import patch_embed_gemm
        x = patch_embed_gemm(x, self.patch_embed.weight,
                             self.patch_embed.bias,
                             self.patch_embed.kernel_size[0])  # B, L, C
        x = self.patch_norm(x)
        if self.ape:
            x = x + self.absolute_pos_embed
        x = self.pos_drop(x)
        for layer in self.layers:
            x = layer(x)
        x = self.norm(x)
        x = self.avgpool(x.transpose(1, 2)).flatten(1)
        return x

    def forward(self, x):
        return self.head(self.forward_features(x))


@register_model
def swin_t(num_classes=1000, **kw):
    return SwinTransformer(embed_dim=96, depths=(2, 2, 6, 2),
                           num_heads=(3, 6, 12, 24), num_classes=num_classes, **kw)


@register_model
def swin_s(num_classes=1000, **kw):
    return SwinTransformer(embed_dim=96, depths=(2, 2, 18, 2),
                           num_heads=(3, 6, 12, 24), num_classes=num_classes, **kw)


@register_model
def swin_b(num_classes=1000, **kw):
    return SwinTransformer(embed_dim=128, depths=(2, 2, 18, 2),
                           num_heads=(4, 8, 16, 32), num_classes=num_classes, **kw)


@register_model
def swin_l(num_classes=1000, **kw):
    return SwinTransformer(embed_dim=192, depths=(2, 2, 18, 2),
                           num_heads=(6, 12, 24, 48), num_classes=num_classes, **kw)


@register_model
def swinv2_t(num_classes=1000, **kw):
    return SwinTransformer(embed_dim=96, depths=(2, 2, 6, 2),
                           num_heads=(3, 6, 12, 24), v2=True,
                           window_size=kw.pop("window_size", 8),
                           img_size=kw.pop("img_size", 256),
                           num_classes=num_classes, **kw)


@register_model
def swinv2_s(num_classes=1000, **kw):
    return SwinTransformer(embed_dim=96, depths=(2, 2, 18, 2),
                           num_heads=(3, 6, 12, 24), v2=True,
                           window_size=kw.pop("window_size", 8),
                           img_size=kw.pop("img_size", 256),
                           num_classes=num_classes, **kw)


@register_model
def swinv2_b(num_classes=1000, **kw):
    return SwinTransformer(embed_dim=128, depths=(2, 2, 18, 2),
                           num_heads=(4, 8, 16, 32), v2=True,
                           window_size=kw.pop("window_size", 8),
                           img_size=kw.pop("img_size", 256),
                           num_classes=num_classes, **kw)
