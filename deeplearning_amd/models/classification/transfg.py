"""TransFG: fine-grained ViT with Part-Attention selection + contrastive loss.

Reference parity: classification/TransFG/models/transfg.py (Part_Attention
:131-143) and train.py contrastive objective — re-designed on this repo's ViT
blocks (HIP LayerNorm/GELU). Part attention multiplies attention maps across
layers and routes the max-attended token per head into a final part layer.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import LayerNorm
from ..registry import register_model
from .vit import Block, PatchEmbed


class PartAttention(nn.Module):
    """Multiply per-layer attention maps, pick max-attended token per head."""

    def forward(self, attn_list):
        joint = attn_list[0]
        for a in attn_list[1:]:
            joint = a @ joint
        # joint: B, heads, N, N ; attention of CLS (row 0) to patches
        _, max_idx = joint[:, :, 0, 1:].max(2)
        return max_idx + 1  # offset past CLS


class AttnBlock(Block):
    """ViT block that also returns its attention map."""

    def forward(self, x):
        B, N, C = x.shape
        h = self.norm1(x)
        qkv = self.attn.qkv(h).reshape(B, N, 3, self.attn.num_heads,
                                       self.attn.head_dim).permute(2, 0, 3, 1, 4)
        q, k, v = qkv.unbind(0)
        attn = (q @ k.transpose(-2, -1)) * self.attn.scale
        attn = attn.softmax(dim=-1)
        out = (attn @ v).transpose(1, 2).reshape(B, N, C)
        out = self.attn.proj_drop(self.attn.proj(out))
        x = x + self.drop_path(out)
        x = x + self.drop_path(self.mlp(self.norm2(x)))
        return x, attn


class TransFG(nn.Module):
    def __init__(self, img_size=448, patch_size=16, num_classes=200,
                 embed_dim=768, depth=12, num_heads=12, mlp_ratio=4.0,
                 qkv_bias=True, norm_layer=LayerNorm):
        super().__init__()
        self.patch_embed = PatchEmbed(img_size, patch_size, 3, embed_dim)
        num_patches = self.patch_embed.num_patches
        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.pos_embed = nn.Parameter(torch.zeros(1, num_patches + 1, embed_dim))
        self.blocks = nn.ModuleList([
            AttnBlock(embed_dim, num_heads, mlp_ratio, qkv_bias,
                      norm_layer=norm_layer)
            for _ in range(depth - 1)])
        self.part_select = PartAttention()
        self.part_layer = Block(embed_dim, num_heads, mlp_ratio, qkv_bias,
                                norm_layer=norm_layer)
        self.part_norm = norm_layer(embed_dim)
        self.head = nn.Linear(embed_dim, num_classes)
        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)

    def forward_features(self, x):
        x = self.patch_embed(x)
        cls = self.cls_token.expand(x.shape[0], -1, -1)
        x = torch.cat((cls, x), 1) + self.pos_embed
        attns = []
        for blk in self.blocks:
            x, a = blk(x)
            attns.append(a)
        part_idx = self.part_select(attns)  # B, heads
        B, num = part_idx.shape
        parts = x[torch.arange(B).unsqueeze(1), part_idx]  # B, heads, C
        concat = torch.cat((x[:, :1], parts), dim=1)
        hidden = self.part_layer(concat)
        return self.part_norm(hidden)[:, 0]

    def forward(self, x):
        return self.head(self.forward_features(x))


def contrastive_loss(features: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """TransFG's pairwise contrastive loss over normalized CLS features
    (ref TransFG train.py con_loss)."""
    f = nn.functional.normalize(features, dim=-1)
    sim = f @ f.t()
    same = labels.unsqueeze(0) == labels.unsqueeze(1)
    pos = (1.0 - sim) * same
    neg = torch.clamp(sim - 0.4, min=0.0) * (~same)
    return (pos.sum() + neg.sum()) / (f.shape[0] ** 2)


@register_model
def transfg_b16(num_classes=200, **kw):
    return TransFG(num_classes=num_classes, **kw)
