"""Swin-MoE: Swin Transformer with mixture-of-experts MLPs in alternating
blocks of the later stages.

Reference parity: classification/swin_transformer/models/swin_transformer_moe.py
(MoEMlp with tutel moe_layer :36-94; cosine router, capacity 1.25, expert
parallelism) — re-designed on this repo's Swin + parallel/moe.MoEMlp (RCCL
all-to-all over xGMI for the EP dispatch).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...parallel.moe import MoEMlp
from ..registry import register_model
from .swin import SwinTransformer


class MoEMlpAdapter(nn.Module):
    """Drop-in replacement for the block's Mlp that records the aux loss."""

    def __init__(self, dim, hidden_dim, num_experts=8, top_k=1,
                 ep_group=None):
        super().__init__()
        self.moe = MoEMlp(dim, hidden_dim, num_experts=num_experts,
                          top_k=top_k, ep_group=ep_group)
        self.last_aux = torch.zeros(())

    def forward(self, x):
        out, aux = self.moe(x)
        self.last_aux = aux
        return out


class SwinMoE(SwinTransformer):
    def __init__(self, num_experts=8, top_k=1, moe_stages=(2, 3),
                 moe_interval=2, ep_group=None, **kw):
        super().__init__(**kw)
        self.moe_adapters = []
        for si in moe_stages:
            layer = self.layers[si]
            for bi, blk in enumerate(layer.blocks):
                if bi % moe_interval == 1:  # every other block (ref style)
                    dim = blk.dim
                    hidden = blk.mlp.fc1.out_features
                    blk.mlp = MoEMlpAdapter(dim, hidden, num_experts, top_k,
                                            ep_group)
                    self.moe_adapters.append(blk.mlp)

    def aux_loss(self):
        return sum(a.last_aux for a in self.moe_adapters)


@register_model
def swin_moe_t(num_classes=1000, num_experts=8, **kw):
    return SwinMoE(embed_dim=96, depths=(2, 2, 6, 2), num_heads=(3, 6, 12, 24),
                   num_classes=num_classes, num_experts=num_experts, **kw)


@register_model
def swin_moe_s(num_classes=1000, num_experts=16, **kw):
    return SwinMoE(embed_dim=96, depths=(2, 2, 18, 2),
                   num_heads=(3, 6, 12, 24), num_classes=num_classes,
                   num_experts=num_experts, **kw)
