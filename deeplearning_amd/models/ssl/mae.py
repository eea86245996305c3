"""MAE: masked-autoencoder ViT pretraining (75% patch masking).

Reference parity: self-supervised/MAE/models/MAE.py (masking/shuffle forward
:72-144) and models/VIT.py — re-designed on this repo's ViT blocks (HIP
LayerNorm/GELU). The shuffle/gather/unshuffle-scatter path uses batched
torch.gather on GPU (one gather per direction; a dedicated kernel buys nothing
over gather's bandwidth-bound copy).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import LayerNorm
from ..classification.vit import Block, PatchEmbed
from ..registry import register_model


class MAE(nn.Module):
    def __init__(self, img_size=224, patch_size=16, in_chans=3,
                 embed_dim=768, depth=12, num_heads=12,
                 decoder_embed_dim=512, decoder_depth=8, decoder_num_heads=16,
                 mlp_ratio=4.0, mask_ratio=0.75, norm_pix_loss=False):
        super().__init__()
        self.mask_ratio = mask_ratio
        self.norm_pix_loss = norm_pix_loss
        self.patch_size = patch_size

        # encoder
        self.patch_embed = PatchEmbed(img_size, patch_size, in_chans, embed_dim)
        num_patches = self.patch_embed.num_patches
        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.pos_embed = nn.Parameter(torch.zeros(1, num_patches + 1, embed_dim))
        self.blocks = nn.ModuleList([
            Block(embed_dim, num_heads, mlp_ratio, qkv_bias=True,
                  norm_layer=LayerNorm) for _ in range(depth)])
        self.norm = LayerNorm(embed_dim)

        # decoder
        self.decoder_embed = nn.Linear(embed_dim, decoder_embed_dim)
        self.mask_token = nn.Parameter(torch.zeros(1, 1, decoder_embed_dim))
        self.decoder_pos_embed = nn.Parameter(
            torch.zeros(1, num_patches + 1, decoder_embed_dim))
        self.decoder_blocks = nn.ModuleList([
            Block(decoder_embed_dim, decoder_num_heads, mlp_ratio,
                  qkv_bias=True, norm_layer=LayerNorm)
            for _ in range(decoder_depth)])
        self.decoder_norm = LayerNorm(decoder_embed_dim)
        self.decoder_pred = nn.Linear(decoder_embed_dim,
                                      patch_size ** 2 * in_chans)

        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.decoder_pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)
        nn.init.trunc_normal_(self.mask_token, std=0.02)

    # --- patch <-> pixel -----------------------------------------------
    def patchify(self, imgs):
        p = self.patch_size
        B, C, H, W = imgs.shape
        h, w = H // p, W // p
        x = imgs.reshape(B, C, h, p, w, p)
        x = torch.einsum("nchpwq->nhwpqc", x)
        return x.reshape(B, h * w, p * p * C)

    def unpatchify(self, x):
        p = self.patch_size
        B, L, D = x.shape
        h = w = int(L ** 0.5)
        C = D // (p * p)
        x = x.reshape(B, h, w, p, p, C)
        x = torch.einsum("nhwpqc->nchpwq", x)
        return x.reshape(B, C, h * p, w * p)

    # --- random shuffle masking (ref MAE.py:86-97) ---------------------
    def random_masking(self, x, mask_ratio):
        B, L, D = x.shape
        len_keep = int(L * (1 - mask_ratio))
        noise = torch.rand(B, L, device=x.device)
        ids_shuffle = torch.argsort(noise, dim=1)
        ids_restore = torch.argsort(ids_shuffle, dim=1)
        ids_keep = ids_shuffle[:, :len_keep]
        x_masked = torch.gather(
            x, 1, ids_keep.unsqueeze(-1).expand(-1, -1, D))
        mask = torch.ones(B, L, device=x.device)
        mask[:, :len_keep] = 0
        mask = torch.gather(mask, 1, ids_restore)
        return x_masked, mask, ids_restore

    def forward_encoder(self, imgs, mask_ratio):
        x = self.patch_embed(imgs)
        x = x + self.pos_embed[:, 1:]
        x, mask, ids_restore = self.random_masking(x, mask_ratio)
        cls = self.cls_token + self.pos_embed[:, :1]
        x = torch.cat([cls.expand(x.shape[0], -1, -1), x], dim=1)
        for blk in self.blocks:
            x = blk(x)
        return self.norm(x), mask, ids_restore

    def forward_decoder(self, x, ids_restore):
        x = self.decoder_embed(x)
        B, _, D = x.shape
        L = ids_restore.shape[1]
        mask_tokens = self.mask_token.expand(B, L + 1 - x.shape[1], -1)
        x_ = torch.cat([x[:, 1:], mask_tokens], dim=1)  # drop cls
        x_ = torch.gather(x_, 1, ids_restore.unsqueeze(-1).expand(-1, -1, D))
        x = torch.cat([x[:, :1], x_], dim=1)
        x = x + self.decoder_pos_embed
        for blk in self.decoder_blocks:
            x = blk(x)
        return self.decoder_pred(self.decoder_norm(x))[:, 1:]

    def forward_loss(self, imgs, pred, mask):
        target = self.patchify(imgs)
        if self.norm_pix_loss:
            mean = target.mean(dim=-1, keepdim=True)
            var = target.var(dim=-1, keepdim=True)
            target = (target - mean) / (var + 1e-6) ** 0.5
        loss = ((pred - target) ** 2).mean(dim=-1)
        return (loss * mask).sum() / mask.sum()

    def forward(self, imgs, mask_ratio=None):
        mask_ratio = mask_ratio if mask_ratio is not None else self.mask_ratio
        latent, mask, ids_restore = self.forward_encoder(imgs, mask_ratio)
        pred = self.forward_decoder(latent, ids_restore)
        loss = self.forward_loss(imgs, pred, mask)
        return loss, pred, mask


@register_model
def mae_vit_base_patch16(**kw):
    return MAE(embed_dim=768, depth=12, num_heads=12, **kw)


@register_model
def mae_vit_large_patch16(**kw):
    return MAE(embed_dim=1024, depth=24, num_heads=16, **kw)
