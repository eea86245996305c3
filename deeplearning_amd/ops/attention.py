"""Fused multi-head attention (HIP MFMA kernel, csrc/attention.hip).

Forward runs softmax(scale * Q K^T + bias + mask) V in ONE kernel straight
from the packed qkv projection — no permute copies, no fp32 score tensor, no
separate softmax launches. Backward uses the kernel-saved bf16 P with batched
GEMMs (dV = P^T dO, dP = dO V^T, dS = P*(dP - rowsum(dP*P)), dQ = dS K,
dK = dS^T Q).

Eager reference (CPU + numerics tests): plain einsum/softmax chain, identical
math. Reference parity: ViT Attention (vit_model.py:88-113), Swin
WindowAttention (swin models/swin_transformer.py:118-151).
"""
from __future__ import annotations

import torch

from ._ext import ext, use_hip


def _eager_attention(qkv: torch.Tensor, num_heads: int, scale: float,
                     bias=None, mask=None):
    B, N, _ = qkv.shape[0], qkv.shape[1], qkv.shape[2]
    qkv = qkv.reshape(B, N, 3, num_heads, -1).permute(2, 0, 3, 1, 4)
    q, k, v = qkv.unbind(0)  # B, H, N, d
    attn = (q @ k.transpose(-2, -1)) * scale
    if bias is not None:
        attn = attn + bias.unsqueeze(0)
    if mask is not None:
        nW = mask.shape[0]
        attn = attn.view(B // nW, nW, num_heads, N, N) + \
            mask.unsqueeze(1).unsqueeze(0)
        attn = attn.view(B, num_heads, N, N)
    attn = attn.softmax(dim=-1)
    out = (attn @ v).transpose(1, 2).reshape(B, N, -1)
    return out


class _AttnFusedBwdFn(torch.autograd.Function):
    """No-bias/mask path: forward saves only per-row softmax stats; backward
    recomputes P in-LDS (two HIP kernels, no [B,H,N,N] tensor ever hits HBM)."""

    @staticmethod
    def forward(ctx, qkv, num_heads, scale):
        needs = qkv.requires_grad
        res = ext().attn_fwd(qkv.contiguous(), num_heads, scale, None, None,
                             False, needs)
        ctx.num_heads = num_heads
        ctx.scale = scale
        if needs:
            ctx.save_for_backward(qkv, res[0], res[1])
        return res[0]

    @staticmethod
    def backward(ctx, dout):
        qkv, out, stats = ctx.saved_tensors
        H = ctx.num_heads
        B, N, _ = qkv.shape
        dout = dout.contiguous()
        # D_row[b,h,n] = sum_d dO * O  (one fused reduce)
        drow = (dout.float() * out.float()).view(B, N, H, -1).sum(-1) \
            .permute(0, 2, 1).contiguous()
        (dqkv,) = ext().attn_bwd(qkv, dout.to(qkv.dtype), stats, drow, H,
                                 ctx.scale)
        return dqkv, None, None


class _AttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, num_heads, scale, bias, mask):
        needs_p = qkv.requires_grad or \
            (bias is not None and bias.requires_grad)
        res = ext().attn_fwd(qkv.contiguous(), num_heads, scale, bias, mask,
                             needs_p, False)
        ctx.num_heads = num_heads
        ctx.scale = scale
        ctx.has_bias = bias is not None
        ctx.has_mask = mask is not None
        if needs_p:
            ctx.save_for_backward(qkv, res[1])
        return res[0]

    @staticmethod
    def backward(ctx, dout):
        qkv, p = ctx.saved_tensors
        H = ctx.num_heads
        B, N, _ = qkv.shape
        d = qkv.shape[2] // (3 * H)
        qkv_v = qkv.reshape(B, N, 3, H, d).permute(2, 0, 3, 1, 4)
        q, k, v = qkv_v.unbind(0)  # B,H,N,d (views)
        dout = dout.reshape(B, N, H, d).permute(0, 2, 1, 3)  # B,H,N,d
        p32 = p  # bf16 GEMMs on MFMA
        dv = p32.transpose(-2, -1) @ dout
        dp = dout @ v.transpose(-2, -1)
        ds = p32 * (dp - (dp * p32).sum(dim=-1, keepdim=True))
        dq = (ds @ k) * ctx.scale
        dk = ds.transpose(-2, -1) @ q * ctx.scale
        dqkv = torch.stack([dq, dk, dv], dim=0)  # 3,B,H,N,d
        dqkv = dqkv.permute(1, 3, 0, 2, 4).reshape(B, N, 3 * H * d)
        dbias = None
        if ctx.has_bias:
            dbias = ds.sum(dim=0)
        return dqkv, None, None, dbias, None


def fused_attention(qkv: torch.Tensor, num_heads: int, scale: float,
                    bias: torch.Tensor | None = None,
                    mask: torch.Tensor | None = None) -> torch.Tensor:
    """qkv: [B, N, 3*H*d] (packed projection). Returns [B, N, H*d].

    bias: [H, N, N] additive (relative position bias); mask: [nW, N, N]
    additive window mask (Swin). Falls back to eager off-GPU or out of the
    kernel's shape domain (d in {32,64}, N <= 256, bf16).
    """
    d = qkv.shape[2] // (3 * num_heads)
    if (use_hip(qkv) and qkv.dtype == torch.bfloat16 and d in (32, 64)
            and qkv.shape[1] <= 256):
        if bias is None and mask is None:
            return _AttnFusedBwdFn.apply(qkv, num_heads, scale)
        return _AttnFn.apply(qkv, num_heads, scale, bias, mask)
    return _eager_attention(qkv, num_heads, scale, bias, mask)


class _AttnCosineFn(torch.autograd.Function):
    """Swin-v2 cosine attention: fused HIP forward (in-kernel q/k row norms
    + per-head logit scale), backward = cosine chain rule from the
    kernel-saved P (same forward-fused/backward-from-P split as the v1
    bias path). Gradients flow to qkv, logit_scale and bias."""

    @staticmethod
    def forward(ctx, qkv, num_heads, lscale, bias, mask):
        # grad mode is off inside Function.forward; the no-grad case was
        # already routed around this Function by the wrapper
        needs = (qkv.requires_grad or lscale.requires_grad or
                 (bias is not None and bias.requires_grad))
        res = ext().attn_fwd_cosine(qkv.contiguous(), num_heads,
                                    lscale.detach(), bias, mask, needs)
        ctx.num_heads = num_heads
        ctx.has_bias = bias is not None
        if needs:
            ctx.save_for_backward(qkv, lscale, res[1])
        return res[0]

    @staticmethod
    def backward(ctx, dout):
        import torch.nn.functional as F

        qkv, lscale, p = ctx.saved_tensors
        H = ctx.num_heads
        B, N, _ = qkv.shape
        d = qkv.shape[2] // (3 * H)
        q, k, v = qkv.reshape(B, N, 3, H, d).permute(2, 0, 3, 1, 4).unbind(0)
        dout = dout.reshape(B, N, H, d).permute(0, 2, 1, 3)
        qh = F.normalize(q.float(), dim=-1)
        kh = F.normalize(k.float(), dim=-1)
        pf = p.float()
        dv = p.transpose(-2, -1) @ dout
        dp = (dout @ v.transpose(-2, -1)).float()
        ds = pf * (dp - (dp * pf).sum(dim=-1, keepdim=True))
        c = qh @ kh.transpose(-2, -1)
        d_ls = (ds * c).sum(dim=(0, 2, 3))           # [H]
        dc = ds * lscale.view(1, -1, 1, 1)
        dqh = dc @ kh
        dkh = dc.transpose(-2, -1) @ qh
        qn = q.float().norm(dim=-1, keepdim=True).clamp_min(1e-12)
        kn = k.float().norm(dim=-1, keepdim=True).clamp_min(1e-12)
        dq = (dqh - (dqh * qh).sum(-1, keepdim=True) * qh) / qn
        dk = (dkh - (dkh * kh).sum(-1, keepdim=True) * kh) / kn
        dqkv = torch.stack([dq.to(qkv.dtype), dk.to(qkv.dtype), dv], dim=0)
        dqkv = dqkv.permute(1, 3, 0, 2, 4).reshape(B, N, 3 * H * d)
        dbias = ds.sum(dim=0) if ctx.has_bias else None
        return dqkv, None, d_ls.to(lscale.dtype), dbias, None


def fused_attention_cosine(qkv: torch.Tensor, num_heads: int,
                           logit_scale: torch.Tensor,
                           bias: torch.Tensor | None = None,
                           mask: torch.Tensor | None = None) -> torch.Tensor:
    """Swin-v2 cosine attention on the fused HIP kernel (train + eval).
    logit_scale: [H] already clamp(exp(param), max=100)."""
    d = qkv.shape[2] // (3 * num_heads)
    if (use_hip(qkv) and qkv.dtype == torch.bfloat16 and d in (32, 64)
            and qkv.shape[1] <= 256):
        if not torch.is_grad_enabled():
            return ext().attn_fwd_cosine(qkv.contiguous(), num_heads,
                                         logit_scale, bias, mask, False)[0]
        return _AttnCosineFn.apply(qkv, num_heads, logit_scale, bias, mask)
    # eager reference
    B, N, _ = qkv.shape
    q, k, v = qkv.reshape(B, N, 3, num_heads, -1).permute(
        2, 0, 3, 1, 4).unbind(0)
    attn = torch.nn.functional.normalize(q.float(), dim=-1) @ \
        torch.nn.functional.normalize(k.float(), dim=-1).transpose(-2, -1)
    attn = attn * logit_scale.view(1, -1, 1, 1)
    if bias is not None:
        attn = attn + bias.unsqueeze(0).float()
    if mask is not None:
        nW = mask.shape[0]
        attn = attn.view(B // nW, nW, num_heads, N, N) + \
            mask.float().unsqueeze(1).unsqueeze(0)
        attn = attn.view(B, num_heads, N, N)
    attn = attn.softmax(dim=-1).to(v.dtype)
    return (attn @ v).transpose(1, 2).reshape(B, N, -1)
