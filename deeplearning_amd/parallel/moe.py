"""Expert-parallel MoE MLP (Swin-MoE parity).

Reference parity: classification/swin_transformer/models/swin_transformer_moe.py
:36-94 (tutel moe_layer in MoEMlp: top-k gating with capacity factor 1.25,
cosine router option, num_local_experts per rank, all-to-all expert dispatch)
— re-designed for RCCL over xGMI: token dispatch is ONE all_to_all_single per
direction on the EP group (xGMI is point-to-point, so all-to-all uses all 7
links of each GPU concurrently — the best collective for expert exchange).

Single-process (no EP group): all experts are local, dispatch is a gather.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from ..core.dist import get_world_size, is_dist


class _AllToAll(torch.autograd.Function):
    """Autograd-aware all_to_all_single (equal splits): raw dist collectives
    are invisible to autograd, so the EP dispatch/return would otherwise cut
    the graph — expert weights and the router would get NO gradient in EP
    mode. Backward of an equal-split all-to-all is the same all-to-all on
    the gradient (send/recv roles swap symmetrically)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        out = torch.empty_like(x)
        dist.all_to_all_single(out, x.contiguous(), group=group)
        return out

    @staticmethod
    def backward(ctx, grad):
        g = torch.empty_like(grad)
        dist.all_to_all_single(g, grad.contiguous(), group=ctx.group)
        return g, None


class CosineRouter(nn.Module):
    """Cosine-similarity router (Swin-MoE style) with learnable temperature."""

    def __init__(self, dim, num_experts, init_t=0.5):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(num_experts, dim))
        nn.init.normal_(self.weight, std=0.02)
        self.temperature = nn.Parameter(torch.log(torch.tensor(1.0 / init_t)))

    def forward(self, x):
        logits = F.normalize(x, dim=-1) @ F.normalize(self.weight, dim=-1).t()
        return logits * self.temperature.exp().clamp(max=100.0)


class MoEMlp(nn.Module):
    """Top-k gated mixture-of-experts MLP with optional expert parallelism.

    With an EP process group of size W and E total experts, each rank holds
    E/W local experts; tokens are exchanged with all_to_all_single.
    """

    def __init__(self, dim, hidden_dim, num_experts=8, top_k=1,
                 capacity_factor=1.25, cosine_router=True, ep_group=None):
        super().__init__()
        self.dim = dim
        self.num_experts = num_experts
        self.top_k = top_k
        self.capacity_factor = capacity_factor
        self.ep_group = ep_group
        self.ep_size = (dist.get_world_size(ep_group)
                        if is_dist() and get_world_size() > 1 else 1)
        assert num_experts % self.ep_size == 0, \
            "num_experts must divide by EP world size"
        self.num_local = num_experts // self.ep_size
        self.router = CosineRouter(dim, num_experts) if cosine_router \
            else nn.Linear(dim, num_experts, bias=False)
        # local experts: [num_local] two-layer MLPs stored as batched weights
        self.w1 = nn.Parameter(torch.empty(self.num_local, dim, hidden_dim))
        self.b1 = nn.Parameter(torch.zeros(self.num_local, hidden_dim))
        self.w2 = nn.Parameter(torch.empty(self.num_local, hidden_dim, dim))
        self.b2 = nn.Parameter(torch.zeros(self.num_local, dim))
        nn.init.trunc_normal_(self.w1, std=0.02)
        nn.init.trunc_normal_(self.w2, std=0.02)
        # mark expert params so DDP skips them (each rank owns its experts;
        # reference pattern: _ddp_params_and_buffers_to_ignore)
        for p in (self.w1, self.b1, self.w2, self.b2):
            p.expert = True
        self.aux_loss = torch.zeros(())

    def _capacity(self, tokens):
        return max(4, int(self.capacity_factor * tokens * self.top_k /
                          self.num_experts))

    def forward(self, x):
        B, N, C = x.shape
        tokens = x.reshape(-1, C)
        T = tokens.shape[0]
        logits = self.router(tokens)                       # T, E
        probs = logits.softmax(dim=-1)
        gate, expert_idx = probs.topk(self.top_k, dim=-1)  # T, k

        # load-balance aux loss (mean prob * mean assignment per expert)
        with torch.no_grad():
            assign = F.one_hot(expert_idx[:, 0], self.num_experts).float()
        self.aux_loss = (probs.mean(0) * assign.mean(0)).sum() * \
            self.num_experts

        cap = self._capacity(T)
        out = torch.zeros_like(tokens)
        # flatten (token, k) pairs
        flat_expert = expert_idx.reshape(-1)               # T*k
        flat_gate = gate.reshape(-1)
        flat_tok = torch.arange(T, device=x.device).repeat_interleave(
            self.top_k)
        order = flat_expert.argsort(stable=True)
        flat_expert = flat_expert[order]
        flat_gate = flat_gate[order]
        flat_tok = flat_tok[order]
        # capacity: drop tokens beyond cap per expert (sorted by expert, so
        # position-within-expert = index - expert_start)
        counts = torch.bincount(flat_expert, minlength=self.num_experts)
        starts = torch.cumsum(
            torch.cat([counts.new_zeros(1), counts[:-1]]), 0)
        pos_in_expert = torch.arange(
            flat_expert.numel(), device=x.device) - starts[flat_expert]
        keep = pos_in_expert < cap
        flat_expert = flat_expert[keep]
        flat_gate = flat_gate[keep]
        flat_tok = flat_tok[keep]

        if self.ep_size == 1:
            # all experts local: batched per-expert GEMMs
            for e in range(self.num_experts):
                sel = flat_expert == e
                if not bool(sel.any()):
                    continue
                toks = tokens[flat_tok[sel]]
                h = F.gelu(toks @ self.w1[e] + self.b1[e])
                y = h @ self.w2[e] + self.b2[e]
                # gate comes from an fp32 softmax under autocast
                out.index_add_(0, flat_tok[sel],
                               (y * flat_gate[sel, None]).to(out.dtype))
            return out.reshape(B, N, C), self.aux_loss

        # ---- expert parallel: pad per-expert to capacity, all-to-all -------
        E, W, L = self.num_experts, self.ep_size, self.num_local
        buf = tokens.new_zeros(E, cap, C)
        slot = pos_in_expert[keep]
        buf[flat_expert, slot] = tokens[flat_tok]
        # send experts to their owner rank: rank r owns experts [r*L,(r+1)*L)
        send = buf.reshape(W, L * cap, C)
        recv = _AllToAll.apply(send, self.ep_group)
        # recv: [W senders][L local experts * cap] -> per local expert batch
        recv = recv.reshape(W, L, cap, C).transpose(0, 1) \
            .reshape(L, W * cap, C)
        h = F.gelu(torch.bmm(recv, self.w1) + self.b1[:, None])
        y = torch.bmm(h, self.w2) + self.b2[:, None]      # L, W*cap, C
        y = y.reshape(L, W, cap, C).transpose(0, 1).reshape(W, L * cap, C)
        back = _AllToAll.apply(y, self.ep_group).reshape(E, cap, C)
        contrib = (back[flat_expert, slot] *
                   flat_gate[:, None]).to(out.dtype)
        out.index_add_(0, flat_tok, contrib)
        return out.reshape(B, N, C), self.aux_loss


def expert_params(module: nn.Module):
    """Parameters marked as expert-local (exclude from DDP all-reduce)."""
    return [p for p in module.parameters() if getattr(p, "expert", False)]
