"""LARS / LARC layer-wise adaptive-rate optimizer wrapper for large-batch SSL.

Reference parity: self-supervised/MAE/utils/LARS.py — re-designed as a clean
torch.optim.Optimizer wrapper (trust-ratio clipping per parameter group).
"""
from __future__ import annotations

import torch


class LARC:
    """Wrap an optimizer; before each step, scale each param's LR by the
    trust ratio eta*||w||/(||g|| + wd*||w||), clipped at 1."""

    def __init__(self, optimizer: torch.optim.Optimizer, trust_coefficient=0.02,
                 clip=True, eps=1e-8):
        self.optim = optimizer
        self.trust_coefficient = trust_coefficient
        self.clip = clip
        self.eps = eps

    def __getattr__(self, name):
        return getattr(self.optim, name)

    @property
    def param_groups(self):
        return self.optim.param_groups

    def state_dict(self):
        return self.optim.state_dict()

    def load_state_dict(self, d):
        self.optim.load_state_dict(d)

    def zero_grad(self, set_to_none=True):
        self.optim.zero_grad(set_to_none=set_to_none)

    @torch.no_grad()
    def step(self, closure=None):
        weight_decays = []
        for group in self.optim.param_groups:
            wd = group.get("weight_decay", 0)
            weight_decays.append(wd)
            group["weight_decay"] = 0
            for p in group["params"]:
                if p.grad is None:
                    continue
                p_norm = torch.norm(p.data)
                g_norm = torch.norm(p.grad.data)
                if p_norm != 0 and g_norm != 0:
                    adaptive_lr = self.trust_coefficient * p_norm / \
                        (g_norm + p_norm * wd + self.eps)
                    if self.clip:
                        adaptive_lr = min(adaptive_lr / group["lr"], 1.0)
                    p.grad.data += wd * p.data
                    p.grad.data *= adaptive_lr
        loss = self.optim.step(closure)
        for group, wd in zip(self.optim.param_groups, weight_decays):
            group["weight_decay"] = wd
        return loss
