#!/usr/bin/env python3
"""Educational BN/LN/IN/GN numeric re-derivations vs torch built-ins (reference: others/normalization/*.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import torch


def batch_norm_manual(x, eps=1e-5):
    mean = x.mean(dim=(0, 2, 3), keepdim=True)
    var = x.var(dim=(0, 2, 3), unbiased=False, keepdim=True)
    return (x - mean) / (var + eps).sqrt()


def layer_norm_manual(x, eps=1e-5):
    mean = x.mean(dim=-1, keepdim=True)
    var = x.var(dim=-1, unbiased=False, keepdim=True)
    return (x - mean) / (var + eps).sqrt()


def instance_norm_manual(x, eps=1e-5):
    mean = x.mean(dim=(2, 3), keepdim=True)
    var = x.var(dim=(2, 3), unbiased=False, keepdim=True)
    return (x - mean) / (var + eps).sqrt()


def group_norm_manual(x, groups, eps=1e-5):
    B, C, H, W = x.shape
    xg = x.view(B, groups, C // groups, H, W)
    mean = xg.mean(dim=(2, 3, 4), keepdim=True)
    var = xg.var(dim=(2, 3, 4), unbiased=False, keepdim=True)
    return ((xg - mean) / (var + eps).sqrt()).view(B, C, H, W)


if __name__ == "__main__":
    x = torch.randn(4, 8, 6, 6)
    checks = {
        "batch_norm": (batch_norm_manual(x),
                       torch.nn.functional.batch_norm(
                           x, None, None, training=True)),
        "layer_norm": (layer_norm_manual(x),
                       torch.nn.functional.layer_norm(x, (6,))),
        "instance_norm": (instance_norm_manual(x),
                          torch.nn.functional.instance_norm(x)),
        "group_norm": (group_norm_manual(x, 4),
                       torch.nn.functional.group_norm(x, 4)),
    }
    for name, (a, b) in checks.items():
        err = (a - b).abs().max().item()
        print(f"{name}: max err {err:.2e}")
        assert err < 1e-5
