#!/usr/bin/env python3
"""Run ONE conv1x1 kernel in a loop so rocprofv3 --pmc counters reflect it.

Usage: python tools/pmc_probe.py {fwd|wgrad|fwd_stats} [iters]
Shape: the weak ResNet-50 class M=802816, K=64, N=256.
"""
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from deeplearning_amd.ops._ext import ext  # noqa: E402

which = sys.argv[1] if len(sys.argv) > 1 else "fwd"
iters = int(sys.argv[2]) if len(sys.argv) > 2 else 30

if which == "attn":
    # ViT-B/16 attention shape: B=256, N=197, H=12, D=64
    qkv = torch.randn(256, 197, 3 * 768, device="cuda").to(torch.bfloat16)
    torch.cuda.synchronize()
    import time
    for _ in range(5):
        ext().attn_fwd(qkv, 12, 0.125)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ext().attn_fwd(qkv, 12, 0.125)
    torch.cuda.synchronize()
    print(f"attn_fwd {(time.perf_counter() - t0) / iters * 1e3:.3f} ms")
elif which == "fwd_fat":
    m, k, n = 200704, 128, 512
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16)
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16)
    torch.cuda.synchronize()
    for _ in range(iters):
        ext().conv1x1_fwd(a, w, None, None, None, None, False, False)
    torch.cuda.synchronize()
else:
    m, k, n = 802816, 64, 256
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16)
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16)
    dy = torch.randn(m, n, device="cuda").to(torch.bfloat16)
    torch.cuda.synchronize()
    for _ in range(iters):
        if which == "fwd":
            ext().conv1x1_fwd(a, w, None, None, None, None, False, False)
        elif which == "fwd_stats":
            ext().conv1x1_fwd(a, w, None, None, None, None, False, True)
        else:
            ext().conv1x1_wgrad(dy, a)
    torch.cuda.synchronize()
print("done", which, iters)
