#!/usr/bin/env python3
"""Fused-attention microbench: kernel vs eager chain, fwd and fwd+bwd,
at the shapes the zoo actually runs (ViT-B 197/64, Swin windows 49/32,
MAE encoder 50/64). No convs -> no MIOpen find -> starts in seconds;
suitable as a rocprofv3 --pmc target.
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from deeplearning_amd.ops.attention import _eager_attention, fused_attention

SHAPES = [  # (tag, B, N, H, d)
    ("vit_b", 256, 197, 12, 64),
    ("swin_w7", 4096, 49, 3, 32),
    ("mae_enc", 256, 50, 12, 64),
]


def timeit(fn, iters=30, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--mode", default="both", choices=["fwd", "bwd", "both"])
    p.add_argument("--iters", type=int, default=30)
    args = p.parse_args()
    assert torch.cuda.is_available()
    for tag, B, N, H, d in SHAPES:
        qkv = torch.randn(B, N, 3 * H * d, device="cuda").bfloat16()
        scale = d ** -0.5
        if args.mode in ("fwd", "both"):
            with torch.no_grad():
                t_f = timeit(lambda: fused_attention(qkv, H, scale),
                             args.iters)
                t_e = timeit(lambda: _eager_attention(qkv, H, scale),
                             args.iters)
            print(f"{tag:8s} fwd      B={B} N={N} H={H} d={d}  "
                  f"ours {t_f:7.3f} ms   eager {t_e:7.3f} ms   "
                  f"{t_e / t_f:5.2f}x")
        if args.mode in ("bwd", "both"):
            g = torch.randn(B, N, H * d, device="cuda").bfloat16()

            def run_fused():
                q = qkv.detach().requires_grad_()
                fused_attention(q, H, scale).backward(g)

            def run_eager():
                q = qkv.detach().requires_grad_()
                _eager_attention(q, H, scale).backward(g)

            t_f = timeit(run_fused, args.iters)
            t_e = timeit(run_eager, args.iters)
            print(f"{tag:8s} fwd+bwd  B={B} N={N} H={H} d={d}  "
                  f"ours {t_f:7.3f} ms   eager {t_e:7.3f} ms   "
                  f"{t_e / t_f:5.2f}x")


if __name__ == "__main__":
    main()
