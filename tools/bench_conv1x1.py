#!/usr/bin/env python3
"""Microbenchmark: conv1x1 MFMA kernel vs MIOpen nn.Conv2d on the ResNet-50
1x1 shapes (batch 256, bf16, channels_last), plus the fused conv+BN chain
vs conv + separate BN, and CE old-vs-new launch overhead.

Run on a GPU box:  python tools/bench_conv1x1.py [--quick]
"""
import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from deeplearning_amd.ops._ext import ext  # noqa: E402


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


# ResNet-50 1x1 conv shapes at batch 256 / 224^2: (HW, Cin, Cout)
SHAPES = [
    (56 * 56, 256, 64),
    (56 * 56, 64, 256),
    (28 * 28, 512, 128),
    (28 * 28, 128, 512),
    (14 * 14, 1024, 256),
    (14 * 14, 256, 1024),
    (7 * 7, 2048, 512),
    (7 * 7, 512, 2048),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--batch", type=int, default=256)
    args = ap.parse_args()
    iters = 20 if args.quick else 50
    B = args.batch
    torch.backends.cudnn.benchmark = True

    print(f"batch={B}  (GEMM M = B*HW)")
    print(f"{'shape':>24} {'miopen':>9} {'dla':>9} {'ratio':>6} "
          f"{'dla GB/s':>9} {'dla TF':>7}")
    for HW, K, N in SHAPES:
        M = B * HW
        a = torch.randn(M, K, device="cuda").to(torch.bfloat16)
        w = torch.randn(N, K, device="cuda").to(torch.bfloat16)
        # 4d view for the conv: [B, K, H, W] channels_last
        H = int(HW ** 0.5)
        x4 = a.view(B, H, H, K).permute(0, 3, 1, 2)
        conv = torch.nn.Conv2d(K, N, 1, bias=False).cuda() \
            .to(memory_format=torch.channels_last).to(torch.bfloat16)
        conv.weight.data.copy_(w.view(N, K, 1, 1))

        t_miopen = timeit(lambda: conv(x4), iters)
        t_dla = timeit(
            lambda: ext().conv1x1_fwd(a, w, None, None, None, None, False,
                                      False), iters)
        t_dla_stats = timeit(
            lambda: ext().conv1x1_fwd(a, w, None, None, None, None, False,
                                      True), iters)
        bytes_moved = (M * K + M * N + N * K) * 2
        gbps = bytes_moved / (t_dla * 1e-3) / 1e9
        tf = 2 * M * K * N / (t_dla * 1e-3) / 1e12
        print(f"{f'{M}x{K}x{N}':>24} {t_miopen:9.3f} {t_dla:9.3f} "
              f"{t_miopen / t_dla:6.2f} {gbps:9.0f} {tf:7.1f}"
              f"   (+stats {t_dla_stats:.3f})")

        # wgrad
        dy = torch.randn(M, N, device="cuda").to(torch.bfloat16)
        t_wg = timeit(lambda: ext().conv1x1_wgrad(dy, a), iters)
        # MIOpen wgrad via autograd on weight only
        x4g = x4.detach()
        wref = conv.weight
        def miopen_wgrad():
            wref.grad = None
            y = torch.nn.functional.conv2d(x4g, wref)
            y.backward(dy.view(B, H, H, N).permute(0, 3, 1, 2))
        wref.requires_grad_(True)
        t_wg_m = timeit(miopen_wgrad, iters // 2, 5)
        print(f"{'wgrad':>24} {t_wg_m:9.3f} {t_wg:9.3f} "
              f"{t_wg_m / t_wg:6.2f}   (miopen incl fwd)")

    # fused conv+bn chain vs conv + BN kernels
    print("\nconv+BN train chain (56x56x256->64 class shape):")
    from deeplearning_amd.ops.batchnorm import BatchNorm2d
    from deeplearning_amd.ops.conv1x1 import conv_bn
    B4, C, Hh, N = 256, 256, 56, 64
    x = torch.randn(B4, C, Hh, Hh, device="cuda").to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    conv = torch.nn.Conv2d(C, N, 1, bias=False).cuda()
    bn = BatchNorm2d(N, relu=True).cuda()
    with torch.no_grad():
        t_fused = timeit(lambda: conv_bn(x, conv, bn), iters)
    convm = torch.nn.Conv2d(C, N, 1, bias=False).cuda() \
        .to(memory_format=torch.channels_last).to(torch.bfloat16)
    with torch.no_grad():
        t_sep = timeit(lambda: bn(convm(x)), iters)
    print(f"  fused {t_fused:.3f} ms  vs miopen-conv + HIP BN {t_sep:.3f} ms")

    # CE: new single-launch mean path
    print("\nCE (B=256, C=1000):")
    from deeplearning_amd.ops import cross_entropy
    logits = torch.randn(256, 1000, device="cuda").to(torch.bfloat16)
    logits.requires_grad_(True)
    tgt = torch.randint(0, 1000, (256,), device="cuda")
    def ce_step():
        logits.grad = None
        loss = cross_entropy(logits, tgt)
        loss.backward()
    t_ce = timeit(ce_step, iters)
    def ce_eager():
        logits.grad = None
        loss = torch.nn.functional.cross_entropy(logits.float(), tgt)
        loss.backward()
    t_cee = timeit(ce_eager, iters)
    print(f"  dla {t_ce:.3f} ms  eager {t_cee:.3f} ms  ratio "
          f"{t_cee / t_ce:.2f}")


if __name__ == "__main__":
    main()
