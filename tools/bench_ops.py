#!/usr/bin/env python3
"""Per-op A/B microbenchmarks: our HIP kernels vs PyTorch eager on GPU.

Run on a GPU box:  python tools/bench_ops.py
Prints median ms per op config; within-process interleaved A/B
(guide §5.4 rule 24: interleave rounds in ONE process).
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from deeplearning_amd import ops  # noqa: E402


def timeit(fn, iters=30, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    times = []
    for _ in range(iters):
        ev0.record()
        fn()
        ev1.record()
        torch.cuda.synchronize()
        times.append(ev0.elapsed_time(ev1))
    times.sort()
    return times[len(times) // 2]


def ab(name, ours, eager, iters=30):
    # interleaved rounds
    t_ours, t_eager = [], []
    for _ in range(3):
        t_ours.append(timeit(ours, iters // 3))
        t_eager.append(timeit(eager, iters // 3))
    o, e = min(t_ours), min(t_eager)
    flag = "OURS" if o <= e else "EAGER"
    print(f"{name:45s} ours {o:8.3f} ms   eager {e:8.3f} ms   ratio {e/o:5.2f}x  [{flag}]")


def bench_bn(N, C, H, W, dtype=torch.bfloat16, nhwc=True):
    x = torch.randn(N, C, H, W, device="cuda", dtype=dtype)
    if nhwc:
        x = x.contiguous(memory_format=torch.channels_last)
    x.requires_grad_(True)
    bn = ops.BatchNorm2d(C).cuda()
    bn_ref = torch.nn.BatchNorm2d(C).cuda()
    if nhwc:
        bn_ref = bn_ref.to(memory_format=torch.channels_last)
    g = torch.randn_like(x)

    def ours():
        y = bn(x)
        y.backward(g)
        x.grad = None

    def eager():
        y = F.relu(bn_ref(x))
        y.backward(g)
        x.grad = None

    lay = "nhwc" if nhwc else "nchw"
    ab(f"bn+relu fwd+bwd {N}x{C}x{H}x{W} {lay}", ours, eager)


def main():
    assert torch.cuda.is_available()
    torch.backends.cudnn.benchmark = True
    dev = "cuda"

    # ---- BN shapes from ResNet-50 @ bs256 ----
    for shape in [(256, 64, 56, 56), (256, 256, 56, 56), (256, 512, 28, 28),
                  (256, 1024, 14, 14), (256, 2048, 7, 7)]:
        bench_bn(*shape)

    # ---- add_relu ----
    a = torch.randn(256, 256, 56, 56, device=dev, dtype=torch.bfloat16)
    b = torch.randn_like(a)
    ab("add_relu 256x256x56x56 bf16",
       lambda: ops.add_relu(a, b), lambda: torch.relu(a + b))

    # ---- LayerNorm fwd+bwd (ViT-B shape) ----
    x = torch.randn(256, 197, 768, device=dev, dtype=torch.bfloat16, requires_grad=True)
    w = torch.ones(768, device=dev, dtype=torch.bfloat16, requires_grad=True)
    bias = torch.zeros(768, device=dev, dtype=torch.bfloat16, requires_grad=True)
    g = torch.randn(256, 197, 768, device=dev, dtype=torch.bfloat16)

    def ln_ours():
        y = ops.layer_norm(x, w, bias)
        y.backward(g)
        x.grad = None; w.grad = None; bias.grad = None

    def ln_eager():
        y = F.layer_norm(x, (768,), w, bias)
        y.backward(g)
        x.grad = None; w.grad = None; bias.grad = None

    ab("layernorm fwd+bwd 256x197x768 bf16", ln_ours, ln_eager)

    # ---- GELU ----
    x2 = torch.randn(256, 197, 3072, device=dev, dtype=torch.bfloat16, requires_grad=True)
    g2 = torch.randn_like(x2)

    def gelu_ours():
        y = ops.gelu(x2)
        y.backward(g2)
        x2.grad = None

    def gelu_eager():
        y = F.gelu(x2)
        y.backward(g2)
        x2.grad = None

    ab("gelu fwd+bwd 256x197x3072 bf16", gelu_ours, gelu_eager)

    # ---- CE ----
    logits = torch.randn(256, 1000, device=dev, dtype=torch.bfloat16, requires_grad=True)
    tgt = torch.randint(0, 1000, (256,), device=dev)

    def ce_ours():
        loss = ops.cross_entropy(logits, tgt)
        loss.backward()
        logits.grad = None

    def ce_eager():
        loss = F.cross_entropy(logits, tgt)
        loss.backward()
        logits.grad = None

    ab("cross_entropy fwd+bwd 256x1000 bf16", ce_ours, ce_eager)

    # ---- NMS ----
    n = 20000
    boxes = torch.rand(n, 4, device=dev) * 500
    boxes[:, 2:] = boxes[:, :2] + torch.rand(n, 2, device=dev) * 100 + 1
    scores = torch.rand(n, device=dev)
    t = timeit(lambda: ops.nms(boxes, scores, 0.5), iters=10)
    print(f"{'nms 20k boxes (ours)':45s} ours {t:8.3f} ms")

    # ---- window partition (Swin canonical shape) ----
    xw = torch.randn(192, 56, 56, 96, device=dev, dtype=torch.bfloat16, requires_grad=True)
    gw = None

    def win_ours():
        y = ops.roll_and_window_partition(xw, 7, 3)
        y.sum().backward()
        xw.grad = None

    def win_eager():
        x2_ = torch.roll(xw, (-3, -3), (1, 2))
        y = ops.window_partition_eager(x2_, 7)
        y.sum().backward()
        xw.grad = None

    ab("roll+window_partition 192x56x56x96 bf16", win_ours, win_eager)


if __name__ == "__main__":
    main()
