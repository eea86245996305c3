"""Parity + timing probe for the DLA_C1X1_RING=1 4-slot glds ring on the
fat-N conv1x1 fwd shapes. Run twice (env on / off) to A/B — the gate is
latched at first kernel call."""
import os
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from deeplearning_amd.ops.conv1x1 import ext  # noqa: E402


def t_ms(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    E = ext()
    tag = "ring" if os.environ.get("DLA_C1X1_RING") == "1" else "base"
    torch.manual_seed(0)
    for M, K, N in [(200704, 128, 512), (200704, 256, 1024),
                    (50176, 512, 1024), (50176, 256, 128)]:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        y = E.conv1x1_fwd(a, b)[0]
        ref = (a.float() @ b.float().t()).to(torch.bfloat16)
        err = (y.float() - ref.float()).abs().max()
        ok = bool(err <= 0.5)  # one bf16 ulp at |y|~2^6 (K-sum scale)
        ys, ss = E.conv1x1_fwd(a, b, want_stats=True)
        sums = ss.sum(0)
        serr = (sums[:N] - y.float().sum(0)).abs().max() / \
            y.float().sum(0).abs().max()
        ms = t_ms(lambda: E.conv1x1_fwd(a, b))
        tf = 2 * M * K * N / (ms * 1e-3) / 1e12
        print(f"{tag} {M}x{K}x{N}: {ms:.3f} ms ({tf:.0f} TF) maxerr {err:.3f}"
              f" parity={'OK' if ok else 'FAIL'} stats_rel {serr:.2e}")
        if not ok:
            sys.exit(1)


if __name__ == "__main__":
    main()
