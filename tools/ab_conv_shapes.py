#!/usr/bin/env python3
"""Quick fwd timing over the ResNet fat-N shapes (BK A/B via env)."""
import os
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from deeplearning_amd.ops._ext import ext  # noqa: E402

tag = os.environ.get("DLA_C1X1_BK32", "0")
for name, m, k, n in [("128x512", 200704, 128, 512),
                      ("256x1024", 50176, 256, 1024),
                      ("512x2048", 12544, 512, 2048),
                      ("64x256", 802816, 64, 256),
                      ("256x64", 802816, 256, 64)]:
    a = torch.randn(m, k, device="cuda").to(torch.bfloat16)
    w = torch.randn(n, k, device="cuda").to(torch.bfloat16)
    for _ in range(10):
        ext().conv1x1_fwd(a, w, None, None, None, None, False, False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(40):
        ext().conv1x1_fwd(a, w, None, None, None, None, False, False)
    torch.cuda.synchronize()
    print(f"{name} bk32={tag} {(time.perf_counter() - t0) / 40 * 1e3:.3f} ms")
