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

# 3x3 s1 shapes (ResNet conv2 classes, batch 64 to bound memory)
import torch.nn.functional as F
print("--- conv3x3 fwd vs MIOpen ---")
for name, B, H, cin, cout in [("56x64", 64, 56, 64, 64),
                              ("28x128", 64, 28, 128, 128),
                              ("14x256", 64, 14, 256, 256),
                              ("7x512", 64, 7, 512, 512)]:
    x = torch.randn(B, H, H, cin, device="cuda").to(torch.bfloat16)
    a = x.reshape(-1, cin)
    w = torch.randn(cout, cin, 3, 3, device="cuda").to(torch.bfloat16)
    w9 = w.permute(0, 2, 3, 1).reshape(cout, 9 * cin).contiguous()
    x4 = x.permute(0, 3, 1, 2)
    conv = torch.nn.Conv2d(cin, cout, 3, padding=1, bias=False).cuda() \
        .to(memory_format=torch.channels_last).to(torch.bfloat16)
    conv.weight.data.copy_(w)
    for _ in range(10):
        ext().conv3x3_fwd(a, w9, H, H, None, None, None, None, False, False)
        conv(x4)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(40):
        ext().conv3x3_fwd(a, w9, H, H, None, None, None, None, False, False)
    torch.cuda.synchronize()
    t_dla = (time.perf_counter() - t0) / 40 * 1e3
    t0 = time.perf_counter()
    for _ in range(40):
        conv(x4)
    torch.cuda.synchronize()
    t_mio = (time.perf_counter() - t0) / 40 * 1e3
    fl = 2 * a.shape[0] * 9 * cin * cout
    print(f"3x3 {name}: dla {t_dla:.3f} ms ({fl/t_dla/1e9:.0f} TF) "
          f"miopen {t_mio:.3f} ms  ratio {t_mio/t_dla:.2f}")
