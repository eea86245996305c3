#!/usr/bin/env python3
"""Kernel-weight and feature-map visualization (reference: others/visual_weight_feature_map_test/) — saves PNG grids."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse

import numpy as np
import torch
from PIL import Image

from deeplearning_amd.models import build_model


def tensor_grid_png(t, path, pad=1):
    """[N, H, W] tensor -> tiled grayscale PNG."""
    t = t.detach().float()
    t = (t - t.amin(dim=(1, 2), keepdim=True)) /         (t.amax(dim=(1, 2), keepdim=True) -
         t.amin(dim=(1, 2), keepdim=True) + 1e-8)
    n, h, w = t.shape
    cols = int(n ** 0.5 + 0.999)
    rows = (n + cols - 1) // cols
    canvas = np.ones((rows * (h + pad), cols * (w + pad)), dtype=np.float32)
    for i in range(n):
        r, c = divmod(i, cols)
        canvas[r * (h + pad):r * (h + pad) + h,
               c * (w + pad):c * (w + pad) + w] = t[i].numpy()
    Image.fromarray((canvas * 255).astype(np.uint8)).save(path)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet18")
    p.add_argument("--out-prefix", default="/tmp/viz")
    args = p.parse_args()
    model = build_model(args.model, num_classes=10)
    model.eval()
    # first-conv kernels
    w0 = model.conv1.weight if hasattr(model, "conv1") else \
        next(m.weight for m in model.modules()
             if isinstance(m, torch.nn.Conv2d))
    tensor_grid_png(w0[:, 0], f"{args.out_prefix}_kernels.png")
    # feature maps after the first conv
    feats = {}
    h = w0 if False else None
    mod = model.conv1 if hasattr(model, "conv1") else \
        next(m for m in model.modules() if isinstance(m, torch.nn.Conv2d))
    handle = mod.register_forward_hook(
        lambda m, i, o: feats.__setitem__("conv1", o))
    with torch.no_grad():
        model(torch.randn(1, 3, 224, 224))
    handle.remove()
    tensor_grid_png(feats["conv1"][0, :16], f"{args.out_prefix}_featmaps.png")
    print(f"saved {args.out_prefix}_kernels.png, "
          f"{args.out_prefix}_featmaps.png")
