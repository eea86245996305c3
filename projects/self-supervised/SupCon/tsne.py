#!/usr/bin/env python3
"""t-SNE embedding visualization of encoder features (reference:
self-supervised/SupCon/t-SNE.py) — sklearn TSNE, scatter rendered to PNG
with PIL (no matplotlib in this image)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse

import numpy as np
import torch
from PIL import Image, ImageDraw
from sklearn.manifold import TSNE

from deeplearning_amd.core.env import seed_everything, select_device
from deeplearning_amd.data import SyntheticClassification
from deeplearning_amd.models import build_model

PALETTE = [(228, 26, 28), (55, 126, 184), (77, 175, 74), (152, 78, 163),
           (255, 127, 0), (255, 255, 51), (166, 86, 40), (247, 129, 191),
           (153, 153, 153), (0, 0, 0)]


def scatter_png(emb, labels, path, size=512):
    emb = (emb - emb.min(0)) / (np.ptp(emb, axis=0) + 1e-9)
    img = Image.new("RGB", (size, size), "white")
    draw = ImageDraw.Draw(img)
    for (x, y), l in zip(emb, labels):
        c = PALETTE[int(l) % len(PALETTE)]
        px, py = int(x * (size - 8)) + 4, int(y * (size - 8)) + 4
        draw.ellipse([px - 3, py - 3, px + 3, py + 3], fill=c)
    img.save(path)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="supcon_resnet50")
    p.add_argument("--weights", default="")
    p.add_argument("--num-samples", type=int, default=64)
    p.add_argument("--device", default="cuda")
    p.add_argument("--out", default="tsne.png")
    args = p.parse_args()

    seed_everything(0)
    device = select_device(args.device)
    model = build_model(args.model).to(device)
    if args.weights:
        from deeplearning_amd.core.checkpoint import load_pretrained
        load_pretrained(model, args.weights)
    model.eval()
    ds = SyntheticClassification(args.num_samples, (3, 64, 64), 10)
    feats, labels = [], []
    with torch.no_grad():
        for i in range(len(ds)):
            x, y = ds[i]
            feats.append(model(x[None].to(device)).cpu().numpy()[0])
            labels.append(y)
    emb = TSNE(n_components=2, init="pca",
               perplexity=min(30, len(feats) - 1)).fit_transform(
                   np.stack(feats))
    scatter_png(emb, labels, args.out)
    print(f"wrote {args.out} ({len(feats)} points)")
