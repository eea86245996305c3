#!/usr/bin/env python3
"""Standalone ResNet50-FPN feature extraction (reference: detection/FPN/fpn_model.py — model-only subproject)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import torch

from deeplearning_amd.models.detection import resnet_fpn_backbone

if __name__ == "__main__":
    model = resnet_fpn_backbone()
    model.eval()
    with torch.no_grad():
        feats = model(torch.randn(1, 3, 224, 224))
    for name, f in feats.items():
        print(f"P{name}: {tuple(f.shape)}")
