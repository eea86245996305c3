#!/usr/bin/env python3
"""Export YOLOX to ONNX/TorchScript (reference: detection/YOLOX/tools/export_onnx.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse

import torch

from deeplearning_amd.models import build_model


class _Decoded(torch.nn.Module):
    def __init__(self, m):
        super().__init__()
        self.m = m

    def forward(self, x):
        return self.m(x)  # decoded [B, n_anchors, 5+nc]


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="yolox_s")
    p.add_argument("--num-classes", type=int, default=80)
    p.add_argument("--img-size", type=int, default=640)
    p.add_argument("--weights", default="")
    p.add_argument("--out", default="yolox.onnx")
    p.add_argument("--opset", type=int, default=17)
    args = p.parse_args()

    model = build_model(args.model, num_classes=args.num_classes)
    if args.weights:
        from deeplearning_amd.core.checkpoint import load_pretrained
        load_pretrained(model, args.weights)
    model.eval()
    wrapped = _Decoded(model)
    wrapped.eval()
    x = torch.randn(1, 3, args.img_size, args.img_size)
    try:
        torch.onnx.export(wrapped, x, args.out, opset_version=args.opset,
                          input_names=["images"], output_names=["pred"],
                          dynamo=False)
        print(f"exported {args.model} -> {args.out}")
    except Exception as e:
        ts = args.out.rsplit(".", 1)[0] + ".torchscript.pt"
        torch.jit.trace(wrapped, x).save(ts)
        print(f"onnx unavailable ({type(e).__name__}); TorchScript -> {ts}")
