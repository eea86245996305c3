#!/usr/bin/env python3
"""ONNX export for any registry model (reference: others/deploy/pytorch2onnx/*, yolov5/export.py, YOLOX tools/export_onnx.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse

import torch

from deeplearning_amd.models import build_model

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--img-size", type=int, default=224)
    p.add_argument("--weights", default="")
    p.add_argument("--out", default="model.onnx")
    p.add_argument("--opset", type=int, default=17)
    p.add_argument("--dynamic-batch", action="store_true")
    args = p.parse_args()

    model = build_model(args.model, num_classes=args.num_classes)
    if args.weights:
        from deeplearning_amd.core.checkpoint import load_pretrained
        load_pretrained(model, args.weights)
    model.eval()
    x = torch.randn(1, 3, args.img_size, args.img_size)
    dyn = {"input": {0: "batch"}, "output": {0: "batch"}} \
        if args.dynamic_batch else None
    try:
        torch.onnx.export(model, x, args.out, opset_version=args.opset,
                          input_names=["input"], output_names=["output"],
                          dynamic_axes=dyn, dynamo=False)
        print(f"exported {args.model} -> {args.out} (onnx)")
    except Exception as e:  # onnx package absent in this image
        ts_out = args.out.rsplit(".", 1)[0] + ".torchscript.pt"
        traced = torch.jit.trace(model, x)
        traced.save(ts_out)
        print(f"onnx unavailable ({type(e).__name__}); "
              f"exported TorchScript -> {ts_out}")
