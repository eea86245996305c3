#!/usr/bin/env python3
"""YOLOX mAP validation from a checkpoint (reference: YOLOX COCOEvaluator
path, yolox/evaluators/coco_evaluator.py)."""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import torch
from torch.utils.data import DataLoader

from deeplearning_amd.core.checkpoint import load_pretrained
from deeplearning_amd.core.env import select_device
from deeplearning_amd.engine.cli_det import SyntheticDetection
from deeplearning_amd.engine.det_eval import DetEvaluator
from deeplearning_amd.models import build_model
from deeplearning_amd.models.detection import yolox_postprocess

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="yolox_s")
    p.add_argument("--weights", required=True)
    p.add_argument("--num-classes", type=int, default=80)
    p.add_argument("--img-size", type=int, default=256)
    p.add_argument("--conf-thres", type=float, default=0.001)
    p.add_argument("--nms-thres", type=float, default=0.65)
    p.add_argument("--batch-size", type=int, default=4)
    p.add_argument("--synthetic-size", type=int, default=16)
    p.add_argument("--device", default="cuda")
    args = p.parse_args()

    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes).to(device)
    load_pretrained(model, args.weights)
    model.eval()
    ds = SyntheticDetection(args.synthetic_size,
                            (3, args.img_size, args.img_size),
                            args.num_classes)
    loader = DataLoader(ds, batch_size=args.batch_size,
                        collate_fn=SyntheticDetection.collate_fn)
    ev = DetEvaluator()
    with torch.no_grad():
        for images, targets in loader:
            x = torch.stack(list(images)).to(device)
            decoded = model(x)
            dets = yolox_postprocess(decoded, args.num_classes,
                                     conf_thre=args.conf_thres,
                                     nms_thre=args.nms_thres)
            ev.update(dets, targets)
    st = ev.summarize()
    print(f"mAP {st['mAP']:.4f}  mAP50 {st['mAP50']:.4f}  "
          f"mAP75 {st['mAP75']:.4f}")
