#!/usr/bin/env python3
"""COCO-style mAP validation for fasterRcnn from a trained checkpoint
(reference: detection/fasterRcnn validation.py pattern)."""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import torch

from deeplearning_amd.core.checkpoint import load_pretrained
from deeplearning_amd.core.env import select_device
from deeplearning_amd.engine.cli_det import \
    build_det_dataset, det_argparser
from deeplearning_amd.engine.det_eval import DetEvaluator
from deeplearning_amd.models import build_model
from torch.utils.data import DataLoader

if __name__ == "__main__":
    p = det_argparser("fasterrcnn_resnet50_fpn", num_classes=21, name="fasterRcnn-val")
    p.add_argument("--weights", required=True)
    args = p.parse_args()
    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes,
                        min_size=args.img_size,
                        max_size=args.img_size + 64).to(device)
    load_pretrained(model, args.weights)
    model.eval()
    ds = build_det_dataset(args)
    loader = DataLoader(ds, batch_size=args.batch_size,
                        collate_fn=getattr(type(ds), "collate_fn", None))
    ev = DetEvaluator()
    with torch.no_grad():
        for images, targets in loader:
            dets = model([im.to(device) for im in images])
            ev.update(dets, targets)
    stats = ev.summarize()
    print(f"mAP {stats['mAP']:.4f}  mAP50 {stats['mAP50']:.4f}  "
          f"mAP75 {stats['mAP75']:.4f}")
