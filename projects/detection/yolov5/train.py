#!/usr/bin/env python3
"""Train YOLOv5 (reference: detection/yolov5/train.py — yaml-scaled model, ComputeLoss, EMA, nominal-batch accumulate)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse
import time

import torch
from torch.utils.data import DataLoader

from deeplearning_amd.core.checkpoint import save_checkpoint, save_weights
from deeplearning_amd.core.dist import (cleanup, get_rank, get_world_size,
                                        init_distributed, is_main_process)
from deeplearning_amd.core.env import (increment_path, seed_everything,
                                       select_device)
from deeplearning_amd.core.logging import create_logger
from deeplearning_amd.engine.cli_det import SyntheticDetection
from deeplearning_amd.models import build_model
from deeplearning_amd.models.detection import ComputeLoss
from deeplearning_amd.ops import ModelEMA


def to_yolo_targets(targets, img_size):
    """list of dicts (xyxy pixels) -> [N,6] (img, cls, cx, cy, w, h) 0-1."""
    rows = []
    for i, t in enumerate(targets):
        if t["boxes"].numel() == 0:
            continue
        b = t["boxes"] / img_size
        cxcywh = torch.stack([(b[:, 0] + b[:, 2]) / 2,
                              (b[:, 1] + b[:, 3]) / 2,
                              b[:, 2] - b[:, 0], b[:, 3] - b[:, 1]], 1)
        idx = torch.full((b.shape[0], 1), float(i))
        rows.append(torch.cat([idx, t["labels"][:, None].float(),
                               cxcywh], 1))
    return torch.cat(rows) if rows else torch.zeros(0, 6)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="yolov5s",
                   choices=["yolov5s", "yolov5m", "yolov5l", "yolov5x"])
    p.add_argument("--num-classes", type=int, default=80)
    p.add_argument("--img-size", type=int, default=256)
    p.add_argument("--epochs", type=int, default=3)
    p.add_argument("--batch-size", type=int, default=4)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--ema", action="store_true", default=True)
    p.add_argument("--device", default="cuda")
    p.add_argument("--output", default="runs")
    p.add_argument("--synthetic-size", type=int, default=16)
    args = p.parse_args()

    init_distributed()
    seed_everything(0, rank=get_rank())
    device = select_device(args.device)
    run_dir = increment_path(Path(args.output) / "yolov5")
    logger = create_logger(str(run_dir) if is_main_process() else None,
                           dist_rank=get_rank())

    model = build_model(args.model, num_classes=args.num_classes).to(device)
    if get_world_size() > 1:
        from deeplearning_amd.parallel import wrap_data_parallel
        model = wrap_data_parallel(model)
    compute_loss = ComputeLoss(model.module if hasattr(model, "module")
                               else model)
    ema = ModelEMA(model) if args.ema else None
    ds = SyntheticDetection(args.synthetic_size,
                            (3, args.img_size, args.img_size),
                            args.num_classes)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True,
                        collate_fn=SyntheticDetection.collate_fn)
    opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.937,
                          weight_decay=5e-4, nesterov=True)
    # nominal batch 64: accumulate gradients to reach it (ref train.py:176)
    accumulate = max(round(64 / args.batch_size), 1)
    amp = device.type == "cuda"
    opt.zero_grad(set_to_none=True)
    it = 0
    for epoch in range(args.epochs):
        model.train()
        t0 = time.time()
        tot = 0.0
        for images, targets in loader:
            x = torch.stack(list(images)).to(device)
            t = to_yolo_targets(targets, args.img_size).to(device)
            with torch.autocast(device.type, dtype=torch.bfloat16,
                                enabled=amp):
                preds = model(x)
                loss, items = compute_loss(preds, t)
            loss.backward()
            it += 1
            if it % accumulate == 0:
                finalize = getattr(model, "finalize", None)
                if finalize is not None:
                    finalize()
                opt.step()
                opt.zero_grad(set_to_none=True)
                if ema:
                    ema.update(model)
            tot += float(loss.detach())
        logger.info(f"epoch {epoch}: loss {tot / len(loader):.4f} "
                    f"({time.time() - t0:.1f}s)")
        if is_main_process():
            save_weights(ema.ema if ema else model,
                         run_dir / "weights" / f"model_{epoch}.pth")
            save_checkpoint(run_dir / "weights" / f"ckpt_epoch_{epoch}.pth",
                            model, opt, epoch=epoch)
    cleanup()
