#!/usr/bin/env python3
"""Supervised-contrastive pretraining (reference: self-supervised/SupCon/trainer/trainer.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse
import time

import torch
from torch.utils.data import DataLoader

from deeplearning_amd.core.checkpoint import save_checkpoint
from deeplearning_amd.core.dist import get_rank, init_distributed, cleanup, \
    is_main_process
from deeplearning_amd.core.env import (increment_path, seed_everything,
                                       select_device)
from deeplearning_amd.core.logging import create_logger
from deeplearning_amd.data import SyntheticClassification
from deeplearning_amd.models import build_model
from deeplearning_amd.models.ssl import SupConLoss

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--temperature", type=float, default=0.07)
    p.add_argument("--device", default="cuda")
    p.add_argument("--output", default="runs")
    p.add_argument("--synthetic-size", type=int, default=32)
    args = p.parse_args()

    init_distributed()
    seed_everything(0, rank=get_rank())
    device = select_device(args.device)
    run_dir = increment_path(Path(args.output) / "supcon")
    logger = create_logger(str(run_dir) if is_main_process() else None,
                           dist_rank=get_rank())

    model = build_model("supcon_resnet50").to(device)
    crit = SupConLoss(temperature=args.temperature)
    opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9,
                          weight_decay=1e-4)
    ds = SyntheticClassification(args.synthetic_size, (3, 128, 128), 10)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True,
                        drop_last=True)
    amp = device.type == "cuda"
    for epoch in range(args.epochs):
        model.train()
        t0, tot = time.time(), 0.0
        for x, y in loader:
            # two views per sample (synthetic noise augmentation)
            x = x.to(device)
            y = y.to(device)
            v1 = model(x + 0.05 * torch.randn_like(x))
            v2 = model(x + 0.05 * torch.randn_like(x))
            feats = torch.stack([v1, v2], dim=1)
            loss = crit(feats, labels=y)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            tot += float(loss.detach())
        logger.info(f"epoch {epoch}: loss {tot / len(loader):.4f} "
                    f"({time.time() - t0:.1f}s)")
        if is_main_process():
            save_checkpoint(run_dir / "weights" / f"ckpt_epoch_{epoch}.pth",
                            model, opt, epoch=epoch)
    cleanup()
