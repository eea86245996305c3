#!/usr/bin/env python3
"""MAE pretraining (reference: self-supervised/MAE/train.py — masked autoencoder, AdamW or LARS for large batch)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse
import time

import torch
from torch.utils.data import DataLoader

from deeplearning_amd.core.checkpoint import save_checkpoint
from deeplearning_amd.core.dist import (cleanup, get_rank, get_world_size,
                                        init_distributed, is_main_process)
from deeplearning_amd.core.env import (increment_path, seed_everything,
                                       select_device)
from deeplearning_amd.core.logging import create_logger
from deeplearning_amd.data import SyntheticClassification
from deeplearning_amd.engine.lars import LARC
from deeplearning_amd.engine.scheduler import WarmupScheduler
from deeplearning_amd.models import build_model

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="mae_vit_base_patch16",
                   choices=["mae_vit_base_patch16", "mae_vit_large_patch16"])
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--lr", type=float, default=1.5e-4)
    p.add_argument("--mask-ratio", type=float, default=0.75)
    p.add_argument("--lars", action="store_true")
    p.add_argument("--device", default="cuda")
    p.add_argument("--output", default="runs")
    p.add_argument("--synthetic-size", type=int, default=32)
    args = p.parse_args()

    init_distributed()
    seed_everything(0, rank=get_rank())
    device = select_device(args.device)
    run_dir = increment_path(Path(args.output) / "mae")
    logger = create_logger(str(run_dir) if is_main_process() else None,
                           dist_rank=get_rank())

    model = build_model(args.model, mask_ratio=args.mask_ratio).to(device)
    if get_world_size() > 1:
        from deeplearning_amd.parallel import wrap_data_parallel
        model = wrap_data_parallel(model)
    opt = torch.optim.AdamW(model.parameters(), lr=args.lr,
                            betas=(0.9, 0.95), weight_decay=0.05)
    if args.lars:
        opt = LARC(opt)
    ds = SyntheticClassification(args.synthetic_size, (3, 224, 224), 1000)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True,
                        drop_last=True)
    sched = WarmupScheduler(opt if not args.lars else opt.optim,
                            total_steps=args.epochs * len(loader),
                            warmup_steps=len(loader) // 2)
    amp = device.type == "cuda"
    for epoch in range(args.epochs):
        model.train()
        t0, tot = time.time(), 0.0
        for x, _ in loader:
            x = x.to(device)
            with torch.autocast(device.type, dtype=torch.bfloat16,
                                enabled=amp):
                loss, _, _ = model(x)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            finalize = getattr(model, "finalize", None)
            if finalize is not None:
                finalize()
            opt.step()
            sched.step()
            tot += float(loss.detach())
        logger.info(f"epoch {epoch}: loss {tot / len(loader):.4f} "
                    f"({time.time() - t0:.1f}s)")
        if is_main_process():
            save_checkpoint(run_dir / "weights" / f"ckpt_epoch_{epoch}.pth",
                            model, opt if not args.lars else opt.optim,
                            epoch=epoch)
    cleanup()
