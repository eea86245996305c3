#!/usr/bin/env python3
"""BDB metric-learning training (reference: metric_learning/BDB/trainers/trainer.py — triplet + softmax, CMC/mAP eval)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse
import time

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader

from deeplearning_amd.core.checkpoint import save_checkpoint
from deeplearning_amd.core.env import seed_everything, select_device, \
    increment_path
from deeplearning_amd.core.logging import create_logger
from deeplearning_amd.data import SyntheticClassification
from deeplearning_amd.models import build_model
from deeplearning_amd.models.metric import TripletLoss, cmc_map

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--num-ids", type=int, default=8)
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--margin", type=float, default=0.3)
    p.add_argument("--device", default="cuda")
    p.add_argument("--output", default="runs")
    args = p.parse_args()

    seed_everything(0)
    device = select_device(args.device)
    run_dir = increment_path(Path(args.output) / "bdb")
    logger = create_logger(str(run_dir))
    model = build_model("bdb_resnet50", num_classes=args.num_ids).to(device)
    triplet = TripletLoss(margin=args.margin)
    opt = torch.optim.AdamW(model.parameters(), lr=args.lr)
    ds = SyntheticClassification(32, (3, 128, 64), args.num_ids)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True,
                        drop_last=True)
    for epoch in range(args.epochs):
        model.train()
        t0, tot = time.time(), 0.0
        for x, y in loader:
            x, y = x.to(device), y.to(device)
            out = model(x)
            loss = (triplet(out["global_feat"], y) +
                    triplet(out["part_feat"], y) +
                    F.cross_entropy(out["global_logits"], y) +
                    F.cross_entropy(out["part_logits"], y))
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            tot += float(loss.detach())
        logger.info(f"epoch {epoch}: loss {tot / len(loader):.4f} "
                    f"({time.time() - t0:.1f}s)")
        save_checkpoint(run_dir / "weights" / f"ckpt_epoch_{epoch}.pth",
                        model, opt, epoch=epoch)
    # retrieval eval (query == gallery on synthetic): CMC/mAP protocol
    model.eval()
    feats, ids = [], []
    with torch.no_grad():
        for x, y in loader:
            feats.append(model(x.to(device)).cpu())
            ids.append(y)
    feats, ids = torch.cat(feats), torch.cat(ids)
    cmc, mAP = cmc_map(feats, ids, feats, ids)
    logger.info(f"CMC@1 {float(cmc[0]):.3f} mAP {float(mAP):.3f}")
