#!/usr/bin/env python3
"""HRNet keypoint training (reference: pose_estimation/Insulator/train.py — heatmap targets + focal heatmap loss + decode)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse
import time

import torch
from torch.utils.data import DataLoader, Dataset

from deeplearning_amd.core.checkpoint import save_checkpoint
from deeplearning_amd.core.dist import get_rank, init_distributed, cleanup, \
    is_main_process
from deeplearning_amd.core.env import (increment_path, seed_everything,
                                       select_device)
from deeplearning_amd.core.logging import create_logger
from deeplearning_amd.models import build_model
from deeplearning_amd.models.pose import (KeypointToHeatMap, decode_heatmaps,
                                          heatmap_focal_loss)


class SyntheticKeypoints(Dataset):
    def __init__(self, length=16, img_size=128, num_joints=4, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.imgs = torch.rand(length, 3, img_size, img_size, generator=g)
        self.kps = torch.rand(length, num_joints, 2, generator=g) * img_size
        self.to_heatmap = KeypointToHeatMap((img_size // 4, img_size // 4))

    def __len__(self):
        return len(self.imgs)

    def __getitem__(self, i):
        return self.imgs[i], self.to_heatmap(self.kps[i][None])[0], self.kps[i]


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--num-joints", type=int, default=4)
    p.add_argument("--img-size", type=int, default=128)
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=4)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--device", default="cuda")
    p.add_argument("--output", default="runs")
    args = p.parse_args()

    init_distributed()
    seed_everything(0, rank=get_rank())
    device = select_device(args.device)
    run_dir = increment_path(Path(args.output) / "insulator")
    logger = create_logger(str(run_dir) if is_main_process() else None,
                           dist_rank=get_rank())

    model = build_model("hrnet_w18_pose",
                        num_joints=args.num_joints).to(device)
    opt = torch.optim.AdamW(model.parameters(), lr=args.lr)
    ds = SyntheticKeypoints(16, args.img_size, args.num_joints)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True)
    amp = device.type == "cuda"
    for epoch in range(args.epochs):
        model.train()
        t0, tot = time.time(), 0.0
        for x, hm, _ in loader:
            x, hm = x.to(device), hm.to(device)
            with torch.autocast(device.type, dtype=torch.bfloat16,
                                enabled=amp):
                pred = model(x)
                loss = heatmap_focal_loss(pred.float(), hm)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            tot += float(loss.detach())
        # PCK-style check: decoded keypoint error
        model.eval()
        errs = []
        with torch.no_grad():
            for x, _, kp in loader:
                coords, _ = decode_heatmaps(model(x.to(device)).float())
                errs.append((coords.cpu() - kp).norm(dim=-1).mean())
        logger.info(f"epoch {epoch}: loss {tot / len(loader):.4f} "
                    f"kp-err {float(torch.stack(errs).mean()):.1f}px "
                    f"({time.time() - t0:.1f}s)")
        if is_main_process():
            save_checkpoint(run_dir / "weights" / f"ckpt_epoch_{epoch}.pth",
                            model, opt, epoch=epoch)
    cleanup()
