#!/usr/bin/env python3
"""Stochastic Weight Averaging fine-tune (reference: self-supervised/SupCon/
swa.py) on torch.optim.swa_utils."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse

import torch
from torch.optim.swa_utils import SWALR, AveragedModel, update_bn
from torch.utils.data import DataLoader

from deeplearning_amd.core.env import seed_everything, select_device
from deeplearning_amd.data import SyntheticClassification
from deeplearning_amd.models import build_model
from deeplearning_amd.ops import cross_entropy

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet18")
    p.add_argument("--num-classes", type=int, default=10)
    p.add_argument("--epochs", type=int, default=3)
    p.add_argument("--swa-start", type=int, default=1)
    p.add_argument("--swa-lr", type=float, default=0.005)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--device", default="cuda")
    p.add_argument("--weights", default="")
    p.add_argument("--out", default="swa_model.pth")
    args = p.parse_args()

    seed_everything(0)
    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes).to(device)
    if args.weights:
        from deeplearning_amd.core.checkpoint import load_pretrained
        load_pretrained(model, args.weights)
    swa_model = AveragedModel(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    swa_sched = SWALR(opt, swa_lr=args.swa_lr)
    ds = SyntheticClassification(32, (3, 64, 64), args.num_classes)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True)
    for epoch in range(args.epochs):
        model.train()
        for x, y in loader:
            x, y = x.to(device), y.to(device)
            loss = cross_entropy(model(x), y)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
        if epoch >= args.swa_start:
            swa_model.update_parameters(model)
            swa_sched.step()
        print(f"epoch {epoch}: loss {float(loss):.4f}")
    update_bn(loader, swa_model, device=device)
    torch.save(swa_model.module.state_dict(), args.out)
    print(f"saved SWA weights -> {args.out}")
