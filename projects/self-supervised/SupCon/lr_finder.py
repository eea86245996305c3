#!/usr/bin/env python3
"""Learning-rate range test (reference: self-supervised/SupCon/
learning_rate_finder.py): exponential LR sweep, report the steepest-descent
LR."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse

import torch
from torch.utils.data import DataLoader

from deeplearning_amd.core.env import seed_everything, select_device
from deeplearning_amd.data import SyntheticClassification
from deeplearning_amd.models import build_model
from deeplearning_amd.ops import cross_entropy


def lr_range_test(model, loader, device, lr_min=1e-6, lr_max=1.0, steps=50,
                  beta=0.9):
    opt = torch.optim.SGD(model.parameters(), lr=lr_min, momentum=0.9)
    mult = (lr_max / lr_min) ** (1.0 / max(steps - 1, 1))
    lrs, losses = [], []
    avg = 0.0
    it = iter(loader)
    for i in range(steps):
        try:
            x, y = next(it)
        except StopIteration:
            it = iter(loader)
            x, y = next(it)
        x, y = x.to(device), y.to(device)
        loss = cross_entropy(model(x), y)
        opt.zero_grad(set_to_none=True)
        loss.backward()
        opt.step()
        avg = beta * avg + (1 - beta) * float(loss)
        smoothed = avg / (1 - beta ** (i + 1))
        lrs.append(opt.param_groups[0]["lr"])
        losses.append(smoothed)
        if i > 10 and smoothed > 4 * min(losses):
            break  # diverged
        for g in opt.param_groups:
            g["lr"] *= mult
    # steepest descent point
    grads = [(losses[i + 1] - losses[i]) for i in range(len(losses) - 1)]
    best = min(range(len(grads)), key=lambda i: grads[i]) if grads else 0
    return lrs, losses, lrs[best]


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet18")
    p.add_argument("--num-classes", type=int, default=10)
    p.add_argument("--steps", type=int, default=40)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--device", default="cuda")
    args = p.parse_args()
    seed_everything(0)
    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes).to(device)
    ds = SyntheticClassification(64, (3, 64, 64), args.num_classes)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True)
    lrs, losses, suggestion = lr_range_test(model, loader, device,
                                            steps=args.steps)
    for lr, ls in zip(lrs[::5], losses[::5]):
        print(f"lr {lr:10.2e}  loss {ls:.4f}")
    print(f"suggested lr: {suggestion:.2e}")
