#!/usr/bin/env python3
"""Happy-Whale retrieval baseline (reference: metric_learning/Happy-Whale/retrieval/train.py — hard-mining triplet, lovasz aux)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import argparse

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader

from deeplearning_amd.core.env import seed_everything, select_device
from deeplearning_amd.data import SyntheticClassification
from deeplearning_amd.models import build_model
from deeplearning_amd.models.metric import TripletLoss

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--num-ids", type=int, default=8)
    p.add_argument("--backbone", default="supcon_resnet50",
                   choices=["supcon_resnet50", "dpn68", "dpn92",
                            "inception_v4", "xception", "senet154",
                            "polynet", "nasnet_a_mobile", "nasnet_a_large"],
                   help="retrieval backbone (reference modelZoo: dpn/"
                        "inceptionV4/xception/senet/ployNet/nasnet)")
    p.add_argument("--epochs", type=int, default=2)
    p.add_argument("--batch-size", type=int, default=8)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--device", default="cuda")
    args = p.parse_args()

    seed_everything(0)
    device = select_device(args.device)
    if args.backbone == "supcon_resnet50":
        model = build_model(args.backbone, feat_dim=256).to(device)
    else:
        # zoo backbone -> pooled features -> 256-d embedding (ref
        # retrieval/train.py feature head)
        bb = build_model(args.backbone, num_classes=1)
        embed = torch.nn.Linear(bb.num_features, 256)

        class _Retrieval(torch.nn.Module):
            def __init__(self, bb, embed):
                super().__init__()
                self.bb, self.embed = bb, embed

            def forward(self, x):
                f = self.bb.forward_features(x)
                f = torch.nn.functional.adaptive_avg_pool2d(f, 1).flatten(1)
                return torch.nn.functional.normalize(self.embed(f), dim=1)

        model = _Retrieval(bb, embed).to(device)
    head = torch.nn.Linear(256, args.num_ids).to(device)
    triplet = TripletLoss(margin=None)  # soft-margin (ref :38-189)
    opt = torch.optim.AdamW(list(model.parameters()) +
                            list(head.parameters()), lr=args.lr)
    ds = SyntheticClassification(32, (3, 128, 128), args.num_ids)
    loader = DataLoader(ds, batch_size=args.batch_size, shuffle=True,
                        drop_last=True)
    for epoch in range(args.epochs):
        tot = 0.0
        for x, y in loader:
            x, y = x.to(device), y.to(device)
            f = model(x)
            loss = triplet(f, y) + F.cross_entropy(head(f), y)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            tot += float(loss.detach())
        print(f"epoch {epoch}: loss {tot / len(loader):.4f}")
