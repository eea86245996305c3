"""Shared segmentation train driver for projects/Image_segmentation/*.

Reference parity: U-Net train.py:104-171, FCN train_multi_GPU.py, DeepLabV3
builder-style train.py:21-120 — one loop: CE(+aux)+optional dice, poly LR,
mIoU ConfusionMatrix eval with cross-rank reduce.
"""
from __future__ import annotations

import argparse
import time
from pathlib import Path

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, Dataset, DistributedSampler

from ..core.checkpoint import save_checkpoint, save_weights
from ..core.dist import (cleanup, get_rank, get_world_size, init_distributed,
                         is_main_process)
from ..core.env import increment_path, seed_everything, select_device
from ..core.logging import create_logger
from ..core.meters import AverageMeter
from ..engine.metrics import ConfusionMatrix, dice_loss
from ..engine.scheduler import WarmupScheduler
from ..models import build_model


class SyntheticSegmentation(Dataset):
    def __init__(self, length=64, image_size=(3, 128, 128), num_classes=21,
                 seed=0):
        g = torch.Generator().manual_seed(seed)
        self.images = torch.randn(length, *image_size, generator=g)
        self.masks = torch.randint(0, num_classes,
                                   (length, *image_size[1:]), generator=g)

    def __len__(self):
        return len(self.images)

    def __getitem__(self, i):
        return self.images[i], self.masks[i]


def seg_argparser(default_model: str, **defaults):
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=default_model)
    p.add_argument("--data-path", default="",
                   help="empty = synthetic data")
    p.add_argument("--num-classes", type=int,
                   default=defaults.get("num_classes", 21))
    p.add_argument("--img-size", type=int,
                   default=defaults.get("img_size", 256))
    p.add_argument("--epochs", type=int, default=defaults.get("epochs", 10))
    p.add_argument("--batch-size", type=int,
                   default=defaults.get("batch_size", 4))
    p.add_argument("--lr", type=float, default=defaults.get("lr", 0.01))
    p.add_argument("--weight-decay", type=float, default=1e-4)
    p.add_argument("--aux-weight", type=float, default=0.5)
    p.add_argument("--dice", action="store_true",
                   default=defaults.get("dice", False),
                   help="add dice loss (U-Net)")
    p.add_argument("--ohem", action="store_true",
                   default=defaults.get("ohem", False),
                   help="OHEM CE (HRNet)")
    p.add_argument("--device", default="cuda")
    p.add_argument("--workers", type=int, default=2)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--deterministic", action="store_true",
                   help="cudnn.deterministic + fixed seeds "
                        "(ref FCOS trainers/trainer.py:57-66)")
    p.add_argument("--amp", action="store_true", default=True)
    p.add_argument("--no-amp", dest="amp", action="store_false")
    p.add_argument("--output", default="runs")
    p.add_argument("--name", default=defaults.get("name", "seg"))
    p.add_argument("--synthetic-size", type=int, default=32)
    return p


def seg_criterion(outputs, target, aux_weight=0.5, use_dice=False,
                  ohem=None):
    out = outputs["out"] if isinstance(outputs, dict) else outputs
    if ohem is not None:
        loss = ohem(out, target)
    else:
        loss = F.cross_entropy(out, target, ignore_index=255)
    if use_dice:
        nc = out.shape[1]
        onehot = F.one_hot(target.clamp(0, nc - 1), nc)
        onehot = onehot.permute(0, 3, 1, 2).float()
        loss = loss + dice_loss(out, onehot)
    if isinstance(outputs, dict) and "aux" in outputs:
        loss = loss + aux_weight * F.cross_entropy(outputs["aux"], target,
                                                   ignore_index=255)
    return loss


def seg_train_main(args) -> dict:
    info = init_distributed()
    if torch.cuda.is_available() and get_world_size() > 1:
        device = torch.device("cuda", info["local_rank"])
        torch.cuda.set_device(device)
    else:
        device = select_device(args.device)
    seed_everything(args.seed, rank=get_rank(),
                    deterministic=getattr(args, 'deterministic', False))

    run_dir = Path(increment_path(Path(args.output) / args.name)) \
        if is_main_process() else Path(args.output) / args.name
    logger = create_logger(str(run_dir) if is_main_process() else None,
                           dist_rank=get_rank())

    model = build_model(args.model, num_classes=args.num_classes).to(device)
    from ..core.complexity import count_params
    logger.info(f"model {args.model}: params {count_params(model) / 1e6:.1f}M")
    if get_world_size() > 1:
        from ..parallel import wrap_data_parallel
        model = wrap_data_parallel(model)

    ds = SyntheticSegmentation(args.synthetic_size,
                               (3, args.img_size, args.img_size),
                               args.num_classes)
    sampler = DistributedSampler(ds) if get_world_size() > 1 else None
    loader = DataLoader(ds, batch_size=args.batch_size,
                        shuffle=sampler is None, sampler=sampler,
                        num_workers=args.workers, drop_last=True)

    optimizer = torch.optim.SGD(
        [p for p in model.parameters() if p.requires_grad], lr=args.lr,
        momentum=0.9, weight_decay=args.weight_decay)
    scheduler = WarmupScheduler(optimizer,
                                total_steps=args.epochs * len(loader),
                                mode="poly")
    ohem = None
    if args.ohem:
        from ..models.segmentation import OhemCrossEntropy
        ohem = OhemCrossEntropy(min_kept=1000)

    amp = args.amp and device.type == "cuda"
    miou = 0.0
    for epoch in range(args.epochs):
        if sampler is not None:
            sampler.set_epoch(epoch)
        model.train()
        loss_m = AverageMeter()
        t0 = time.time()
        for x, y in loader:
            x, y = x.to(device), y.to(device)
            with torch.autocast(device.type, dtype=torch.bfloat16,
                                enabled=amp):
                out = model(x)
                loss = seg_criterion(out, y, args.aux_weight, args.dice,
                                     ohem)
            optimizer.zero_grad(set_to_none=True)
            loss.backward()
            finalize = getattr(model, "finalize", None)
            if finalize is not None:
                finalize()
            optimizer.step()
            scheduler.step()
            loss_m.update(float(loss.detach()), x.shape[0])
        # quick mIoU on the train set (synthetic): protocol check, not a metric
        cm = ConfusionMatrix(args.num_classes)
        model.eval()
        with torch.no_grad():
            for x, y in loader:
                x, y = x.to(device), y.to(device)
                out = model(x)
                out = out["out"] if isinstance(out, dict) else out
                cm.update(y.flatten(), out.argmax(1).flatten())
        cm.reduce_from_all_processes()
        acc, _, iou = cm.compute()
        miou = float(iou[iou == iou].mean())
        logger.info(f"epoch {epoch}: loss {loss_m.avg:.4f} mIoU {miou:.4f} "
                    f"({time.time() - t0:.1f}s)")
        if is_main_process():
            save_weights(model, run_dir / "weights" / f"model_{epoch}.pth")
            save_checkpoint(run_dir / "weights" / f"ckpt_epoch_{epoch}.pth",
                            model, optimizer, scheduler, epoch)
    cleanup()
    return {"miou": miou, "run_dir": str(run_dir)}


def seg_predict_main(default_model: str, num_classes: int = 21):
    """Single-image segmentation CLI: write the predicted mask as a palette
    PNG (ref Image_segmentation/U-Net predict, DeepLabV3 predict)."""
    import argparse as _ap

    import numpy as np
    from PIL import Image

    from ..core.checkpoint import load_pretrained
    from ..core.env import select_device
    from ..data.transforms import pil_to_tensor
    from ..models import build_model

    p = _ap.ArgumentParser()
    p.add_argument("image")
    p.add_argument("--model", default=default_model)
    p.add_argument("--weights", required=True)
    p.add_argument("--num-classes", type=int, default=num_classes)
    p.add_argument("--device", default="cuda")
    p.add_argument("--out", default="mask.png")
    args = p.parse_args()

    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes).to(device)
    load_pretrained(model, args.weights)
    model.eval()
    img = Image.open(args.image).convert("RGB")
    x = pil_to_tensor(img)[None].to(device)
    with torch.no_grad():
        out = model(x)
        logits = out["out"] if isinstance(out, dict) else out
    mask = logits.argmax(1)[0].byte().cpu().numpy()
    pal = Image.fromarray(mask, mode="P")
    # simple deterministic palette
    palette = []
    for c in range(256):
        palette += [(c * 37) % 256, (c * 91) % 256, (c * 173) % 256]
    pal.putpalette(palette)
    pal.save(args.out)
    classes = sorted(int(v) for v in np.unique(mask))
    print(f"classes present: {classes}; wrote {args.out}")
    return mask
