"""Shared detection train driver for projects/detection/*.

Reference parity: fasterRcnn train_resnet50_fpn.py:180-221 / train_multi_gpu.py,
RetinaNet train_utils/train_eval_utils.py:12-116, yolov5 train.py, YOLOX
trainer — one loop: warmup LR, loss-dict sum, reduce_dict logging, EMA option,
DetEvaluator eval with cross-rank merge.
"""
from __future__ import annotations

import argparse
import math
import time
from pathlib import Path

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler

from ..core.checkpoint import save_checkpoint, save_weights
from ..core.dist import (cleanup, get_rank, get_world_size, init_distributed,
                         is_main_process, reduce_dict)
from ..core.env import increment_path, seed_everything, select_device
from ..core.logging import create_logger
from ..core.meters import AverageMeter
from ..engine.det_eval import DetEvaluator
from ..engine.scheduler import WarmupScheduler
from ..models import build_model
from ..ops import ModelEMA


class SyntheticDetection(Dataset):
    """Random images with a few axis-aligned box targets."""

    def __init__(self, length=32, image_size=(3, 256, 256), num_classes=5,
                 max_boxes=4, seed=0):
        self.g = torch.Generator().manual_seed(seed)
        self.length = length
        self.image_size = image_size
        self.num_classes = num_classes
        self.max_boxes = max_boxes
        self.items = []
        H, W = image_size[1:]
        for i in range(length):
            n = int(torch.randint(1, max_boxes + 1, (1,),
                                  generator=self.g))
            xy = torch.rand(n, 2, generator=self.g) * \
                torch.tensor([W, H]) * 0.6
            wh = torch.rand(n, 2, generator=self.g) * \
                torch.tensor([W, H]) * 0.3 + 8
            boxes = torch.cat([xy, (xy + wh).clamp(
                max=torch.tensor([W - 1.0, H - 1.0]))], 1)
            labels = torch.randint(1, num_classes, (n,), generator=self.g)
            self.items.append((torch.rand(*image_size, generator=self.g),
                               {"boxes": boxes, "labels": labels,
                                "iscrowd": torch.zeros(n, dtype=torch.long),
                                "image_id": torch.tensor([i])}))

    def __len__(self):
        return self.length

    def __getitem__(self, i):
        return self.items[i]

    @staticmethod
    def collate_fn(batch):
        return tuple(zip(*batch))


def det_argparser(default_model: str, **defaults):
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=default_model)
    p.add_argument("--data-path", default="", help="VOC root or COCO dir; "
                   "empty = synthetic")
    p.add_argument("--dataset", default="voc", choices=["voc", "coco"])
    p.add_argument("--num-classes", type=int,
                   default=defaults.get("num_classes", 21))
    p.add_argument("--img-size", type=int,
                   default=defaults.get("img_size", 256))
    p.add_argument("--epochs", type=int, default=defaults.get("epochs", 10))
    p.add_argument("--batch-size", type=int,
                   default=defaults.get("batch_size", 2))
    p.add_argument("--lr", type=float, default=defaults.get("lr", 0.005))
    p.add_argument("--weight-decay", type=float, default=1e-4)
    p.add_argument("--ema", action="store_true",
                   default=defaults.get("ema", False))
    p.add_argument("--device", default="cuda")
    p.add_argument("--workers", type=int, default=2)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--deterministic", action="store_true",
                   help="cudnn.deterministic + fixed seeds "
                        "(ref FCOS trainers/trainer.py:57-66)")
    p.add_argument("--amp", action="store_true", default=True)
    p.add_argument("--no-amp", dest="amp", action="store_false")
    p.add_argument("--output", default="runs")
    p.add_argument("--name", default=defaults.get("name", "det"))
    p.add_argument("--synthetic-size", type=int, default=16)
    p.add_argument("--eval-every", type=int, default=0,
                   help="0 = eval only at the end")
    p.add_argument("--aspect-ratio-group-factor", type=int, default=-1,
                   help=">=0: group batches by aspect ratio (2k+1 buckets; "
                        "ref fasterRcnn utils/group_by_aspect_ratio.py)")
    return p


def build_det_dataset(args):
    if not args.data_path:
        return SyntheticDetection(args.synthetic_size,
                                  (3, args.img_size, args.img_size),
                                  args.num_classes)
    from ..data import COCODetectionDataset, VOCDetectionDataset
    if args.dataset == "coco":
        return COCODetectionDataset(
            Path(args.data_path) / "images",
            Path(args.data_path) / "annotations.json")
    return VOCDetectionDataset(args.data_path)


def det_train_main(args, model_kwargs=None) -> dict:
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

    model = build_model(args.model, num_classes=args.num_classes,
                        **(model_kwargs or {})).to(device)
    from ..core.complexity import count_params
    logger.info(f"model {args.model}: params {count_params(model) / 1e6:.1f}M")
    if get_world_size() > 1:
        from ..parallel import wrap_data_parallel
        model = wrap_data_parallel(model)
    ema = ModelEMA(model) if args.ema else None

    ds = build_det_dataset(args)
    sampler = DistributedSampler(ds) if get_world_size() > 1 else None
    group_factor = getattr(args, "aspect_ratio_group_factor", -1)
    if group_factor >= 0:
        from torch.utils.data import RandomSampler

        from ..data.samplers import (GroupedBatchSampler,
                                     compute_aspect_ratios,
                                     create_aspect_ratio_groups)
        base = sampler if sampler is not None else RandomSampler(ds)
        group_ids = create_aspect_ratio_groups(
            compute_aspect_ratios(ds), k=group_factor)
        batch_sampler = GroupedBatchSampler(base, group_ids, args.batch_size)
        loader = DataLoader(ds, batch_sampler=batch_sampler,
                            num_workers=args.workers,
                            collate_fn=getattr(type(ds), "collate_fn", None))
    else:
        loader = DataLoader(ds, batch_size=args.batch_size,
                            shuffle=sampler is None, sampler=sampler,
                            num_workers=args.workers,
                            collate_fn=getattr(type(ds), "collate_fn", None))

    optimizer = torch.optim.SGD(
        [p for p in model.parameters() if p.requires_grad], lr=args.lr,
        momentum=0.9, weight_decay=args.weight_decay)
    scheduler = WarmupScheduler(optimizer,
                                total_steps=args.epochs * len(loader),
                                warmup_steps=min(500, len(loader)))

    amp = args.amp and device.type == "cuda"
    for epoch in range(args.epochs):
        if sampler is not None:
            sampler.set_epoch(epoch)
        model.train()
        loss_m = AverageMeter()
        t0 = time.time()
        for images, targets in loader:
            images = [im.to(device) for im in images]
            targets = [{k: v.to(device) for k, v in t.items()}
                       for t in targets]
            with torch.autocast(device.type, dtype=torch.bfloat16,
                                enabled=amp):
                loss_dict = model(images, targets)
                loss = sum(loss_dict.values())
            if not math.isfinite(float(loss)):
                raise RuntimeError(f"non-finite loss {loss_dict}")
            optimizer.zero_grad(set_to_none=True)
            loss.backward()
            finalize = getattr(model, "finalize", None)
            if finalize is not None:
                finalize()
            optimizer.step()
            scheduler.step()
            if ema:
                ema.update(model)
            reduced = reduce_dict(loss_dict)
            loss_m.update(float(sum(reduced.values())), len(images))
        logger.info(f"epoch {epoch}: loss {loss_m.avg:.4f} "
                    f"({time.time() - t0:.1f}s)")
        if is_main_process():
            save_weights(model, run_dir / "weights" / f"model_{epoch}.pth")
            save_checkpoint(run_dir / "weights" / f"ckpt_epoch_{epoch}.pth",
                            model, optimizer, scheduler, epoch)

    # final eval: mAP on the train-set images (synthetic protocol check)
    evaluator = DetEvaluator()
    eval_model = ema.ema if ema else model
    eval_model.eval()
    with torch.no_grad():
        for images, targets in loader:
            images = [im.to(device) for im in images]
            dets = eval_model(images)
            evaluator.update(dets, targets)
    evaluator.synchronize_between_processes()
    stats = evaluator.summarize()
    logger.info(
        f"final mAP {stats['mAP']:.4f} mAP50 {stats['mAP50']:.4f} "
        f"mAP75 {stats['mAP75']:.4f} | s/m/l "
        f"{stats.get('mAP_small', 0):.4f}/{stats.get('mAP_medium', 0):.4f}/"
        f"{stats.get('mAP_large', 0):.4f} | AR1/10/100 "
        f"{stats.get('AR1', 0):.4f}/{stats.get('AR10', 0):.4f}/"
        f"{stats.get('AR100', 0):.4f}")
    cleanup()
    return {"mAP": stats["mAP"], "run_dir": str(run_dir)}


def detect_main(default_model: str, num_classes: int = 21):
    """Single-image detection CLI: load weights, print + optionally draw
    detections (ref detection/yolov5/detect.py, fasterRcnn predict.py)."""
    import argparse as _ap

    from PIL import Image, ImageDraw

    from ..core.checkpoint import load_pretrained
    from ..core.env import select_device
    from ..data.transforms import pil_to_tensor
    from ..models import build_model

    p = _ap.ArgumentParser()
    p.add_argument("image")
    p.add_argument("--model", default=default_model)
    p.add_argument("--weights", required=True)
    p.add_argument("--num-classes", type=int, default=num_classes)
    p.add_argument("--score-thresh", type=float, default=0.5)
    p.add_argument("--device", default="cuda")
    p.add_argument("--save", default="", help="write annotated PNG here")
    p.add_argument("--min-size", type=int, default=800)
    p.add_argument("--max-size", type=int, default=1333)
    args = p.parse_args()

    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes,
                        min_size=args.min_size,
                        max_size=args.max_size).to(device)
    load_pretrained(model, args.weights)
    model.eval()
    img = Image.open(args.image).convert("RGB")
    x = pil_to_tensor(img).to(device)
    with torch.no_grad():
        det = model([x])[0]
    keep = det["scores"] > args.score_thresh
    boxes = det["boxes"][keep].cpu()
    scores = det["scores"][keep].cpu()
    labels = det["labels"][keep].cpu()
    print(f"{len(boxes)} detection(s) above score {args.score_thresh}")
    for b, s, l in zip(boxes.tolist(), scores.tolist(), labels.tolist()):
        print(f"class {l}  score {s:.3f}  box "
              f"[{b[0]:.1f}, {b[1]:.1f}, {b[2]:.1f}, {b[3]:.1f}]")
    if args.save:
        draw = ImageDraw.Draw(img)
        for b, l in zip(boxes.tolist(), labels.tolist()):
            draw.rectangle(b, outline="red", width=2)
            draw.text((b[0], max(b[1] - 10, 0)), str(int(l)), fill="red")
        img.save(args.save)
        print(f"saved {args.save}")
    return boxes, scores, labels
