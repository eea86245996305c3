"""Shared CLI drivers for the per-subproject train.py / predict.py scripts
under projects/.

Each reference subproject ships its own full train loop;
here every projects/*/train.py is a thin wrapper over these drivers, keeping
the reference CLI surface (--data-path, --epochs, --batch-size, --lr,
--device, --weights, --resume, --amp ...) and the reference checkpoint layout
runs/<name>/weights/model_{e}.pth + best_model.pth
(ref classification/mnist/train.py:141-186, others/train_with_DDP/train.py).
"""
from __future__ import annotations

import argparse
import time
from pathlib import Path

import torch
from torch.utils.data import DataLoader, DistributedSampler

from ..core.checkpoint import (load_checkpoint, load_pretrained,
                               save_checkpoint, save_weights)
from ..core.dist import (cleanup, get_rank, get_world_size, init_distributed,
                         is_main_process)
from ..core.env import increment_path, seed_everything, select_device
from ..core.logging import create_logger
from ..core.meters import AverageMeter
from ..core.tensorboard import SummaryWriter
from ..data import (ClassificationDataset, SyntheticClassification,
                    classification_eval_transform,
                    classification_train_transform, read_split_data)
from ..engine.metrics import accuracy
from ..engine.scheduler import WarmupScheduler
from ..models import build_model
from ..ops import cross_entropy
from ..parallel import wrap_data_parallel


def classification_argparser(default_model: str, **defaults):
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=default_model)
    p.add_argument("--data-path", default="", help="class-per-folder root; "
                   "empty = synthetic data")
    p.add_argument("--num-classes", type=int,
                   default=defaults.get("num_classes", 1000))
    p.add_argument("--img-size", type=int,
                   default=defaults.get("img_size", 224))
    p.add_argument("--in-channels", type=int,
                   default=defaults.get("in_channels", 3))
    p.add_argument("--epochs", type=int, default=defaults.get("epochs", 10))
    p.add_argument("--batch-size", type=int,
                   default=defaults.get("batch_size", 32))
    p.add_argument("--lr", type=float, default=defaults.get("lr", 0.01))
    p.add_argument("--weight-decay", type=float,
                   default=defaults.get("weight_decay", 5e-4))
    p.add_argument("--optimizer", default=defaults.get("optimizer", "sgd"),
                   choices=["sgd", "adamw"])
    p.add_argument("--warmup-epochs", type=int, default=1)
    p.add_argument("--workers", type=int, default=4)
    p.add_argument("--device", default="cuda")
    p.add_argument("--weights", default="", help="pretrained weights")
    p.add_argument("--resume", default="", help="checkpoint to resume, or "
                   "'auto'")
    p.add_argument("--amp", action="store_true", default=True)
    p.add_argument("--no-amp", dest="amp", action="store_false")
    p.add_argument("--amp-dtype", default="bf16", choices=["bf16", "fp16"],
                   help="fp16 adds a GradScaler (reference NativeScaler "
                        "semantics, swin utils/torch_utils.py:297-323)")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--deterministic", action="store_true",
                   help="cudnn.deterministic + fixed seeds "
                        "(ref FCOS trainers/trainer.py:57-66)")
    p.add_argument("--output", default="runs")
    p.add_argument("--name", default=defaults.get("name", "exp"))
    p.add_argument("--syncbn", action="store_true")
    p.add_argument("--rand-augment", action="store_true",
                   help="RandAugment rand-m9-mstd0.5 on the train pipeline "
                        "(swin recipe; data/autoaugment.py)")
    p.add_argument("--ra-magnitude", type=float, default=9.0)
    p.add_argument("--mixup", action="store_true",
                   help="Mixup/CutMix soft-target training "
                        "(swin dataLoader/build.py:90-96)")
    p.add_argument("--label-smoothing", type=float, default=0.0)
    p.add_argument("--accumulate-steps", type=int, default=1)
    p.add_argument("--clip-grad", type=float, default=0.0)
    p.add_argument("--synthetic-size", type=int, default=256,
                   help="synthetic dataset length when --data-path is empty")
    p.add_argument("--throughput", action="store_true",
                   help="50 warmup + 30 timed forward passes, then exit "
                        "(ref swin main.py:280-297)")
    p.add_argument("--cfg", default="", help="YAML config (CfgNode, supports "
                   "_BASE_ inheritance); keys override CLI defaults")
    p.add_argument("--opts", nargs="*", default=None,
                   help="dotted-key config overrides, e.g. train.lr 0.1")
    return p


def build_classification_loaders(args):
    if args.data_path:
        tp, tl, vp, vl, classes = read_split_data(args.data_path)
        train_ds = ClassificationDataset(
            tp, tl, classification_train_transform(
                args.img_size,
                rand_augment=getattr(args, "rand_augment", False),
                ra_magnitude=getattr(args, "ra_magnitude", 9.0)))
        val_ds = ClassificationDataset(
            vp, vl, classification_eval_transform(args.img_size))
    else:
        c = getattr(args, "in_channels", 3)
        train_ds = SyntheticClassification(
            args.synthetic_size, (c, args.img_size, args.img_size),
            args.num_classes)
        val_ds = SyntheticClassification(
            max(args.synthetic_size // 4, 8),
            (c, args.img_size, args.img_size), args.num_classes)
    train_sampler = DistributedSampler(train_ds) if get_world_size() > 1 \
        else None
    train_loader = DataLoader(
        train_ds, batch_size=args.batch_size,
        shuffle=train_sampler is None, sampler=train_sampler,
        num_workers=args.workers, pin_memory=torch.cuda.is_available(),
        drop_last=True)
    val_loader = DataLoader(val_ds, batch_size=args.batch_size,
                            shuffle=False, num_workers=args.workers)
    return train_loader, val_loader, train_sampler


def apply_cfg(args):
    """Merge --cfg YAML + --opts into args (one config system for all
    subprojects; ref swin yacs config.py semantics)."""
    if not getattr(args, "cfg", "") and not getattr(args, "opts", None):
        return args
    from ..core.config import load_config
    defaults = {k.replace("-", "_"): v for k, v in vars(args).items()}
    cfg = load_config(defaults, args.cfg or None, args.opts, freeze=False)
    for k, v in cfg.items():
        if k not in ("cfg", "opts"):
            setattr(args, k, v)
    return args


def classification_train_main(args) -> dict:
    args = apply_cfg(args)
    if getattr(args, "throughput", False):
        return throughput_main(args)
    info = init_distributed()
    if torch.cuda.is_available() and get_world_size() > 1:
        device = torch.device("cuda", info["local_rank"])
        torch.cuda.set_device(device)
    else:
        device = select_device(args.device)
    seed_everything(args.seed, rank=get_rank(),
                    deterministic=getattr(args, 'deterministic', False))

    run_dir = Path(increment_path(Path(args.output) / args.name,
                                  exist_ok=False)) \
        if is_main_process() else Path(args.output) / args.name
    weights_dir = run_dir / "weights"
    logger = create_logger(str(run_dir) if is_main_process() else None,
                           dist_rank=get_rank())
    writer = SummaryWriter(str(run_dir)) if is_main_process() else None
    if is_main_process():  # dump the resolved config (ref swin main.py:349-353)
        import json
        with open(run_dir / "config.json", "w") as f:
            json.dump({k: v for k, v in vars(args).items()
                       if isinstance(v, (int, float, str, bool, list,
                                         type(None)))}, f, indent=2)

    try:  # size-dependent models (swin, vit, transfg) take img_size
        model = build_model(args.model, num_classes=args.num_classes,
                            img_size=args.img_size)
    except TypeError:
        model = build_model(args.model, num_classes=args.num_classes)
    from ..core.complexity import complexity_str
    logger.info(f"model {args.model}: "
                f"{complexity_str(model, torch.randn(1, 3, args.img_size, args.img_size))}")
    model = model.to(device)
    if args.weights:
        load_pretrained(model, args.weights, logger=logger)
    if args.syncbn and get_world_size() > 1:
        from ..parallel.syncbn import convert_sync_batchnorm
        model = convert_sync_batchnorm(model)
    if get_world_size() > 1:
        model = wrap_data_parallel(model)

    # norm scales / biases / model-declared keys train without weight decay
    # (ref swin utils/optimizer.py set_weight_decay)
    from .optim_groups import param_groups_weight_decay
    params = param_groups_weight_decay(model, args.weight_decay)
    if args.optimizer == "adamw":
        optimizer = torch.optim.AdamW(params, lr=args.lr,
                                      weight_decay=args.weight_decay)
    else:
        optimizer = torch.optim.SGD(params, lr=args.lr, momentum=0.9,
                                    weight_decay=args.weight_decay)

    train_loader, val_loader, train_sampler = \
        build_classification_loaders(args)
    steps_per_epoch = len(train_loader)
    scheduler = WarmupScheduler(
        optimizer, total_steps=args.epochs * steps_per_epoch,
        warmup_steps=args.warmup_epochs * steps_per_epoch)

    start_epoch = 0
    best_acc = 0.0
    resume = args.resume
    if resume == "auto":
        from ..core.checkpoint import auto_resume_helper
        resume = auto_resume_helper(weights_dir) or ""
    if resume:
        ckpt = load_checkpoint(resume, model, optimizer, scheduler)
        start_epoch = ckpt.get("epoch", -1) + 1
        best_acc = ckpt.get("max_accuracy", 0.0)
        logger.info(f"resumed from {resume} at epoch {start_epoch}")

    mixup_fn = None
    if getattr(args, "mixup", False):
        from ..data import Mixup
        mixup_fn = Mixup(num_classes=args.num_classes,
                         label_smoothing=getattr(args, "label_smoothing", 0.1))
    smoothing = getattr(args, "label_smoothing", 0.0)

    amp = args.amp and device.type == "cuda"
    amp_dtype = torch.float16 if getattr(args, "amp_dtype", "bf16") == "fp16" \
        else torch.bfloat16
    # fp16 needs a loss scaler (reference NativeScalerWithGradNormCount);
    # bf16 has the dynamic range to train unscaled
    scaler = torch.amp.GradScaler(
        "cuda", enabled=amp and amp_dtype == torch.float16)
    for epoch in range(start_epoch, args.epochs):
        if train_sampler is not None:
            train_sampler.set_epoch(epoch)
        model.train()
        loss_m, acc_m = AverageMeter(), AverageMeter()
        t0 = time.time()
        optimizer.zero_grad(set_to_none=True)
        for it, (x, y) in enumerate(train_loader):
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            y_hard = y
            if mixup_fn is not None:
                x, y = mixup_fn(x, y)  # y becomes soft [B, C]
            with torch.autocast(device.type, dtype=amp_dtype, enabled=amp):
                out = model(x)
                logits = out[0] if isinstance(out, tuple) else out
                if mixup_fn is not None:
                    from ..ops import soft_target_cross_entropy
                    loss = soft_target_cross_entropy(logits, y)
                else:
                    loss = cross_entropy(logits, y, smoothing=smoothing)
                if isinstance(out, tuple):  # aux heads (GoogLeNet)
                    for aux in out[1:]:
                        if aux is not None:
                            loss = loss + 0.3 * cross_entropy(aux, y_hard)
            scaler.scale(loss / args.accumulate_steps).backward()
            if (it + 1) % args.accumulate_steps == 0:
                # sync FIRST: finalize() writes the all-reduced averaged
                # grads into p.grad — clipping before it would be overwritten
                finalize = getattr(model, "finalize", None)
                if finalize is not None:
                    finalize()
                if args.clip_grad > 0:
                    scaler.unscale_(optimizer)
                    torch.nn.utils.clip_grad_norm_(model.parameters(),
                                                   args.clip_grad)
                scaler.step(optimizer)
                scaler.update()
                optimizer.zero_grad(set_to_none=True)
                scheduler.step()
            with torch.no_grad():
                acc1 = accuracy(logits.float(), y_hard)[0]
            loss_m.update(float(loss.detach()), x.shape[0])
            acc_m.update(float(acc1), x.shape[0])
        logger.info(f"epoch {epoch}: loss {loss_m.avg:.4f} "
                    f"acc {acc_m.avg:.2f} ({time.time() - t0:.1f}s)")

        # eval + checkpoint (rank 0)
        val_acc = evaluate_classification(model, val_loader, device, amp)
        if is_main_process():
            writer.add_scalar("train/loss", loss_m.avg, epoch)
            writer.add_scalar("val/acc1", val_acc, epoch)
            save_weights(model, weights_dir / f"model_{epoch}.pth")
            # update best BEFORE writing the checkpoint so a resumed run
            # sees this epoch's accuracy as max_accuracy
            if val_acc >= best_acc:
                best_acc = val_acc
                save_weights(model, weights_dir / "best_model.pth")
            save_checkpoint(weights_dir / f"ckpt_epoch_{epoch}.pth", model,
                            optimizer, scheduler, epoch,
                            max_accuracy=best_acc)
        else:
            best_acc = max(best_acc, val_acc)
        logger.info(f"epoch {epoch}: val acc1 {val_acc:.2f} "
                    f"(best {best_acc:.2f})")
    if writer:
        writer.close()
    cleanup()
    return {"best_acc": best_acc, "run_dir": str(run_dir)}


@torch.no_grad()
def evaluate_classification(model, loader, device, amp=True) -> float:
    model.eval()
    correct = torch.zeros(2, device=device)
    for x, y in loader:
        x = x.to(device, non_blocking=True)
        y = y.to(device, non_blocking=True)
        with torch.autocast(device.type, dtype=torch.bfloat16, enabled=amp):
            logits = model(x)
        pred = logits.float().argmax(1)
        correct[0] += (pred == y).sum()
        correct[1] += y.numel()
    from ..core.dist import reduce_value
    correct = reduce_value(correct, average=False)
    return float(correct[0] / correct[1].clamp(min=1) * 100)


def predict_main(default_model: str, num_classes: int = 1000,
                 img_size: int = 224):
    """Single-image predict CLI (ref classification/*/predict.py:12-59)."""
    p = argparse.ArgumentParser()
    p.add_argument("image", help="path to image")
    p.add_argument("--model", default=default_model)
    p.add_argument("--weights", required=True)
    p.add_argument("--num-classes", type=int, default=num_classes)
    p.add_argument("--img-size", type=int, default=img_size)
    p.add_argument("--device", default="cuda")
    p.add_argument("--topk", type=int, default=5)
    args = p.parse_args()

    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes).to(device)
    load_pretrained(model, args.weights)
    model.eval()

    from PIL import Image
    img = Image.open(args.image).convert("RGB")
    x = classification_eval_transform(args.img_size)(img)[None].to(device)
    with torch.no_grad():
        prob = model(x).softmax(1)[0]
    topk = prob.topk(min(args.topk, prob.numel()))
    for score, idx in zip(topk.values.tolist(), topk.indices.tolist()):
        print(f"class {idx}: {score:.4f}")
    return topk


@torch.no_grad()
def throughput_main(args) -> dict:
    """Throughput mode: 50 warmup + 30 timed forward passes
    (ref swin main.py:280-297)."""
    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes).to(device)
    model.eval()
    c = getattr(args, "in_channels", 3)
    x = torch.randn(args.batch_size, c, args.img_size, args.img_size,
                    device=device)
    amp = args.amp and device.type == "cuda"
    warmup, iters = (50, 30) if device.type == "cuda" else (3, 5)
    with torch.autocast(device.type, dtype=torch.bfloat16, enabled=amp):
        for _ in range(warmup):
            model(x)
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(iters):
            model(x)
        if device.type == "cuda":
            torch.cuda.synchronize()
    elapsed = time.time() - t0
    ips = args.batch_size * iters / elapsed
    print(f"throughput: {ips:.1f} images/s "
          f"({elapsed / iters * 1000:.2f} ms/batch of {args.batch_size})")
    return {"images_per_sec": ips}


def evaluate_main(default_model: str, num_classes: int = 1000,
                  img_size: int = 224, in_channels: int = 3):
    """Dataset evaluation CLI: load a checkpoint, run the val split, print
    accuracy + confusion matrix (ref classification/*/test.py pattern)."""
    p = argparse.ArgumentParser()
    p.add_argument("--model", default=default_model)
    p.add_argument("--weights", required=True)
    p.add_argument("--data-path", default="", help="empty = synthetic")
    p.add_argument("--num-classes", type=int, default=num_classes)
    p.add_argument("--img-size", type=int, default=img_size)
    p.add_argument("--in-channels", type=int, default=in_channels)
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--device", default="cuda")
    p.add_argument("--workers", type=int, default=2)
    p.add_argument("--synthetic-size", type=int, default=64)
    args = p.parse_args()

    device = select_device(args.device)
    model = build_model(args.model, num_classes=args.num_classes).to(device)
    load_pretrained(model, args.weights)
    model.eval()
    _, val_loader, _ = build_classification_loaders(args)

    from .metrics import ConfusionMatrix
    cm = ConfusionMatrix(args.num_classes)
    correct = top5 = total = 0
    with torch.no_grad():
        for x, y in val_loader:
            x, y = x.to(device), y.to(device)
            with torch.autocast(device.type, dtype=torch.bfloat16,
                                enabled=device.type == "cuda"):
                logits = model(x).float()
            pred = logits.argmax(1)
            correct += int((pred == y).sum())
            k = min(5, logits.shape[1])
            top5 += int((logits.topk(k, dim=1).indices ==
                         y[:, None]).any(1).sum())
            total += y.numel()
            cm.update(y.flatten(), pred.flatten())
    print(f"top1 {100 * correct / max(total, 1):.2f}%  "
          f"top5 {100 * top5 / max(total, 1):.2f}%  ({total} images)")
    print(cm)
    return correct / max(total, 1)
