"""Training engines.

Three reference styles unified:
 - train_one_epoch / evaluate functions (script style, mnist/U-Net/fasterRcnn)
 - Trainer class with before/after hooks (YOLOX yolox/core/trainer.py:33-329)
 - config-driven loop features: bf16 autocast, grad accumulation, grad clip,
   per-iter LR, auto-resume (swin main.py:84-359)
"""
from __future__ import annotations

import time

import torch
import torch.nn as nn

from ..core import (MetricLogger, SmoothedValue, create_logger, is_main_process)
from ..core.checkpoint import unwrap_model
from .metrics import accuracy


def _autocast(enabled: bool, dtype: torch.dtype):
    if torch.cuda.is_available():
        return torch.autocast("cuda", dtype=dtype, enabled=enabled)
    return torch.autocast("cpu", dtype=torch.bfloat16, enabled=enabled)


def train_one_epoch(model, criterion, data_loader, optimizer, device, epoch,
                    lr_scheduler=None, accum_steps: int = 1, clip_grad: float = 0.0,
                    amp: bool = True, amp_dtype: torch.dtype = torch.bfloat16,
                    ema=None, print_freq: int = 50, logger=None,
                    mixup_fn=None, aux_loss_weight: float = 0.01) -> dict:
    model.train()
    metric = MetricLogger(logger=logger)
    metric.add_meter("lr", SmoothedValue(window_size=1, fmt="{value:.6f}"))
    optimizer.zero_grad(set_to_none=True)
    finalize = getattr(model, "finalize", None)
    no_sync = getattr(model, "no_sync", None)
    # MoE models expose a load-balance aux loss (swin_moe); the reference
    # adds it to the criterion (swin-moe main: cfg TRAIN.MOE.AUX_LOSS_WEIGHT)
    aux_fn = getattr(unwrap_model(model), "aux_loss", None)

    for it, (samples, targets) in enumerate(
            metric.log_every(data_loader, print_freq, f"Epoch [{epoch}]")):
        samples = samples.to(device, non_blocking=True)
        targets = targets.to(device, non_blocking=True)
        if mixup_fn is not None:
            samples, targets = mixup_fn(samples, targets)

        is_accum = (it + 1) % accum_steps != 0
        with _autocast(amp, amp_dtype):
            outputs = model(samples)
            loss = criterion(outputs, targets)
            if aux_fn is not None:
                loss = loss + aux_loss_weight * aux_fn()
            loss = loss / accum_steps
        if is_accum and no_sync is not None:
            with no_sync():
                loss.backward()
        else:
            loss.backward()
        if not is_accum:
            if finalize is not None:
                finalize()
            if clip_grad > 0:
                torch.nn.utils.clip_grad_norm_(model.parameters(), clip_grad)
            optimizer.step()
            optimizer.zero_grad(set_to_none=True)
            if ema is not None:
                ema.update(model)
            if lr_scheduler is not None:
                lr_scheduler.step()
        metric.update(loss=loss.item() * accum_steps)
        metric.meters["lr"].update(optimizer.param_groups[0]["lr"])
    metric.synchronize_between_processes()
    return {k: m.global_avg for k, m in metric.meters.items()}


@torch.no_grad()
def evaluate(model, criterion, data_loader, device, amp: bool = True,
             amp_dtype: torch.dtype = torch.bfloat16, logger=None,
             print_freq: int = 50) -> dict:
    model.eval()
    metric = MetricLogger(logger=logger)
    for samples, targets in metric.log_every(data_loader, print_freq, "Test:"):
        samples = samples.to(device, non_blocking=True)
        targets = targets.to(device, non_blocking=True)
        with _autocast(amp, amp_dtype):
            outputs = model(samples)
            loss = criterion(outputs, targets)
        acc1, acc5 = accuracy(outputs.float(), targets, topk=(1, min(5, outputs.shape[1])))
        n = samples.shape[0]
        metric.update(loss=loss.item())
        metric.meters["acc1"].update(acc1.item(), n=n)
        metric.meters["acc5"].update(acc5.item(), n=n)
    metric.synchronize_between_processes()
    return {k: m.global_avg for k, m in metric.meters.items()}


@torch.no_grad()
def throughput_test(model, batch, warmup: int = 50, iters: int = 30,
                    amp_dtype: torch.dtype = torch.bfloat16) -> float:
    """swin --throughput mode (main.py:280-297): images/sec forward-only."""
    model.eval()
    for _ in range(warmup):
        with _autocast(True, amp_dtype):
            model(batch)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        with _autocast(True, amp_dtype):
            model(batch)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return iters * batch.shape[0] / (time.time() - t0)


class Trainer:
    """Hook-based trainer (YOLOX style): override the before_*/after_* hooks or
    train_one_iter in subclasses; the base class implements classification."""

    def __init__(self, model, optimizer, train_loader, device, max_epoch,
                 criterion=None, lr_scheduler=None, val_loader=None, ema=None,
                 accum_steps=1, clip_grad=0.0, amp=True,
                 amp_dtype=torch.bfloat16, output_dir=None, logger=None,
                 eval_interval=1, save_interval=1, callbacks=None):
        self.model = model
        self.optimizer = optimizer
        self.train_loader = train_loader
        self.val_loader = val_loader
        self.device = device
        self.max_epoch = max_epoch
        self.criterion = criterion or nn.CrossEntropyLoss()
        self.lr_scheduler = lr_scheduler
        self.ema = ema
        self.accum_steps = accum_steps
        self.clip_grad = clip_grad
        self.amp = amp
        self.amp_dtype = amp_dtype
        self.output_dir = output_dir
        self.logger = logger or create_logger(output_dir, 0)
        self.eval_interval = eval_interval
        self.save_interval = save_interval
        self.epoch = 0
        self.best_metric = 0.0
        from .callbacks import Callbacks
        self.callbacks = callbacks or Callbacks()

    # hooks ------------------------------------------------------------
    def before_train(self):
        pass

    def after_train(self):
        pass

    def before_epoch(self):
        if hasattr(self.train_loader, "sampler") and hasattr(
                self.train_loader.sampler, "set_epoch"):
            self.train_loader.sampler.set_epoch(self.epoch)

    def after_epoch(self):
        stats = {}
        if self.val_loader is not None and (self.epoch + 1) % self.eval_interval == 0:
            eval_model = self.ema.ema if self.ema is not None else self.model
            stats = evaluate(eval_model, self.criterion, self.val_loader,
                             self.device, self.amp, self.amp_dtype, self.logger)
            metric = stats.get("acc1", 0.0)
            if metric > self.best_metric:
                self.best_metric = metric
                self._save("best.pth")
        if self.output_dir and is_main_process() and \
                (self.epoch + 1) % self.save_interval == 0:
            self._save(f"ckpt_epoch_{self.epoch}.pth")
        return stats

    def _save(self, name):
        if self.output_dir is None or not is_main_process():
            return
        from ..core.checkpoint import save_checkpoint

        save_checkpoint(f"{self.output_dir}/{name}", self.model, self.optimizer,
                        epoch=self.epoch, ema=self.ema,
                        best_metric=self.best_metric)

    # loop -------------------------------------------------------------
    def train(self):
        self.callbacks.run("on_train_start", self)
        self.before_train()
        try:
            for self.epoch in range(self.epoch, self.max_epoch):
                self.callbacks.run("on_train_epoch_start", self)
                self.before_epoch()
                train_one_epoch(self.model, self.criterion, self.train_loader,
                                self.optimizer, self.device, self.epoch,
                                self.lr_scheduler, self.accum_steps,
                                self.clip_grad, self.amp, self.amp_dtype,
                                self.ema, logger=self.logger)
                stats = self.after_epoch()
                self.callbacks.run("on_fit_epoch_end", self, stats)
        finally:
            self.after_train()
            self.callbacks.run("on_train_end", self)
        return self.best_metric
