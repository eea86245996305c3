"""Config-as-code Exp classes (YOLOX style).

Reference parity: detection/YOLOX/yolox/exp/base_exp.py:17-63 (BaseExp with
merge(opts)) and yolox_base.py:15-286 (Exp with model/data/optimizer
factories) — re-designed on this repo's registry/engine.
"""
from __future__ import annotations

import ast
import importlib.util
from pathlib import Path


class BaseExp:
    """Subclass and override attributes; `merge` applies CLI KV overrides."""

    seed = 0
    output_dir = "runs"
    print_interval = 10
    eval_interval = 1

    def merge(self, opts):
        """opts: flat [k, v, k, v, ...] list; values parsed as literals."""
        assert len(opts) % 2 == 0, "opts must be key-value pairs"
        for k, v in zip(opts[0::2], opts[1::2]):
            if not hasattr(self, k):
                raise AttributeError(f"Exp has no attribute '{k}'")
            old = getattr(self, k)
            try:
                v = ast.literal_eval(v)
            except (ValueError, SyntaxError):
                pass
            if old is not None and not isinstance(v, type(old)):
                v = type(old)(v)
            setattr(self, k, v)
        return self

    def __repr__(self):
        rows = [f"  {k} = {getattr(self, k)!r}"
                for k in sorted(dir(self))
                if not k.startswith("_") and
                not callable(getattr(self, k))]
        return f"{type(self).__name__}(\n" + "\n".join(rows) + "\n)"


class YoloxExp(BaseExp):
    """Default YOLOX experiment (ref yolox_base.py Exp)."""

    model_name = "yolox_s"
    num_classes = 80
    depth = 0.33
    width = 0.50
    input_size = (640, 640)
    basic_lr_per_img = 0.01 / 64.0
    max_epoch = 300
    no_aug_epochs = 15
    warmup_epochs = 5
    momentum = 0.9
    weight_decay = 5e-4
    ema = True
    mosaic = True
    test_conf = 0.01
    nms_thre = 0.65

    def get_model(self):
        from ..models import build_model

        return build_model(self.model_name, num_classes=self.num_classes)

    def get_optimizer(self, model, batch_size):
        import torch

        lr = self.basic_lr_per_img * batch_size
        # BN/bias without weight decay (ref yolox_base.py get_optimizer)
        decay, no_decay = [], []
        for name, p in model.named_parameters():
            if not p.requires_grad:
                continue
            (no_decay if p.ndim <= 1 or name.endswith(".bias")
             else decay).append(p)
        return torch.optim.SGD(
            [{"params": decay, "weight_decay": self.weight_decay},
             {"params": no_decay, "weight_decay": 0.0}],
            lr=lr, momentum=self.momentum, nesterov=True)

    def get_data_loader(self, batch_size, synthetic_size=16):
        from torch.utils.data import DataLoader

        from ..data import MosaicDetection
        from .cli_det import SyntheticDetection

        ds = SyntheticDetection(synthetic_size,
                                (3, *self.input_size), self.num_classes)
        if self.mosaic:
            ds = MosaicDetection(ds, out_size=self.input_size[0])
        return DataLoader(ds, batch_size=batch_size, shuffle=True,
                          collate_fn=SyntheticDetection.collate_fn)

    def close_mosaic_if_due(self, loader, epoch) -> bool:
        """Disable mosaic for the last `no_aug_epochs` epochs (ref
        yolox/core/trainer.py before_epoch: close_mosaic + L1 switch).
        Returns True when augmentation is off for this epoch."""
        off = epoch >= self.max_epoch - self.no_aug_epochs
        ds = getattr(loader, "dataset", None)
        if off and hasattr(ds, "enabled"):
            ds.enabled = False
        return off


def get_exp(exp_file: str | None = None, exp_name: str | None = None):
    """Load an Exp from a python file (ref yolox/exp/build.py get_exp)."""
    if exp_file:
        spec = importlib.util.spec_from_file_location(
            Path(exp_file).stem, exp_file)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        return mod.Exp()
    table = {"yolox_s": dict(depth=0.33, width=0.50),
             "yolox_m": dict(depth=0.67, width=0.75),
             "yolox_l": dict(depth=1.0, width=1.0),
             "yolox_x": dict(depth=1.33, width=1.25)}
    exp = YoloxExp()
    if exp_name:
        assert exp_name in table, f"unknown exp '{exp_name}'"
        exp.model_name = exp_name
        for k, v in table[exp_name].items():
            setattr(exp, k, v)
    return exp
