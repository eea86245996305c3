"""Loggers facade: one `log_metrics` fan-out to CSV + TensorBoard(+JSONL
fallback) + wandb (when importable; silently disabled otherwise).

Reference parity: detection/yolov5/utils/loggers/__init__.py:17-55 (the
csv/tb/wandb Loggers facade) — the reference degrades the same way when
wandb is absent.
"""
from __future__ import annotations

import csv
from pathlib import Path

from ..core.tensorboard import SummaryWriter


class Loggers:
    def __init__(self, save_dir, use_csv=True, use_tb=True, use_wandb=True,
                 project="deeplearning_amd", run_name=None):
        self.save_dir = Path(save_dir)
        self.save_dir.mkdir(parents=True, exist_ok=True)
        self.csv_path = self.save_dir / "results.csv" if use_csv else None
        self._csv_keys = None
        self.tb = SummaryWriter(str(self.save_dir)) if use_tb else None
        self.wandb = None
        if use_wandb:
            try:
                import wandb  # not installed in this image: facade degrades

                self.wandb = wandb.init(project=project, name=run_name,
                                        dir=str(self.save_dir))
            except Exception:
                self.wandb = None

    def log_metrics(self, metrics: dict, step: int):
        if self.csv_path is not None:
            keys = ["step"] + sorted(metrics)
            exists = self.csv_path.exists()
            # header on a fresh file or when the metric set changes mid-run;
            # resuming into an existing file appends without re-writing it
            write_header = not exists or (self._csv_keys is not None and
                                          self._csv_keys != keys)
            with open(self.csv_path, "a" if exists else "w", newline="") as f:
                w = csv.writer(f)
                if write_header:
                    w.writerow(keys)
                w.writerow([step] + [metrics[k] for k in sorted(metrics)])
            self._csv_keys = keys
        if self.tb is not None:
            for k, v in metrics.items():
                self.tb.add_scalar(k, v, step)
        if self.wandb is not None:
            self.wandb.log(metrics, step=step)

    def log_images(self, tag: str, images, step: int):
        if self.tb is not None and hasattr(self.tb, "add_image"):
            for i, im in enumerate(images):
                try:
                    self.tb.add_image(f"{tag}/{i}", im, step)
                except Exception:
                    pass

    def close(self):
        if self.tb is not None:
            self.tb.close()
        if self.wandb is not None:
            self.wandb.finish()
