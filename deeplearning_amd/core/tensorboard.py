"""SummaryWriter: TensorBoard-compatible scalar logging with a JSONL/CSV
fallback when the tensorboard package is absent (it is not in this image).

Reference parity: the SummaryWriter usage in nearly every subproject
(SURVEY.md §5 Metrics/logging — mnist/train.py:29,149-156, U-Net, swin).
API subset: add_scalar, add_scalars, add_text, add_image (no-op in fallback),
flush, close.
"""
from __future__ import annotations

import json
import time
from pathlib import Path

try:
    from torch.utils.tensorboard import SummaryWriter as _TBWriter
    _HAS_TB = True
except Exception:
    _HAS_TB = False


class SummaryWriter:
    def __init__(self, log_dir: str = "runs"):
        self.log_dir = Path(log_dir)
        self.log_dir.mkdir(parents=True, exist_ok=True)
        self._tb = _TBWriter(str(log_dir)) if _HAS_TB else None
        self._f = None
        if self._tb is None:
            self._f = open(self.log_dir / "scalars.jsonl", "a")

    def add_scalar(self, tag, value, step=None, walltime=None):
        if self._tb:
            self._tb.add_scalar(tag, value, step, walltime)
        else:
            self._f.write(json.dumps({
                "tag": tag, "value": float(value), "step": step,
                "time": walltime or time.time()}) + "\n")

    def add_scalars(self, main_tag, tag_scalar_dict, step=None):
        for k, v in tag_scalar_dict.items():
            self.add_scalar(f"{main_tag}/{k}", v, step)

    def add_text(self, tag, text, step=None):
        if self._tb:
            self._tb.add_text(tag, text, step)
        else:
            self._f.write(json.dumps({"tag": tag, "text": text,
                                      "step": step}) + "\n")

    def add_image(self, *a, **kw):
        if self._tb:
            self._tb.add_image(*a, **kw)

    def add_histogram(self, *a, **kw):
        if self._tb:
            self._tb.add_histogram(*a, **kw)

    def add_graph(self, *a, **kw):
        if self._tb:
            self._tb.add_graph(*a, **kw)

    def flush(self):
        (self._tb.flush() if self._tb else self._f.flush())

    def close(self):
        if self._tb:
            self._tb.close()
        if self._f:
            self._f.close()
