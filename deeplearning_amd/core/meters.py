"""Metric meters: AverageMeter, SmoothedValue, MetricLogger.

Reference parity: AverageMeter (swin utils/torch_utils.py:342), SmoothedValue +
MetricLogger.log_every w/ ETA (RetinaNet train_utils/distributed_utils.py:12-233),
MeterBuffer (YOLOX yolox/utils/metric.py:98-109).
"""
from __future__ import annotations

import datetime
import time
from collections import defaultdict, deque

import torch

from .dist import is_dist, reduce_value


class AverageMeter:
    def __init__(self):
        self.reset()

    def reset(self):
        self.val, self.sum, self.count, self.avg = 0.0, 0.0, 0, 0.0

    def update(self, val: float, n: int = 1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / max(1, self.count)


class SmoothedValue:
    """Track a series, report median/avg over a window + global avg."""

    def __init__(self, window_size: int = 20, fmt: str = "{median:.4f} ({global_avg:.4f})"):
        self.deque: deque = deque(maxlen=window_size)
        self.total = 0.0
        self.count = 0
        self.fmt = fmt

    def update(self, value: float, n: int = 1):
        self.deque.append(value)
        self.count += n
        self.total += value * n

    def synchronize_between_processes(self):
        """All-reduce count/total (reference RetinaNet distributed_utils.py:29-41)."""
        if not is_dist():
            return
        t = torch.tensor([self.count, self.total], dtype=torch.float64)
        if torch.cuda.is_available():
            t = t.cuda()
        t = reduce_value(t, average=False)
        t = t.tolist()
        self.count = int(t[0])
        self.total = t[1]

    @property
    def median(self) -> float:
        return float(torch.tensor(list(self.deque)).median()) if self.deque else 0.0

    @property
    def avg(self) -> float:
        return float(torch.tensor(list(self.deque)).mean()) if self.deque else 0.0

    @property
    def global_avg(self) -> float:
        return self.total / max(1, self.count)

    @property
    def value(self) -> float:
        return self.deque[-1] if self.deque else 0.0

    def __str__(self):
        return self.fmt.format(median=self.median, avg=self.avg,
                               global_avg=self.global_avg, value=self.value)


class MetricLogger:
    def __init__(self, delimiter: str = "  ", logger=None):
        self.meters: dict[str, SmoothedValue] = defaultdict(SmoothedValue)
        self.delimiter = delimiter
        self.logger = logger

    def update(self, **kwargs):
        for k, v in kwargs.items():
            if isinstance(v, torch.Tensor):
                v = v.item()
            self.meters[k].update(float(v))

    def __getattr__(self, attr):
        if attr in self.meters:
            return self.meters[attr]
        raise AttributeError(attr)

    def add_meter(self, name: str, meter: SmoothedValue):
        self.meters[name] = meter

    def synchronize_between_processes(self):
        for m in self.meters.values():
            m.synchronize_between_processes()

    def _print(self, msg: str):
        if self.logger is not None:
            self.logger.info(msg)
        else:
            print(msg, flush=True)

    def log_every(self, iterable, print_freq: int, header: str = ""):
        i = 0
        start = time.time()
        iter_time = SmoothedValue(fmt="{avg:.4f}")
        data_time = SmoothedValue(fmt="{avg:.4f}")
        end = time.time()
        n = len(iterable) if hasattr(iterable, "__len__") else None
        for obj in iterable:
            data_time.update(time.time() - end)
            yield obj
            iter_time.update(time.time() - end)
            if i % print_freq == 0 or (n is not None and i == n - 1):
                eta = ""
                if n is not None:
                    eta_sec = iter_time.global_avg * (n - i)
                    eta = f" eta: {datetime.timedelta(seconds=int(eta_sec))}"
                meters = self.delimiter.join(f"{k}: {v}" for k, v in self.meters.items())
                total = f"[{i}/{n}]" if n is not None else f"[{i}]"
                self._print(f"{header} {total}{eta} {meters} "
                            f"time: {iter_time} data: {data_time}")
            i += 1
            end = time.time()
        total_time = time.time() - start
        self._print(f"{header} Total time: {datetime.timedelta(seconds=int(total_time))}")
