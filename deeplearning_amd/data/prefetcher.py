"""DataPrefetcher: overlap H2D copies with compute on a side HIP stream.

Reference parity: detection/YOLOX/yolox/data/data_prefetcher.py:8-50 —
re-designed for ROCm: copies ride a dedicated HIP stream; the compute stream
waits on a recorded event (not a blanket stream sync), and tensors are pinned
once at the loader.
"""
from __future__ import annotations

import torch


class DataPrefetcher:
    def __init__(self, loader, device=None):
        self.loader = iter(loader)
        self.device = device or torch.device("cuda")
        self.stream = torch.cuda.Stream()
        self.next_batch = None
        self.event = torch.cuda.Event()
        self.preload()

    def _to_device(self, x):
        if torch.is_tensor(x):
            return x.to(self.device, non_blocking=True)
        if isinstance(x, tuple) and hasattr(x, "_make"):  # namedtuple
            return type(x)._make(self._to_device(v) for v in x)
        if isinstance(x, (list, tuple)):
            return type(x)(self._to_device(v) for v in x)
        if isinstance(x, dict):
            return {k: self._to_device(v) for k, v in x.items()}
        return x

    def preload(self):
        try:
            batch = next(self.loader)
        except StopIteration:
            self.next_batch = None
            return
        with torch.cuda.stream(self.stream):
            self.next_batch = self._to_device(batch)
            self.event.record(self.stream)

    def _record_stream(self, x, stream):
        # Every copied tensor was allocated on the side stream but is consumed
        # on the compute stream; without record_stream the caching allocator
        # may hand its memory to the next preload's H2D copy while compute is
        # still reading it (mirror _to_device's recursion over nested batches).
        if torch.is_tensor(x):
            if x.is_cuda:
                x.record_stream(stream)
        elif isinstance(x, (list, tuple)):
            for v in x:
                self._record_stream(v, stream)
        elif isinstance(x, dict):
            for v in x.values():
                self._record_stream(v, stream)

    def next(self):
        if self.next_batch is None:
            return None
        torch.cuda.current_stream().wait_event(self.event)
        batch = self.next_batch
        # keep the copied tensors alive until the compute stream used them
        self._record_stream(batch, torch.cuda.current_stream())
        self.preload()
        return batch

    def __iter__(self):
        while True:
            b = self.next()
            if b is None:
                return
            yield b
