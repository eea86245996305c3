"""Synthetic datasets for benchmarking / testing (no network, no real data).

DeviceBatchLoader serves pre-generated device-resident batches — the hot-loop
feeding style the bench uses ("data": "synthetic" in bench.py output).
"""
from __future__ import annotations

import torch
from torch.utils.data import Dataset


class SyntheticClassification(Dataset):
    """Random images + labels with fixed seed (deterministic per index)."""

    def __init__(self, length=1024, image_size=(3, 224, 224), num_classes=1000,
                 seed=0):
        self.length = length
        self.image_size = tuple(image_size)
        self.num_classes = num_classes
        self.seed = seed

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        img = torch.randn(self.image_size, generator=g)
        label = torch.randint(0, self.num_classes, (1,), generator=g).item()
        return img, label


class DeviceBatchLoader:
    """Iterates `steps` pre-built device batches (rotating over `distinct`
    actual tensors to defeat trivial caching)."""

    def __init__(self, batch_size, image_size, num_classes, steps, device,
                 dtype=torch.float32, distinct=2, seed=0, channels_last=False):
        g = torch.Generator(device="cpu").manual_seed(seed)
        self.batches = []
        for _ in range(min(distinct, steps) or 1):
            x = torch.randn(batch_size, *image_size, generator=g).to(device=device, dtype=dtype)
            if channels_last and x.dim() == 4:
                x = x.contiguous(memory_format=torch.channels_last)
            y = torch.randint(0, num_classes, (batch_size,), generator=g).to(device)
            self.batches.append((x, y))
        self.steps = steps

    def __len__(self):
        return self.steps

    def __iter__(self):
        for i in range(self.steps):
            yield self.batches[i % len(self.batches)]
