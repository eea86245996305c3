"""Samplers: infinite, yolo-batch (mosaic flag), aspect-ratio grouped batch.

Reference parity: detection/YOLOX/yolox/data/samplers.py:14-84
(InfiniteSampler, YoloBatchSampler),
detection/fasterRcnn/utils/group_by_aspect_ratio.py (GroupedBatchSampler),
swin dataLoader/samplers.py (SubsetRandomSampler w/ set_epoch).
"""
from __future__ import annotations

import itertools
from collections import defaultdict

import torch
from torch.utils.data.sampler import BatchSampler, Sampler


class InfiniteSampler(Sampler):
    """Shuffled infinite index stream, rank-sharded (ref samplers.py:14-50)."""

    def __init__(self, size: int, shuffle: bool = True, seed: int = 0,
                 rank: int = 0, world_size: int = 1):
        self._size = size
        self._shuffle = shuffle
        self._seed = seed
        self._rank = rank
        self._world = world_size

    def __iter__(self):
        yield from itertools.islice(self._infinite(), self._rank, None,
                                    self._world)

    def _infinite(self):
        g = torch.Generator()
        g.manual_seed(self._seed)
        while True:
            if self._shuffle:
                yield from torch.randperm(self._size, generator=g).tolist()
            else:
                yield from range(self._size)

    def __len__(self):
        return self._size // self._world


class YoloBatchSampler(BatchSampler):
    """BatchSampler whose batches carry a mosaic on/off flag
    (ref samplers.py:52-84)."""

    def __init__(self, *args, mosaic: bool = True, **kw):
        super().__init__(*args, **kw)
        self.mosaic = mosaic

    def __iter__(self):
        for batch in super().__iter__():
            yield [(self.mosaic, idx) for idx in batch]


class SubsetRandomSampler(Sampler):
    """Epoch-seeded shuffled subset (ref swin dataLoader/samplers.py:11-28)."""

    def __init__(self, indices):
        self.indices = list(indices)
        self.epoch = 0

    def __iter__(self):
        g = torch.Generator()
        g.manual_seed(self.epoch)
        perm = torch.randperm(len(self.indices), generator=g)
        return (self.indices[i] for i in perm.tolist())

    def __len__(self):
        return len(self.indices)

    def set_epoch(self, epoch: int):
        self.epoch = epoch


def compute_aspect_ratios(dataset):
    ratios = []
    for i in range(len(dataset)):
        if hasattr(dataset, "get_height_and_width"):
            h, w = dataset.get_height_and_width(i)
        else:
            img = dataset[i][0]
            h, w = img.shape[-2:]
        ratios.append(w / h)
    return ratios


class GroupedBatchSampler(BatchSampler):
    """Batches drawn from the same aspect-ratio group so pad waste stays low
    (ref fasterRcnn utils/group_by_aspect_ratio.py)."""

    def __init__(self, sampler, group_ids, batch_size):
        self.sampler = sampler
        self.group_ids = group_ids
        self.batch_size = batch_size

    def __iter__(self):
        buffers = defaultdict(list)
        for idx in self.sampler:
            g = self.group_ids[idx]
            buffers[g].append(idx)
            if len(buffers[g]) == self.batch_size:
                yield buffers[g]
                buffers[g] = []
        # flush leftovers (repeat to fill, keeping batch size constant)
        for g, buf in buffers.items():
            if buf:
                while len(buf) < self.batch_size:
                    buf.append(buf[-1])
                yield buf

    def __len__(self):
        # __iter__ yields full batches per group plus one padded flush batch
        # for every group with a remainder, so count from the group-id
        # distribution (len(sampler)//batch_size under-counts and schedulers
        # sized from len(loader) would step past their total).
        counts = defaultdict(int)
        for idx in self.sampler:
            counts[self.group_ids[idx]] += 1
        return sum((n + self.batch_size - 1) // self.batch_size
                   for n in counts.values())


def create_aspect_ratio_groups(aspect_ratios, k=3):
    """Quantize ratios into 2k+1 buckets (fbnet-style grouping)."""
    import bisect
    bins = (2 ** torch.linspace(-1, 1, 2 * k + 1)).tolist() if k > 0 else [1.0]
    return [bisect.bisect_right(bins, r) for r in aspect_ratios]
