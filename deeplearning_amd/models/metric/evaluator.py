"""Retrieval evaluation: CMC curve + mean average precision.

Reference parity: metric_learning/BDB/trainers/evaluator.py:52-241 —
re-designed as pure-tensor math (distance matrix -> per-query ranked match
vector -> CMC/AP), identical protocol: same-camera same-id gallery entries
are discarded per query when cam ids are given.
"""
from __future__ import annotations

import torch

from .bdb import pairwise_dist


def cmc_map(query_feats: torch.Tensor, query_ids: torch.Tensor,
            gallery_feats: torch.Tensor, gallery_ids: torch.Tensor,
            query_cams: torch.Tensor | None = None,
            gallery_cams: torch.Tensor | None = None,
            topk: int = 10):
    """Returns (cmc[topk], mAP)."""
    dist = pairwise_dist(query_feats, gallery_feats)
    nq = dist.shape[0]
    cmc = torch.zeros(topk)
    aps = []
    for q in range(nq):
        order = dist[q].argsort()
        matches = gallery_ids[order] == query_ids[q]
        if query_cams is not None and gallery_cams is not None:
            junk = (gallery_ids[order] == query_ids[q]) & \
                (gallery_cams[order] == query_cams[q])
            keep = ~junk
            matches = matches[keep]
        good = matches.nonzero().flatten()
        if good.numel() == 0:
            continue
        first = int(good[0])
        if first < topk:
            cmc[first:] += 1
        # AP
        hits = torch.arange(1, good.numel() + 1, dtype=torch.float32)
        precision = hits / (good.float() + 1)
        aps.append(precision.mean())
    n_valid = max(len(aps), 1)
    return cmc / n_valid, (torch.stack(aps).mean() if aps else torch.tensor(0.0))
