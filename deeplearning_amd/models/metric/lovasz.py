"""Lovász-hinge / Lovász-softmax losses (direct IoU surrogate).

Reference parity: metric_learning/Happy-Whale/retrieval/models/lovasz.py —
re-designed from the Lovász extension definition (gradient of the sorted
error vector against cumulative IoU).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F


def lovasz_grad(gt_sorted: torch.Tensor) -> torch.Tensor:
    p = len(gt_sorted)
    gts = gt_sorted.sum()
    intersection = gts - gt_sorted.float().cumsum(0)
    union = gts + (1 - gt_sorted).float().cumsum(0)
    jaccard = 1.0 - intersection / union
    if p > 1:
        jaccard[1:p] = jaccard[1:p] - jaccard[0:-1]
    return jaccard


def lovasz_hinge(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Binary Lovász hinge. logits/labels: flattened [P]."""
    signs = 2.0 * labels.float() - 1.0
    errors = 1.0 - logits * signs
    errors_sorted, perm = torch.sort(errors, dim=0, descending=True)
    grad = lovasz_grad(labels[perm])
    return torch.dot(F.relu(errors_sorted), grad)


def lovasz_softmax(probs: torch.Tensor, labels: torch.Tensor,
                   classes="present") -> torch.Tensor:
    """Multi-class Lovász-softmax. probs: [P, C]; labels: [P]."""
    C = probs.shape[1]
    losses = []
    for c in range(C):
        fg = (labels == c).float()
        if classes == "present" and fg.sum() == 0:
            continue
        errors = (fg - probs[:, c]).abs()
        errors_sorted, perm = torch.sort(errors, 0, descending=True)
        losses.append(torch.dot(errors_sorted, lovasz_grad(fg[perm])))
    if not losses:
        return probs.sum() * 0.0
    return torch.stack(losses).mean()
