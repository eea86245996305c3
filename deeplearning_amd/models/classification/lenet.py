"""MNIST nets — the canonical minimal subproject (classification/mnist
models/network.py:7 mnist_cnn, :34 mnist_fcn)."""
from __future__ import annotations

import torch
import torch.nn as nn

from ..registry import register_model


@register_model
def mnist_cnn(num_classes=10, in_chans=1, **kw):
    """LeNet-style CNN: two conv+pool stages then FC head."""
    return nn.Sequential(
        nn.Conv2d(in_chans, 32, 3, padding=1), nn.ReLU(inplace=True),
        nn.MaxPool2d(2),
        nn.Conv2d(32, 64, 3, padding=1), nn.ReLU(inplace=True),
        nn.MaxPool2d(2),
        nn.Flatten(),
        nn.Linear(64 * 7 * 7, 128), nn.ReLU(inplace=True),
        nn.Dropout(0.25),
        nn.Linear(128, num_classes),
    )


class MnistFCN(nn.Module):
    """All-convolutional MNIST net (global pooling head)."""

    def __init__(self, num_classes=10, in_chans=1):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(in_chans, 32, 3, padding=1), nn.BatchNorm2d(32), nn.ReLU(inplace=True),
            nn.MaxPool2d(2),
            nn.Conv2d(32, 64, 3, padding=1), nn.BatchNorm2d(64), nn.ReLU(inplace=True),
            nn.MaxPool2d(2),
            nn.Conv2d(64, num_classes, 1),
        )
        self.pool = nn.AdaptiveAvgPool2d(1)

    def forward(self, x):
        return torch.flatten(self.pool(self.features(x)), 1)


@register_model
def mnist_fcn(num_classes=10, in_chans=1, **kw):
    return MnistFCN(num_classes=num_classes, in_chans=in_chans)
