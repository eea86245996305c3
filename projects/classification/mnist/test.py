#!/usr/bin/env python3
"""Dataset evaluation for mnist (reference: classification/mnist/test.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli import evaluate_main

if __name__ == "__main__":
    evaluate_main("mnist_cnn", num_classes=10, img_size=28, in_channels=1)
