#!/usr/bin/env python3
"""Dataset evaluation for efficientNet (reference: classification/efficientNet/test.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli import evaluate_main

if __name__ == "__main__":
    evaluate_main("efficientnet_b0", num_classes=1000, img_size=224)
