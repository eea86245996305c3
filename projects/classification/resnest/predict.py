#!/usr/bin/env python3
"""Single-image inference for resnest
(reference: classification/resnest/predict.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli import predict_main

if __name__ == "__main__":
    predict_main("resnest50", num_classes=1000, img_size=224)
