#!/usr/bin/env python3
"""Single-image detection for RetinaNet (reference: detection/RetinaNet predict/detect)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli_det import detect_main

if __name__ == "__main__":
    detect_main("retinanet_resnet50_fpn", num_classes=21)
