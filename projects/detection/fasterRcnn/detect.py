#!/usr/bin/env python3
"""Single-image detection for fasterRcnn (reference: detection/fasterRcnn predict/detect)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli_det import detect_main

if __name__ == "__main__":
    detect_main("fasterrcnn_resnet50_fpn", num_classes=21)
