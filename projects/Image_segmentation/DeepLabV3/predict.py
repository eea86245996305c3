#!/usr/bin/env python3
"""Single-image segmentation for DeepLabV3
(reference: Image_segmentation/DeepLabV3 predict)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli_seg import seg_predict_main

if __name__ == "__main__":
    seg_predict_main("deeplabv3_resnet50", num_classes=21)
