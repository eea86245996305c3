#!/usr/bin/env python3
"""Single-image segmentation for HR-Net-Seg
(reference: Image_segmentation/HR-Net-Seg predict)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli_seg import seg_predict_main

if __name__ == "__main__":
    seg_predict_main("hrnet_w18_seg", num_classes=19)
