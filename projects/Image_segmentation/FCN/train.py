#!/usr/bin/env python3
"""Train FCN (reference: Image_segmentation/FCN/train.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli_seg import seg_argparser, seg_train_main

if __name__ == "__main__":
    args = seg_argparser("fcn_resnet50", num_classes=21, name="FCN").parse_args()
    seg_train_main(args)
