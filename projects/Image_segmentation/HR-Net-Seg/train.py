#!/usr/bin/env python3
"""Train HR-Net-Seg (reference: Image_segmentation/HR-Net-Seg/train.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli_seg import seg_argparser, seg_train_main

if __name__ == "__main__":
    args = seg_argparser("hrnet_w18_seg", num_classes=19, name="HR-Net-Seg", ohem=True).parse_args()
    seg_train_main(args)
