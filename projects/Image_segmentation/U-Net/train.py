#!/usr/bin/env python3
"""Train U-Net (reference: Image_segmentation/U-Net/train.py)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli_seg import seg_argparser, seg_train_main

if __name__ == "__main__":
    args = seg_argparser("unet", num_classes=2, name="U-Net", dice=True).parse_args()
    seg_train_main(args)
