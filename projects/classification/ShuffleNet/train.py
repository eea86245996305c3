#!/usr/bin/env python3
"""Train ShuffleNet (reference: classification/ShuffleNet/train.py, same CLI surface)
on the shared MI355X engine."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli import (classification_argparser,
                                         classification_train_main)

if __name__ == "__main__":
    args = classification_argparser(
        "shufflenet_v2_x1_0", num_classes=1000, img_size=224, name="ShuffleNet"
    ).parse_args()
    classification_train_main(args)
