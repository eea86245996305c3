#!/usr/bin/env python3
"""Train convNext (reference: classification/convNext/train.py, same CLI surface)
on the shared MI355X engine."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli import (classification_argparser,
                                         classification_train_main)

if __name__ == "__main__":
    args = classification_argparser(
        "convnext_tiny", num_classes=1000, img_size=224, name="convNext", optimizer='adamw', lr=0.004
    ).parse_args()
    classification_train_main(args)
