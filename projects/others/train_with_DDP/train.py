#!/usr/bin/env python3
"""Canonical DDP training recipe (reference: others/train_with_DDP/train.py:82-313 — zero-first barrier, init-weight sync, SyncBN, lr x WORLD_SIZE, rank-0 eval/save). Launch: torchrun --nproc-per-node N train.py"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

from deeplearning_amd.engine.cli import (classification_argparser,
                                         classification_train_main)
from deeplearning_amd.core.dist import get_world_size

if __name__ == "__main__":
    args = classification_argparser("resnet50", num_classes=100,
                                    img_size=64, name="ddp",
                                    batch_size=16).parse_args()
    args.lr = args.lr * max(get_world_size(), 1)  # linear scaling (ref :199)
    args.syncbn = True
    classification_train_main(args)
