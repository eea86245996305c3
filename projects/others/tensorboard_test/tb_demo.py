#!/usr/bin/env python3
"""SummaryWriter walkthrough (reference: others/tensorboard_test/) — works with or without the tensorboard package (JSONL fallback)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import math

from deeplearning_amd.core.tensorboard import SummaryWriter

if __name__ == "__main__":
    writer = SummaryWriter("/tmp/tb_demo")
    for step in range(100):
        writer.add_scalar("demo/sine", math.sin(step / 10), step)
        writer.add_scalar("demo/lr", 0.1 * 0.99 ** step, step)
    writer.add_text("demo/info", "hello from MI355X")
    writer.close()
    print("wrote scalars to /tmp/tb_demo")
