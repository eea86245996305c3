#!/usr/bin/env python3
"""Partial/pretrained weight loading walkthrough (reference: others/load_weights_test/)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))


from deeplearning_amd.core.checkpoint import (load_pretrained, save_weights,
                                              strip_module_prefix)
from deeplearning_amd.models import build_model

if __name__ == "__main__":
    src = build_model("resnet18", num_classes=1000)
    save_weights(src, "/tmp/r18_1000.pth")
    # different head: fc dropped, trunk loaded
    dst = build_model("resnet18", num_classes=5)
    dropped = load_pretrained(dst, "/tmp/r18_1000.pth")
    print(f"dropped keys (head mismatch): {dropped}")
    assert any("fc" in k for k in dropped)
    # module.-prefixed dict round trip
    pref = {"module." + k: v for k, v in src.state_dict().items()}
    clean = strip_module_prefix(pref)
    dst2 = build_model("resnet18", num_classes=1000)
    dst2.load_state_dict(clean)
    print("module.-prefix strip OK")
