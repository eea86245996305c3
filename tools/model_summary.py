"""Print params + MACs for registry models (swin `flops()` log parity,
generalized to the whole zoo): python tools/model_summary.py [names...]"""
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from deeplearning_amd.core.complexity import count_params, estimate_macs  # noqa: E402
from deeplearning_amd.models import build_model, list_models  # noqa: E402

SIZES = {"inception_v4": 299, "swinv2_t": 256, "nasnet_a_large": 331,
         "mnist_cnn": 28, "mnist_fcn": 28}


def main(names):
    for n in names:
        try:
            m = build_model(n, num_classes=10)
        except TypeError:
            m = build_model(n)
        size = SIZES.get(n, 224)
        ch = 1 if n.startswith("mnist") else 3
        row = f"{n:28s} params {count_params(m) / 1e6:8.1f}M"
        try:
            macs = estimate_macs(m, torch.randn(1, ch, size, size))
            row += f"  MACs {macs / 1e9:8.2f}G @{size}"
        except Exception as e:
            row += f"  (forward probe n/a: {type(e).__name__})"
        print(row)
        del m


if __name__ == "__main__":
    args = sys.argv[1:]
    main(args if args else sorted(list_models()))
