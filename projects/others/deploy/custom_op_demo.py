#!/usr/bin/env python3
"""Custom-op walkthrough: native my_add (3a+2b) + torch.autograd.Function +
TorchScript/ONNX export hook (reference: others/deploy/pytorch2onnx/
{my_add.cpp, support_new_ops.py, support_TorchScript_ops.py})."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[3]))

import torch

from deeplearning_amd.ops import ext


class MyAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        return ext().my_add(a, b)

    @staticmethod
    def backward(ctx, g):
        return 3 * g, 2 * g

    @staticmethod
    def symbolic(g, a, b):  # ONNX export mapping
        return g.op("custom::MyAdd", a, b)


@torch.jit.script
def my_add_torchscript(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Way 3: TorchScript op (ref support_TorchScript_ops.py)."""
    return 3 * a + 2 * b


def my_add_aten(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Way 2: composed from ATen ops only (ref support_Aten_ops.py)."""
    return torch.add(torch.mul(a, 3), torch.mul(b, 2))


if __name__ == "__main__":
    a = torch.randn(4, requires_grad=True)
    b = torch.randn(4, requires_grad=True)
    # way 1: native extension + autograd.Function + ONNX symbolic
    y = MyAdd.apply(a, b)
    assert torch.allclose(y, 3 * a + 2 * b)
    y.sum().backward()
    assert torch.allclose(a.grad, torch.full((4,), 3.0))
    # ways 2 & 3 agree
    assert torch.allclose(my_add_aten(a, b), y)
    assert torch.allclose(my_add_torchscript(a, b), y)
    print("my_add custom op OK (native / aten / torchscript):", y.tolist())
