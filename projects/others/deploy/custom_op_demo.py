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


if __name__ == "__main__":
    a = torch.randn(4, requires_grad=True)
    b = torch.randn(4, requires_grad=True)
    y = MyAdd.apply(a, b)
    assert torch.allclose(y, 3 * a + 2 * b)
    y.sum().backward()
    assert torch.allclose(a.grad, torch.full((4,), 3.0))
    print("my_add custom op OK:", y.tolist())
