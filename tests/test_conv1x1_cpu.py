"""CPU-side behavior of the conv1x1 fused-path router: gating logic and the
eager fallback (the kernel itself is GPU-only; parity lives in
test_conv1x1_gpu.py)."""
import torch

from deeplearning_amd.ops.batchnorm import BatchNorm2d
from deeplearning_amd.ops.conv1x1 import can_fuse_conv1x1, conv_bn


def test_conv_bn_cpu_fallback_matches_eager():
    torch.manual_seed(0)
    conv = torch.nn.Conv2d(64, 128, 1, bias=False)
    bn = BatchNorm2d(128, relu=True)
    x = torch.randn(2, 64, 8, 8)
    y = conv_bn(x, conv, bn)
    torch.manual_seed(0)
    conv2 = torch.nn.Conv2d(64, 128, 1, bias=False)
    bn2 = BatchNorm2d(128, relu=True)
    y2 = bn2(conv2(x))
    assert torch.allclose(y, y2, atol=1e-6)


def test_can_fuse_gating_cpu():
    conv = torch.nn.Conv2d(64, 128, 1, bias=False)
    x = torch.randn(2, 64, 8, 8)
    assert not can_fuse_conv1x1(x, conv)  # CPU -> never


def test_can_fuse_rejects_bad_shapes():
    # channels not %64, kernel 3x3, groups — all must be rejected even
    # before the GPU checks (order of conditions shouldn't matter)
    x = torch.randn(2, 64, 8, 8)
    assert not can_fuse_conv1x1(x, torch.nn.Conv2d(64, 48, 1))
    assert not can_fuse_conv1x1(x, torch.nn.Conv2d(64, 128, 3, padding=1))
    assert not can_fuse_conv1x1(x, torch.nn.Conv2d(64, 128, 1, groups=2))


def test_resnet50_cpu_forward_unchanged():
    """The conv_bn routing must leave CPU resnet50 numerics identical."""
    from deeplearning_amd.models import build_model

    torch.manual_seed(3)
    model = build_model("resnet50", num_classes=10)
    model.eval()
    x = torch.randn(1, 3, 64, 64)
    with torch.no_grad():
        y = model(x)
    assert y.shape == (1, 10)
    assert torch.isfinite(y).all()


def test_conv3x3_routing_default_off(monkeypatch):
    """3x3 routing is opt-in (measured: integrated step regresses until the
    backward kernels are tuned) — the gate must default off and respond to
    DLA_CONV3X3."""
    from deeplearning_amd.ops import conv1x1 as m

    monkeypatch.delenv("DLA_CONV3X3", raising=False)
    assert not m._conv3x3_enabled()
    monkeypatch.setenv("DLA_CONV3X3", "1")
    assert m._conv3x3_enabled()
