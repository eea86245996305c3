"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32 reference of the
same op — forward AND backward, fp32 and bf16 (the generalized form of the
reference's unit test, swin kernels/window_process/unit_test.py:123-246)."""
import pytest
import torch
import torch.nn.functional as F

from deeplearning_amd import ops

pytestmark = pytest.mark.gpu

DEV = "cuda"


def _tol(dtype):
    return dict(atol=1e-5, rtol=1e-5) if dtype == torch.float32 else dict(atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(8, 197, 768), (4, 3136, 96), (2, 49, 1536), (3, 7, 131)])
def test_layernorm(dtype, shape):
    torch.manual_seed(0)
    C = shape[-1]
    x = torch.randn(shape, device=DEV, dtype=dtype, requires_grad=True)
    w = torch.randn(C, device=DEV, dtype=dtype, requires_grad=True)
    b = torch.randn(C, device=DEV, dtype=dtype, requires_grad=True)
    y = ops.layer_norm(x, w, b)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True)
    yr = F.layer_norm(xr, (C,), wr, br)
    torch.testing.assert_close(y.float(), yr, **_tol(dtype))
    g = torch.randn_like(yr).to(dtype)  # same (rounded) grad to both paths
    y.backward(g)
    yr.backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, **_tol(dtype))
    M = x.numel() // C
    sum_tol = dict(atol=max(1e-2, 2e-3 * M ** 0.5), rtol=2e-2) \
        if dtype != torch.float32 else dict(atol=1e-3, rtol=1e-3)
    torch.testing.assert_close(w.grad.float(), wr.grad, **sum_tol)
    torch.testing.assert_close(b.grad.float(), br.grad, **sum_tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("fn,ref", [(ops.gelu, F.gelu), (ops.silu, F.silu)])
def test_activations(dtype, fn, ref):
    torch.manual_seed(0)
    x = torch.randn(1024, 1000, device=DEV, dtype=dtype, requires_grad=True)
    y = fn(x)
    xr = x.detach().float().requires_grad_(True)
    yr = ref(xr)
    torch.testing.assert_close(y.float(), yr, **_tol(dtype))
    g = torch.randn_like(yr)
    y.backward(g.to(dtype))
    yr.backward(g)
    torch.testing.assert_close(x.grad.float(), xr.grad, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_add_relu(dtype):
    torch.manual_seed(0)
    a = torch.randn(64, 256, 14, 14, device=DEV, dtype=dtype, requires_grad=True)
    b = torch.randn_like(a, requires_grad=True)
    y = ops.add_relu(a, b)
    ar = a.detach().float().requires_grad_(True)
    br = b.detach().float().requires_grad_(True)
    yr = torch.relu(ar + br)
    torch.testing.assert_close(y.float(), yr, **_tol(dtype))
    g = torch.randn_like(yr)
    y.backward(g.to(dtype))
    yr.backward(g)
    torch.testing.assert_close(a.grad.float(), ar.grad, **_tol(dtype))
    torch.testing.assert_close(b.grad.float(), br.grad, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("relu", [False, True])
def test_batchnorm_train(dtype, relu):
    torch.manual_seed(0)
    N, C, H, W = 16, 64, 28, 28
    bn = ops.BatchNorm2d(C, relu=relu).to(DEV).to(dtype)
    bn_ref = torch.nn.BatchNorm2d(C).to(DEV).float()
    bn_ref.load_state_dict({k: v.float() for k, v in bn.state_dict().items()})
    x = torch.randn(N, C, H, W, device=DEV, dtype=dtype, requires_grad=True)
    y = bn(x)
    xr = x.detach().float().requires_grad_(True)
    yr = bn_ref(xr)
    if relu:
        yr = torch.relu(yr)
    torch.testing.assert_close(y.float(), yr, **_tol(dtype))
    g = torch.randn_like(yr).to(dtype)  # same (rounded) grad to both paths
    y.backward(g)
    yr.backward(g.float())
    torch.testing.assert_close(x.grad.float(), xr.grad, atol=2e-2 if dtype != torch.float32 else 1e-4,
                               rtol=2e-2 if dtype != torch.float32 else 1e-4)
    rows = N * H * W
    torch.testing.assert_close(bn.weight.grad.float(), bn_ref.weight.grad,
                               atol=max(5e-2, 2e-3 * rows ** 0.5), rtol=2e-2)
    torch.testing.assert_close(bn.running_mean.float(), bn_ref.running_mean,
                               atol=1e-2, rtol=1e-2)
    torch.testing.assert_close(bn.running_var.float(), bn_ref.running_var,
                               atol=1e-2, rtol=1e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("relu", [False, True])
def test_batchnorm_train_channels_last(dtype, relu):
    """NHWC path: same numerics as NCHW reference, layout preserved."""
    torch.manual_seed(0)
    N, C, H, W = 8, 64, 14, 14
    bn = ops.BatchNorm2d(C, relu=relu).to(DEV).to(dtype)
    bn_ref = torch.nn.BatchNorm2d(C).to(DEV).float()
    bn_ref.load_state_dict({k: v.float() for k, v in bn.state_dict().items()})
    x = torch.randn(N, C, H, W, device=DEV, dtype=dtype).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    y = bn(x)
    assert y.is_contiguous(memory_format=torch.channels_last)
    xr = x.detach().float().contiguous().requires_grad_(True)
    yr = bn_ref(xr)
    if relu:
        yr = torch.relu(yr)
    torch.testing.assert_close(y.float().contiguous(), yr, **_tol(dtype))
    g = torch.randn_like(yr).to(dtype)
    y.backward(g.contiguous(memory_format=torch.channels_last))
    yr.backward(g.float())
    torch.testing.assert_close(x.grad.float().contiguous(), xr.grad,
                               atol=2e-2 if dtype != torch.float32 else 1e-4,
                               rtol=2e-2 if dtype != torch.float32 else 1e-4)


def test_add_relu_channels_last():
    torch.manual_seed(0)
    a = torch.randn(4, 32, 8, 8, device=DEV, dtype=torch.bfloat16).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    b = torch.randn(4, 32, 8, 8, device=DEV, dtype=torch.bfloat16).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    y = ops.add_relu(a, b)
    assert y.is_contiguous(memory_format=torch.channels_last)
    yr = torch.relu(a.detach().float() + b.detach().float())
    torch.testing.assert_close(y.float().contiguous(), yr.contiguous(),
                               atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_batchnorm_eval(dtype):
    C = 32
    bn = ops.BatchNorm2d(C).to(DEV).to(dtype).eval()
    with torch.no_grad():
        bn.running_mean.uniform_(-1, 1)
        bn.running_var.uniform_(0.5, 2)
        x = torch.randn(4, C, 8, 8, device=DEV, dtype=dtype)
        y = bn(x)
        yr = F.batch_norm(x.float(), bn.running_mean.float(), bn.running_var.float(),
                          bn.weight.float(), bn.bias.float(), False, 0.0, bn.eps)
    torch.testing.assert_close(y.float(), yr, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("smoothing", [0.0, 0.1])
def test_cross_entropy(dtype, smoothing):
    torch.manual_seed(0)
    B, C = 128, 1000
    logits = torch.randn(B, C, device=DEV, dtype=dtype, requires_grad=True)
    target = torch.randint(0, C, (B,), device=DEV)
    loss = ops.cross_entropy(logits, target, smoothing=smoothing)
    lr_ = logits.detach().float().requires_grad_(True)
    loss_ref = F.cross_entropy(lr_, target, label_smoothing=smoothing)
    torch.testing.assert_close(loss.float(), loss_ref, atol=1e-2 if dtype != torch.float32 else 1e-5,
                               rtol=1e-2 if dtype != torch.float32 else 1e-5)
    loss.backward()
    loss_ref.backward()
    torch.testing.assert_close(logits.grad.float(), lr_.grad, **_tol(dtype))


def test_cross_entropy_ignore_index():
    B, C = 64, 10
    logits = torch.randn(B, C, device=DEV, requires_grad=True)
    target = torch.randint(0, C, (B,), device=DEV)
    target[::4] = -100
    loss = ops.cross_entropy(logits, target)
    lr_ = logits.detach().clone().requires_grad_(True)
    loss_ref = F.cross_entropy(lr_, target, ignore_index=-100)
    torch.testing.assert_close(loss, loss_ref, atol=1e-5, rtol=1e-5)
    loss.backward()
    loss_ref.backward()
    torch.testing.assert_close(logits.grad, lr_.grad, atol=1e-5, rtol=1e-5)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_soft_target_ce(dtype):
    torch.manual_seed(0)
    B, C = 64, 100
    logits = torch.randn(B, C, device=DEV, dtype=dtype, requires_grad=True)
    target = torch.softmax(torch.randn(B, C, device=DEV, dtype=dtype), dim=-1)
    loss = ops.soft_target_cross_entropy(logits, target)
    lr_ = logits.detach().float().requires_grad_(True)
    loss_ref = torch.sum(-target.float() * F.log_softmax(lr_, dim=-1), dim=-1).mean()
    torch.testing.assert_close(loss.float(), loss_ref, atol=1e-2, rtol=1e-2)
    loss.backward()
    loss_ref.backward()
    torch.testing.assert_close(logits.grad.float(), lr_.grad, **_tol(dtype))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("alpha", [0.25, -1.0])
def test_focal_loss(dtype, alpha):
    torch.manual_seed(0)
    logits = (torch.randn(5000, device=DEV, dtype=dtype) * 3).requires_grad_(True)
    targets = (torch.rand(5000, device=DEV) > 0.9).to(dtype)
    loss = ops.sigmoid_focal_loss(logits, targets, alpha=alpha, reduction="sum")
    lr_ = logits.detach().float().requires_grad_(True)
    p = torch.sigmoid(lr_)
    tf = targets.float()
    ce = F.binary_cross_entropy_with_logits(lr_, tf, reduction="none")
    p_t = p * tf + (1 - p) * (1 - tf)
    lref = ce * ((1 - p_t) ** 2.0)
    if alpha >= 0:
        lref = lref * (alpha * tf + (1 - alpha) * (1 - tf))
    lref = lref.sum()
    torch.testing.assert_close(loss.float(), lref, atol=5e-1 if dtype != torch.float32 else 1e-3,
                               rtol=1e-2 if dtype != torch.float32 else 1e-5)
    loss.backward()
    lref.backward()
    torch.testing.assert_close(logits.grad.float(), lr_.grad,
                               atol=5e-2 if dtype != torch.float32 else 1e-4,
                               rtol=5e-2 if dtype != torch.float32 else 1e-4)


def test_box_iou():
    torch.manual_seed(0)
    a = torch.rand(200, 4, device=DEV) * 100
    a[:, 2:] += a[:, :2]
    b = torch.rand(150, 4, device=DEV) * 100
    b[:, 2:] += b[:, :2]
    from deeplearning_amd.ops.boxes import _box_iou_eager

    got = ops.box_iou(a, b)
    ref = _box_iou_eager(a.cpu(), b.cpu())
    torch.testing.assert_close(got.cpu(), ref, atol=1e-5, rtol=1e-5)
    giou = ops.generalized_box_iou(a, b)
    assert giou.shape == (200, 150)
    assert (giou <= got + 1e-5).all()


def test_nms_matches_eager():
    torch.manual_seed(0)
    n = 2000
    boxes = torch.rand(n, 4, device=DEV) * 200
    boxes[:, 2:] = boxes[:, :2] + torch.rand(n, 2, device=DEV) * 50 + 1
    scores = torch.rand(n, device=DEV)
    keep_gpu = ops.nms(boxes, scores, 0.5)
    from deeplearning_amd.ops.boxes import _nms_eager

    keep_ref = _nms_eager(boxes.cpu(), scores.cpu(), 0.5)
    assert keep_gpu.cpu().tolist() == keep_ref.tolist()


def test_batched_nms():
    torch.manual_seed(1)
    n = 500
    boxes = torch.rand(n, 4, device=DEV) * 100
    boxes[:, 2:] = boxes[:, :2] + 10
    scores = torch.rand(n, device=DEV)
    idxs = torch.randint(0, 5, (n,), device=DEV)
    keep = ops.batched_nms(boxes, scores, idxs, 0.5)
    # identical boxes in different classes never suppress each other
    assert keep.numel() >= 5 or n < 5


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("aligned", [False, True])
def test_roi_align(dtype, aligned):
    torch.manual_seed(0)
    x = torch.randn(2, 16, 32, 32, device=DEV, dtype=dtype, requires_grad=True)
    rois = torch.tensor([[0, 2.0, 3.0, 20.0, 25.0], [1, 0.0, 0.0, 31.0, 31.0],
                         [0, 10.0, 10.0, 12.0, 14.0]], device=DEV)
    y = ops.roi_align(x, rois, (7, 7), spatial_scale=0.5, sampling_ratio=2,
                      aligned=aligned)
    from deeplearning_amd.ops.roi_align import _roi_align_eager

    yr = _roi_align_eager(x.detach().float().cpu(), rois.cpu(), (7, 7), 0.5, 2, aligned)
    torch.testing.assert_close(y.float().cpu(), yr, **_tol(dtype))
    y.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shift", [0, 2, 3])
def test_window_ops(dtype, shift):
    """The reference unit test's canonical shape (unit_test.py:124-131)."""
    torch.manual_seed(0)
    B, H, W, C, ws = 16, 56, 56, 96, 7
    x = torch.randn(B, H, W, C, device=DEV, dtype=dtype, requires_grad=True)
    win = ops.roll_and_window_partition(x, ws, shift)
    xr = x.detach().clone().requires_grad_(True)
    xr2 = torch.roll(xr, (-shift, -shift), (1, 2)) if shift else xr
    win_ref = ops.window_partition_eager(xr2, ws)
    assert torch.equal(win, win_ref)  # pure gather: bitwise equal
    g = torch.randn_like(win)
    win.backward(g)
    win_ref.backward(g)
    assert torch.equal(x.grad, xr.grad)

    # merge + roll inverse
    w2 = win.detach().clone().requires_grad_(True)
    y = ops.window_merge_and_roll(w2, B, H, W, ws, shift)
    assert torch.equal(y, x.detach())
    g2 = torch.randn_like(y)
    y.backward(g2)
    w3 = win.detach().clone().requires_grad_(True)
    ym = ops.window_reverse_eager(w3, ws, H, W)
    if shift:
        ym = torch.roll(ym, (shift, shift), (1, 2))
    ym.backward(g2)
    assert torch.equal(w2.grad, w3.grad)


def test_native_ext_loaded():
    """Assert the in-tree HIP extension is what's running on this GPU box."""
    assert ops.has_ext(), "HIP extension must be built and loadable on a GPU box"
    import deeplearning_amd.ops._dla_hip as m

    assert "_dla_hip" in m.__file__
