"""CPU tests for the eager reference paths of deeplearning_amd.ops."""
import torch
import torch.nn.functional as F

from deeplearning_amd import ops
from deeplearning_amd.ops.boxes import _box_iou_eager, _nms_eager


def test_box_iou_known():
    a = torch.tensor([[0.0, 0, 10, 10]])
    b = torch.tensor([[0.0, 0, 10, 10], [5.0, 5, 15, 15], [20.0, 20, 30, 30]])
    iou = ops.box_iou(a, b)
    assert abs(iou[0, 0].item() - 1.0) < 1e-6
    assert abs(iou[0, 1].item() - 25.0 / 175.0) < 1e-6
    assert iou[0, 2].item() == 0.0


def test_nms_eager_basic():
    boxes = torch.tensor([[0.0, 0, 10, 10], [1.0, 1, 11, 11], [20.0, 20, 30, 30]])
    scores = torch.tensor([0.9, 0.8, 0.7])
    keep = ops.nms(boxes, scores, 0.5)
    assert keep.tolist() == [0, 2]


def test_batched_nms_classes_dont_suppress():
    boxes = torch.tensor([[0.0, 0, 10, 10], [0.0, 0, 10, 10]])
    scores = torch.tensor([0.9, 0.8])
    idxs = torch.tensor([0, 1])
    keep = ops.batched_nms(boxes, scores, idxs, 0.5)
    assert len(keep) == 2


def test_bbox_iou_aligned_ciou():
    b1 = torch.tensor([[5.0, 5, 4, 4]])  # xywh
    b2 = torch.tensor([[5.0, 5, 4, 4]])
    iou = ops.bbox_iou_aligned(b1, b2, xywh=True, CIoU=True)
    assert abs(iou.item() - 1.0) < 1e-5


def test_focal_cpu_matches_formula():
    logits = torch.randn(100)
    targets = (torch.rand(100) > 0.5).float()
    loss = ops.sigmoid_focal_loss(logits, targets, alpha=0.25, gamma=2.0)
    p = torch.sigmoid(logits)
    ce = F.binary_cross_entropy_with_logits(logits, targets, reduction="none")
    p_t = p * targets + (1 - p) * (1 - targets)
    ref = ce * (1 - p_t) ** 2 * (0.25 * targets + 0.75 * (1 - targets))
    torch.testing.assert_close(loss, ref)


def test_window_roundtrip_cpu():
    x = torch.randn(2, 14, 14, 8)
    win = ops.roll_and_window_partition(x, 7, 3)
    y = ops.window_merge_and_roll(win, 2, 14, 14, 7, 3)
    assert torch.equal(x, y)


def test_layernorm_cpu():
    x = torch.randn(4, 10, 32)
    ln = ops.LayerNorm(32)
    y = ln(x)
    yr = F.layer_norm(x, (32,), ln.weight, ln.bias)
    torch.testing.assert_close(y, yr)


def test_layernorm2d_cpu():
    x = torch.randn(2, 16, 8, 8)
    ln = ops.LayerNorm2d(16)
    y = ln(x)
    assert y.shape == x.shape
    mu = y.mean(dim=1)
    torch.testing.assert_close(mu, torch.zeros_like(mu), atol=1e-5, rtol=0)


def test_drop_path():
    x = torch.ones(8, 4)
    assert torch.equal(ops.drop_path(x, 0.5, training=False), x)
    torch.manual_seed(0)
    y = ops.drop_path(x, 0.5, training=True)
    rows = y.sum(dim=1)
    assert set(rows.tolist()) <= {0.0, 8.0}


def test_ema_update():
    m = torch.nn.Linear(4, 4)
    ema = ops.ModelEMA(m, decay=0.5, tau=1.0)
    with torch.no_grad():
        for p in m.parameters():
            p.add_(1.0)
    ema.update(m)
    # after 1 update decay = 0.5*(1-e^-1) ~ 0.316
    d = ema.decay(1)
    for pe, pm in zip(ema.ema.parameters(), m.parameters()):
        expected = d * (pm - 1.0) + (1 - d) * pm
        torch.testing.assert_close(pe, expected, atol=1e-5, rtol=1e-5)


def test_roi_align_cpu_smoke():
    x = torch.arange(16.0).reshape(1, 1, 4, 4)
    rois = torch.tensor([[0.0, 0, 0, 4, 4]])
    y = ops.roi_align(x, rois, (2, 2), spatial_scale=1.0, sampling_ratio=1)
    assert y.shape == (1, 1, 2, 2)
    assert y.mean().item() > 0


def test_cross_entropy_cpu():
    logits = torch.randn(16, 10)
    target = torch.randint(0, 10, (16,))
    torch.testing.assert_close(ops.cross_entropy(logits, target),
                               F.cross_entropy(logits, target))


def test_eager_attention_mask_broadcast_vs_loop():
    """The eager fallback's [nW,N,N] window-mask broadcast must equal a
    per-window loop (guards the CPU path of fused_attention)."""
    import torch

    from deeplearning_amd.ops.attention import _eager_attention

    torch.manual_seed(0)
    nW, B_per, N, H, d = 4, 3, 16, 2, 8
    B = nW * B_per
    qkv = torch.randn(B, N, 3 * H * d)
    bias = torch.randn(H, N, N)
    mask = torch.randn(nW, N, N)
    out = _eager_attention(qkv, H, 0.3, bias=bias, mask=mask)

    q, k, v = qkv.reshape(B, N, 3, H, d).permute(2, 0, 3, 1, 4).unbind(0)
    ref = torch.empty(B, H, N, d)
    for b in range(B):
        w = b % nW  # windows are the FAST index in [B//nW, nW] batching
        a = (q[b] @ k[b].transpose(-2, -1)) * 0.3 + bias + mask[w]
        ref[b] = a.softmax(-1) @ v[b]
    ref = ref.transpose(1, 2).reshape(B, N, H * d)
    assert torch.allclose(out, ref, atol=1e-5)
