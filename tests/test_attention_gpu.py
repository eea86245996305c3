"""GPU numerics tests for the fused MFMA attention kernel vs a plain fp32
PyTorch reference (SURVEY.md §4 strategy: generalize the Swin unit_test
pattern — kernel vs eager, fwd and bwd)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


def eager_ref_f32(qkv, H, scale, bias=None, mask=None):
    B, N, _ = qkv.shape
    qkv = qkv.float().reshape(B, N, 3, H, -1).permute(2, 0, 3, 1, 4)
    q, k, v = qkv.unbind(0)
    attn = (q @ k.transpose(-2, -1)) * scale
    if bias is not None:
        attn = attn + bias.float().unsqueeze(0)
    if mask is not None:
        nW = mask.shape[0]
        attn = attn.view(B // nW, nW, H, N, N) + \
            mask.float().unsqueeze(1).unsqueeze(0)
        attn = attn.view(B, H, N, N)
    attn = attn.softmax(dim=-1)
    return (attn @ v).transpose(1, 2).reshape(B, N, -1)


@requires_gpu
def test_mfma_fragment_layout_probe():
    """A @ B with asymmetric B pins the 16x16x32 fragment mapping."""
    from deeplearning_amd.ops import ext
    torch.manual_seed(0)
    A = torch.randn(16, 32, device="cuda").bfloat16()
    B = torch.randn(32, 16, device="cuda").bfloat16()
    D = ext().mfma_probe(A, B)
    ref = A.float() @ B.float()
    assert (D - ref).abs().max().item() < 0.15, (D - ref).abs().max()


@requires_gpu
@pytest.mark.parametrize("B,N,H,d", [(4, 197, 12, 64), (8, 49, 3, 32),
                                     (2, 50, 12, 64), (3, 256, 4, 64),
                                     (2, 48, 2, 64)])
def test_attn_fwd_matches_eager(B, N, H, d):
    from deeplearning_amd.ops.attention import fused_attention
    torch.manual_seed(0)
    qkv = torch.randn(B, N, 3 * H * d, device="cuda").bfloat16()
    scale = d ** -0.5
    out = fused_attention(qkv, H, scale)
    ref = eager_ref_f32(qkv, H, scale)
    err = (out.float() - ref).abs().max().item()
    assert err < 0.02, f"max err {err}"


@requires_gpu
def test_attn_fwd_with_bias_and_mask():
    from deeplearning_amd.ops.attention import fused_attention
    torch.manual_seed(1)
    B, N, H, d = 8, 49, 3, 32   # swin_t stage-1 windows, nW=4
    nW = 4
    qkv = torch.randn(B, N, 3 * H * d, device="cuda").bfloat16()
    bias = torch.randn(H, N, N, device="cuda")
    mask = torch.zeros(nW, N, N, device="cuda")
    mask[:, :10, 10:] = -100.0
    out = fused_attention(qkv, H, d ** -0.5, bias=bias, mask=mask)
    ref = eager_ref_f32(qkv, H, d ** -0.5, bias=bias, mask=mask)
    err = (out.float() - ref).abs().max().item()
    assert err < 0.02, f"max err {err}"


@requires_gpu
def test_attn_backward_matches_eager():
    from deeplearning_amd.ops.attention import fused_attention
    torch.manual_seed(0)
    B, N, H, d = 2, 197, 12, 64
    qkv = torch.randn(B, N, 3 * H * d, device="cuda").bfloat16()
    qkv_f = qkv.clone().requires_grad_()
    qkv_r = qkv.clone().float().requires_grad_()
    scale = d ** -0.5
    out = fused_attention(qkv_f, H, scale)
    ref = eager_ref_f32(qkv_r, H, scale)
    g = torch.randn_like(ref)
    out.float().backward(g)
    ref.backward(g)
    err = (qkv_f.grad.float() - qkv_r.grad).abs().max().item()
    scale_ref = qkv_r.grad.abs().max().item()
    assert err < 0.05 * max(scale_ref, 1.0), \
        f"bwd err {err} vs ref magnitude {scale_ref}"


@requires_gpu
def test_attn_bias_gradient():
    from deeplearning_amd.ops.attention import fused_attention
    torch.manual_seed(0)
    B, N, H, d = 4, 49, 3, 32
    qkv = torch.randn(B, N, 3 * H * d, device="cuda").bfloat16()
    bias_f = torch.randn(H, N, N, device="cuda", requires_grad=True)
    bias_r = bias_f.detach().clone().requires_grad_()
    qkv_f = qkv.clone().requires_grad_()
    qkv_r = qkv.clone().float().requires_grad_()
    out = fused_attention(qkv_f, H, d ** -0.5, bias=bias_f)
    ref = eager_ref_f32(qkv_r, H, d ** -0.5, bias=bias_r)
    g = torch.randn_like(ref)
    out.float().backward(g)
    ref.backward(g)
    err = (bias_f.grad - bias_r.grad).abs().max().item()
    mag = bias_r.grad.abs().max().item()
    assert err < 0.05 * max(mag, 1.0), f"dbias err {err} vs {mag}"


@requires_gpu
def test_swin_block_fused_vs_eager_path():
    """Whole swin_t forward: HIP fused path vs DLA_FORCE_EAGER reference."""
    import os
    from deeplearning_amd.models import build_model
    torch.manual_seed(0)
    m = build_model("swin_t", num_classes=10).cuda().bfloat16()
    m.eval()
    x = torch.randn(2, 3, 224, 224, device="cuda").bfloat16()
    with torch.no_grad():
        y_fused = m(x)
    os.environ["DLA_FORCE_EAGER"] = "1"
    try:
        with torch.no_grad():
            y_eager = m(x)
    finally:
        del os.environ["DLA_FORCE_EAGER"]
    err = (y_fused.float() - y_eager.float()).abs().max().item()
    assert err < 0.1, f"swin fused-vs-eager {err}"


def test_cosine_attention_train_parity():
    """SwinV2 fused cosine attention (default-on): fwd + bwd vs fp32 eager,
    including logit_scale and bias grads (VERDICT round-1 item 4)."""
    import torch.nn.functional as F

    torch.manual_seed(0)
    B, N, H, d = 16, 49, 4, 32
    qkv = torch.randn(B, N, 3 * H * d, device="cuda").to(torch.bfloat16) \
        .requires_grad_(True)
    ls_param = torch.randn(H, device="cuda").mul(0.1).requires_grad_(True)
    bias = torch.randn(H, N, N, device="cuda").mul(0.5).requires_grad_(True)
    mask = torch.zeros(4, N, N, device="cuda")
    mask[1, :10, 10:] = -100.0

    from deeplearning_amd.ops.attention import fused_attention_cosine
    lscale = torch.clamp(ls_param, max=torch.log(
        torch.tensor(100.0, device="cuda"))).exp()
    out = fused_attention_cosine(qkv, H, lscale, bias=bias, mask=mask)
    out.float().square().mean().backward()

    # fp32 eager reference
    qkv2 = qkv.detach().float().requires_grad_(True)
    ls2 = ls_param.detach().clone().requires_grad_(True)
    bias2 = bias.detach().float().requires_grad_(True)
    q, k, v = qkv2.reshape(B, N, 3, H, d).permute(2, 0, 3, 1, 4).unbind(0)
    attn = F.normalize(q, dim=-1) @ F.normalize(k, dim=-1).transpose(-2, -1)
    attn = attn * torch.clamp(ls2, max=torch.log(
        torch.tensor(100.0, device="cuda"))).exp().view(1, -1, 1, 1)
    attn = attn + bias2.unsqueeze(0)
    attn = attn.view(B // 4, 4, H, N, N) + mask.unsqueeze(1).unsqueeze(0)
    attn = attn.view(B, H, N, N).softmax(-1)
    ref = (attn @ v).transpose(1, 2).reshape(B, N, H * d)
    ref.square().mean().backward()

    def relerr(a, b):
        return (a.float() - b.float()).abs().max() / \
            b.float().abs().max().clamp(min=1e-6)

    assert relerr(out, ref.detach()) < 4e-2, relerr(out, ref.detach())
    assert relerr(qkv.grad, qkv2.grad) < 8e-2, relerr(qkv.grad, qkv2.grad)
    assert relerr(ls_param.grad, ls2.grad) < 8e-2
    assert relerr(bias.grad, bias2.grad) < 8e-2


def test_swinv2_block_trains_through_fused_path():
    """swinv2_t forward+backward on GPU exercises the fused cosine kernel
    (no env gate) and produces finite grads incl. logit_scale."""
    from deeplearning_amd.models import build_model

    torch.manual_seed(1)
    m = build_model("swinv2_t", num_classes=10).cuda()
    x = torch.randn(2, 3, 256, 256, device="cuda")  # v2 default img 256
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = m(x)
    y.float().square().mean().backward()
    ls = next(p for n, p in m.named_parameters() if "logit_scale" in n)
    assert ls.grad is not None and torch.isfinite(ls.grad).all()
