"""Unit tests for small engine/core pieces without dedicated coverage:
LARC trust-ratio optimizer wrapper (ref self-supervised/MAE/utils/LARS.py)
and rank-aware create_logger (ref swin utils/logger.py:9)."""
import logging

import torch

from deeplearning_amd.core.logging import TensorBoardWriter, create_logger
from deeplearning_amd.engine.lars import LARC


def test_larc_step_matches_manual_trust_ratio():
    torch.manual_seed(0)
    p = torch.randn(8, 4)
    g = torch.randn(8, 4)
    lr, wd, tc, eps = 0.1, 0.01, 0.02, 1e-8

    param = p.clone().requires_grad_()
    param.grad = g.clone()
    opt = LARC(torch.optim.SGD([param], lr=lr, weight_decay=wd),
               trust_coefficient=tc, eps=eps)
    opt.step()

    p_norm, g_norm = p.norm(), g.norm()
    adaptive = tc * p_norm / (g_norm + p_norm * wd + eps)
    adaptive = min(adaptive / lr, 1.0)
    expected = p - lr * (g + wd * p) * adaptive
    assert torch.allclose(param.detach(), expected, atol=1e-6)
    # weight_decay must be restored on the wrapped group (it is zeroed
    # during the step so SGD doesn't apply it twice)
    assert opt.param_groups[0]["weight_decay"] == wd


def test_larc_clip_caps_effective_lr():
    # tiny gradient -> raw trust ratio far above lr; with clip the update
    # must be exactly plain SGD (adaptive factor == 1)
    p = torch.ones(4)
    g = torch.full((4,), 1e-6)
    param = p.clone().requires_grad_()
    param.grad = g.clone()
    opt = LARC(torch.optim.SGD([param], lr=0.5), trust_coefficient=0.02)
    opt.step()
    assert torch.allclose(param.detach(), p - 0.5 * g, atol=1e-9)


def test_larc_skips_grad_none_and_passes_state_dict():
    a = torch.randn(3, requires_grad=True)
    b = torch.randn(3, requires_grad=True)
    a.grad = torch.randn(3)
    opt = LARC(torch.optim.SGD([a, b], lr=0.1, momentum=0.9))
    opt.step()  # b.grad is None -> must not raise
    sd = opt.state_dict()
    opt2 = LARC(torch.optim.SGD([a, b], lr=0.1, momentum=0.9))
    opt2.load_state_dict(sd)
    assert opt2.param_groups[0]["lr"] == 0.1
    opt.zero_grad(set_to_none=True)
    assert a.grad is None


def test_create_logger_rank0_console_and_file(tmp_path):
    lg = create_logger(str(tmp_path), dist_rank=0, name="dla_test_r0")
    lg.info("hello-r0")
    for h in lg.handlers:
        h.flush()
    text = (tmp_path / "log_rank0.txt").read_text()
    assert "hello-r0" in text and "rank0" in text
    assert any(isinstance(h, logging.StreamHandler) and
               not isinstance(h, logging.FileHandler) for h in lg.handlers)


def test_create_logger_nonzero_rank_no_console(tmp_path):
    lg = create_logger(str(tmp_path), dist_rank=1, name="dla_test_r1")
    lg.info("hello-r1")
    for h in lg.handlers:
        h.flush()
    assert "hello-r1" in (tmp_path / "log_rank1.txt").read_text()
    assert not any(type(h) is logging.StreamHandler for h in lg.handlers)


def test_create_logger_cached_per_name(tmp_path):
    a = create_logger(str(tmp_path), dist_rank=0, name="dla_test_cache")
    b = create_logger(str(tmp_path), dist_rank=0, name="dla_test_cache")
    assert a is b
    assert len(a.handlers) == 2  # not duplicated by the second call


def test_tensorboard_writer_noop_off_rank(tmp_path):
    w = TensorBoardWriter(str(tmp_path), rank=1)
    w.add_scalar("x", 1.0, 0)
    w.flush()
    w.close()
    assert w.writer is None


def test_complexity_counts_match_published():
    """Hook-based MAC counter vs published model complexities (swin main.py
    parity: n_parameters + flops logged at startup)."""
    from deeplearning_amd.core.complexity import (complexity_str,
                                                  count_params,
                                                  estimate_macs)
    from deeplearning_amd.models import build_model

    x = torch.randn(1, 3, 224, 224)
    m = build_model("resnet18", num_classes=1000)
    assert abs(count_params(m) / 1e6 - 11.69) < 0.02
    assert abs(estimate_macs(m, x) / 1e9 - 1.814) < 0.01
    m50 = build_model("resnet50", num_classes=1000)
    assert abs(estimate_macs(m50, x) / 1e9 - 4.089) < 0.01
    s = complexity_str(m, x)
    assert "params 11.7M" in s and "MACs 1.81G" in s


def test_complexity_batch_scales_linearly():
    from deeplearning_amd.core.complexity import estimate_macs
    from deeplearning_amd.models import build_model

    m = build_model("mnist_fcn", num_classes=10)
    m1 = estimate_macs(m, torch.randn(1, 1, 28, 28))
    m4 = estimate_macs(m, torch.randn(4, 1, 28, 28))
    assert m4 == 4 * m1 > 0


def test_weight_decay_groups():
    """set_weight_decay parity (swin utils/optimizer.py): biases, 1-D norm
    scales and model-declared keys land in the wd=0 group."""
    from deeplearning_amd.engine.optim_groups import param_groups_weight_decay
    from deeplearning_amd.models import build_model

    m = build_model("swin_t", num_classes=10)
    groups = param_groups_weight_decay(m, 0.05)
    assert len(groups) == 2
    assert groups[0]["weight_decay"] == 0.05
    assert groups[1]["weight_decay"] == 0.0
    nd_ids = {id(p) for p in groups[1]["params"]}
    for name, p in m.named_parameters():
        if "relative_position_bias_table" in name or name.endswith(".bias") \
                or p.ndim <= 1:
            assert id(p) in nd_ids, name
    total = sum(len(g["params"]) for g in groups)
    assert total == sum(1 for p in m.parameters() if p.requires_grad)

    v = build_model("vit_b16", num_classes=10)
    gv = param_groups_weight_decay(v, 0.05)
    ndv = {id(p) for p in gv[1]["params"]}
    named = dict(v.named_parameters())
    assert id(named["pos_embed"]) in ndv
    assert id(named["cls_token"]) in ndv
