"""CPU tests for deeplearning_amd.core: config, meters, checkpoint, env."""

import pytest
import torch
import torch.nn as nn

from deeplearning_amd.core import (AverageMeter, CfgNode, MetricLogger,
                                   SmoothedValue, auto_resume_helper,
                                   increment_path, load_checkpoint,
                                   load_config, load_pretrained,
                                   save_checkpoint, save_weights,
                                   seed_everything, select_device,
                                   strip_module_prefix)


def test_cfg_basic():
    cfg = CfgNode({"model": {"name": "resnet50", "depth": 50}, "lr": 0.1})
    assert cfg.model.name == "resnet50"
    cfg.model.depth = 101
    assert cfg["model"]["depth"] == 101
    c2 = cfg.clone()
    c2.lr = 0.2
    assert cfg.lr == 0.1


def test_cfg_freeze():
    cfg = CfgNode({"a": 1})
    cfg.freeze()
    with pytest.raises(AttributeError):
        cfg.a = 2


def test_cfg_merge_list():
    cfg = CfgNode({"train": {"lr": 0.1, "epochs": 10}, "name": "x"})
    cfg.merge_from_list(["train.lr", "0.5", "train.epochs", "20", "name", "y"])
    assert cfg.train.lr == 0.5 and cfg.train.epochs == 20 and cfg.name == "y"


def test_cfg_yaml_base(tmp_path):
    base = tmp_path / "base.yaml"
    base.write_text("model:\n  name: resnet18\ntrain:\n  lr: 0.1\n")
    child = tmp_path / "child.yaml"
    child.write_text(f"_BASE_: {base}\ntrain:\n  lr: 0.2\n")
    cfg = load_config({"model": {"name": ""}, "train": {"lr": 0.0, "epochs": 5}},
                      yaml_file=str(child), freeze=False)
    assert cfg.model.name == "resnet18"
    assert cfg.train.lr == 0.2
    assert cfg.train.epochs == 5


def test_meters():
    m = AverageMeter()
    m.update(1.0)
    m.update(3.0)
    assert m.avg == 2.0
    s = SmoothedValue(window_size=2)
    for v in [1.0, 2.0, 3.0]:
        s.update(v)
    assert s.value == 3.0
    assert s.global_avg == 2.0
    ml = MetricLogger()
    ml.update(loss=torch.tensor(0.5))
    assert abs(ml.meters["loss"].value - 0.5) < 1e-6


def test_checkpoint_roundtrip(tmp_path):
    model = nn.Sequential(nn.Linear(4, 4), nn.ReLU(), nn.Linear(4, 2))
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    sched = torch.optim.lr_scheduler.StepLR(opt, 1)
    path = tmp_path / "ckpt_epoch_3.pth"
    save_checkpoint(path, model, opt, sched, epoch=3, max_accuracy=0.9)
    model2 = nn.Sequential(nn.Linear(4, 4), nn.ReLU(), nn.Linear(4, 2))
    opt2 = torch.optim.SGD(model2.parameters(), lr=0.1)
    ck = load_checkpoint(path, model2, opt2)
    assert ck["epoch"] == 3 and ck["max_accuracy"] == 0.9
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p1, p2)


def test_auto_resume(tmp_path):
    for e in [1, 5, 3]:
        (tmp_path / f"ckpt_epoch_{e}.pth").write_bytes(b"x")
    assert auto_resume_helper(tmp_path).endswith("ckpt_epoch_5.pth")
    assert auto_resume_helper(tmp_path / "empty") is None


def test_partial_load(tmp_path):
    m1 = nn.Sequential(nn.Linear(4, 4), nn.Linear(4, 2))
    save_weights(m1, tmp_path / "w.pth")
    m2 = nn.Sequential(nn.Linear(4, 4), nn.Linear(4, 3))  # last layer differs
    dropped = load_pretrained(m2, tmp_path / "w.pth")
    assert any("1." in k for k in dropped)
    assert torch.equal(m1[0].weight, m2[0].weight)


def test_strip_module_prefix():
    sd = {"module.layer.weight": torch.zeros(1), "other": torch.ones(1)}
    out = strip_module_prefix(sd)
    assert "layer.weight" in out and "other" in out


def test_increment_path(tmp_path):
    p = tmp_path / "exp"
    assert increment_path(p) == p
    p.mkdir()
    assert increment_path(p).name == "exp2"


def test_env():
    seed_everything(0)
    a = torch.rand(3)
    seed_everything(0)
    b = torch.rand(3)
    assert torch.equal(a, b)
    assert select_device("cpu").type == "cpu"
