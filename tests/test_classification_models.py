"""CPU smoke tests: every classification family builds, forwards, backwards.

Mirrors the reference's __main__ smoke blocks as real pytest.
"""
import pytest
import torch

from deeplearning_amd.models import build_model, list_models

SMALL = 64  # small spatial size where the net allows it

CASES = [
    # (name, input shape)
    ("vgg11", (1, 3, 224, 224)),
    ("vgg16_bn", (1, 3, 224, 224)),
    ("googlenet", (1, 3, 224, 224)),
    ("resnet18", (1, 3, SMALL, SMALL)),
    ("resnet50", (1, 3, SMALL, SMALL)),
    ("resnext50_32x4d", (1, 3, SMALL, SMALL)),
    ("wide_resnet50_2", (1, 3, SMALL, SMALL)),
    ("se_resnet50", (1, 3, SMALL, SMALL)),
    ("sk_resnet50", (2, 3, SMALL, SMALL)),
    ("resnest50", (2, 3, SMALL, SMALL)),
    ("convnext_tiny", (1, 3, 224, 224)),
    ("repvgg_a0", (1, 3, SMALL, SMALL)),
    ("shufflenet_v1_g3", (1, 3, 224, 224)),
    ("shufflenet_v2_x1_0", (1, 3, 224, 224)),
    ("efficientnet_b0", (1, 3, 224, 224)),
    ("coatnet_0", (1, 3, 224, 224)),
    ("swin_t", (1, 3, 224, 224)),
    ("swinv2_t", (1, 3, 256, 256)),
    ("vit_b16", (1, 3, 224, 224)),
]


@pytest.mark.parametrize("name,shape", CASES, ids=[c[0] for c in CASES])
def test_forward_backward(name, shape):
    torch.manual_seed(0)
    m = build_model(name, num_classes=10)
    m.train()
    out = m(torch.randn(*shape))
    if isinstance(out, tuple):  # googlenet aux heads
        loss = sum(o.sum() for o in out if o is not None)
    else:
        assert out.shape == (shape[0], 10)
        loss = out.sum()
    loss.backward()
    grads = [p for p in m.parameters() if p.grad is not None]
    assert len(grads) > 0


def test_registry_has_reference_inventory():
    names = set(list_models())
    # one representative per reference subproject
    for required in ["mnist_cnn", "vgg16", "resnet50", "resnext50_32x4d",
                     "resnest50", "se_resnet50", "sk_resnet50", "googlenet",
                     "coatnet_0", "convnext_tiny", "efficientnet_b0",
                     "shufflenet_v1_g3", "shufflenet_v2_x1_0", "repvgg_a0",
                     "vit_b16", "swin_t", "swinv2_t", "transfg_b16"]:
        assert required in names, f"missing {required}"


def test_repvgg_reparam_equivalence():
    from deeplearning_amd.models.classification.repvgg import \
        repvgg_model_convert
    torch.manual_seed(0)
    m = build_model("repvgg_a0", num_classes=10)
    m.eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        y_train = m(x)
        repvgg_model_convert(m)
        y_deploy = m(x)
    assert torch.allclose(y_train, y_deploy, atol=1e-4), \
        (y_train - y_deploy).abs().max()


def test_transfg_contrastive_loss():
    from deeplearning_amd.models.classification.transfg import contrastive_loss
    f = torch.randn(8, 16, requires_grad=True)
    labels = torch.randint(0, 3, (8,))
    loss = contrastive_loss(f, labels)
    loss.backward()
    assert loss.item() >= 0 and f.grad is not None


@pytest.mark.parametrize("name,size", [("dpn68", 224), ("inception_v4", 299),
                                       ("swin_moe_t", 224)])
def test_zoo_extra_forward_backward(name, size):
    torch.manual_seed(0)
    m = build_model(name, num_classes=10)
    m.train()
    out = m(torch.randn(2, 3, size, size))
    out.sum().backward()
    assert out.shape == (2, 10)


@pytest.mark.parametrize("name,size,feat", [
    ("xception", 96, 2048),
    ("senet154", 96, 2048),
    ("polynet", 128, 2048),
    ("nasnet_a_mobile", 96, 1056),
])
def test_zoo_tail_models(name, size, feat):
    """Happy-Whale model-zoo tail (ref modelZoo/{xception,senet,ployNet,
    nasnet}.py): forward shape + feature dim + one backward step."""
    from deeplearning_amd.models import build_model

    m = build_model(name, num_classes=5)
    assert m.num_features == feat
    x = torch.randn(2, 3, size, size)
    y = m(x)
    assert y.shape == (2, 5)
    y.square().mean().backward()
    g = next(p.grad for p in m.parameters() if p.grad is not None)
    assert torch.isfinite(g).all()


def test_nasnet_large_constructs():
    from deeplearning_amd.models import build_model

    m = build_model("nasnet_a_large", num_classes=3)
    n = sum(p.numel() for p in m.parameters())
    assert n > 50e6  # the @large config (published 88.7M)
