"""CPU smoke + loss tests for the segmentation family (SURVEY.md §2.1:
U-Net, FCN, DeepLabV3/V3+, HR-Net-Seg + OHEM, few-shot SSP; §2.4 Dice/mIoU)."""
import pytest
import torch

from deeplearning_amd.engine.metrics import (ConfusionMatrix, dice_coeff,
                                             dice_loss)
from deeplearning_amd.models import build_model
from deeplearning_amd.models.segmentation import OhemCrossEntropy

CASES = ["unet", "fcn_resnet50", "deeplabv3_resnet50",
         "deeplabv3plus_resnet50", "hrnet_w18_seg"]


@pytest.mark.parametrize("name", CASES)
def test_seg_forward_backward(name):
    torch.manual_seed(0)
    m = build_model(name, num_classes=5)
    m.train()
    out = m(torch.randn(2, 3, 96, 96))
    assert out["out"].shape == (2, 5, 96, 96)
    sum(v.sum() for v in out.values()).backward()


def test_sspnet_episode():
    m = build_model("sspnet")
    s = torch.randn(2, 3, 96, 96)
    mask = (torch.rand(2, 96, 96) > 0.5).long()
    q = torch.randn(2, 3, 96, 96)
    pred = m(s, mask, q)
    assert pred.shape == (2, 2, 96, 96)
    pred.sum().backward()


def test_ohem_keeps_hard_pixels():
    torch.manual_seed(0)
    logits = torch.randn(2, 5, 16, 16, requires_grad=True)
    target = torch.randint(0, 5, (2, 16, 16))
    loss_all = torch.nn.functional.cross_entropy(logits, target)
    loss_ohem = OhemCrossEntropy(min_kept=10, thres=0.9)(logits, target)
    # OHEM over hardest pixels must be >= plain mean CE
    assert loss_ohem.item() >= loss_all.item()
    loss_ohem.backward()


def test_ohem_ignore_label():
    logits = torch.randn(1, 3, 8, 8)
    target = torch.full((1, 8, 8), 255)
    loss = OhemCrossEntropy(min_kept=10)(logits, target)
    assert loss.item() == 0.0


def test_dice_and_confusion_matrix():
    pred = torch.tensor([[1.0, 0.0], [0.0, 1.0]])
    assert dice_coeff(pred, pred).item() == pytest.approx(1.0, abs=1e-4)
    logits = torch.randn(2, 3, 8, 8, requires_grad=True)
    onehot = torch.nn.functional.one_hot(
        torch.randint(0, 3, (2, 8, 8)), 3).permute(0, 3, 1, 2).float()
    dl = dice_loss(logits, onehot)
    assert 0.0 <= dl.item() <= 1.0
    dl.backward()

    cm = ConfusionMatrix(3)
    t = torch.randint(0, 3, (100,))
    cm.update(t, t)
    acc, _, iou = cm.compute()
    assert acc.item() == pytest.approx(1.0)
    assert iou[iou == iou].mean().item() == pytest.approx(1.0)  # ignore NaN rows


def test_ohem_uniform_logits_no_nan():
    """OHEM edge: identical confidences make the strict threshold select
    nothing (reference code NaNs there) — must fall back to the hardest-k."""
    import torch

    from deeplearning_amd.models.segmentation.hrnet import OhemCrossEntropy

    crit = OhemCrossEntropy(thres=0.7, min_kept=10)
    logits = torch.zeros(1, 3, 8, 8, requires_grad=True)  # uniform softmax
    target = torch.randint(0, 3, (1, 8, 8))
    loss = crit(logits, target)
    assert torch.isfinite(loss)
    loss.backward()
    assert torch.isfinite(logits.grad).all()
