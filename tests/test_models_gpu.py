"""GPU smoke tests: each model family forwards+backwards on MI355X with the
HIP kernels engaged (window kernels in swin, NMS/RoIAlign/focal in detection,
fused BN/LN/activations everywhere)."""
import pytest
import torch

from deeplearning_amd.models import build_model

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


def _fwd_bwd_cls(name, shape=(4, 3, 224, 224), nc=10, channels_last=False):
    torch.manual_seed(0)
    m = build_model(name, num_classes=nc).cuda()
    if channels_last:
        m = m.to(memory_format=torch.channels_last)
    m.train()
    x = torch.randn(*shape, device="cuda")
    if channels_last:
        x = x.contiguous(memory_format=torch.channels_last)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = m(x)
        if isinstance(out, tuple):
            loss = sum(o.float().sum() for o in out if o is not None)
        elif isinstance(out, dict):
            loss = sum(v.float().sum() for v in out.values())
        else:
            loss = out.float().sum()
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@requires_gpu
@pytest.mark.parametrize("name", ["swin_t", "convnext_tiny", "efficientnet_b0",
                                  "repvgg_a0", "se_resnet50", "resnest50",
                                  "shufflenet_v2_x1_0", "googlenet", "vgg16_bn"])
def test_classification_gpu(name):
    _fwd_bwd_cls(name)


@requires_gpu
def test_resnet50_channels_last_gpu():
    _fwd_bwd_cls("resnet50", channels_last=True)


@requires_gpu
@pytest.mark.parametrize("name", ["unet", "deeplabv3_resnet50",
                                  "hrnet_w18_seg"])
def test_segmentation_gpu(name):
    _fwd_bwd_cls(name, shape=(2, 3, 128, 128), nc=5)


@requires_gpu
def test_mae_gpu():
    m = build_model("mae_vit_base_patch16").cuda()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss, _, _ = m(torch.randn(2, 3, 224, 224, device="cuda"))
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@requires_gpu
def test_retinanet_gpu():
    torch.manual_seed(0)
    m = build_model("retinanet_resnet50_fpn", num_classes=5,
                    min_size=256, max_size=320).cuda()
    imgs = [torch.rand(3, 256, 200, device="cuda"),
            torch.rand(3, 224, 256, device="cuda")]
    targets = [{"boxes": torch.tensor([[10.0, 10.0, 100.0, 120.0]],
                                      device="cuda"),
                "labels": torch.tensor([1], device="cuda")}] * 2
    m.train()
    losses = m(imgs, targets)
    sum(losses.values()).backward()
    m.eval()
    with torch.no_grad():
        dets = m(imgs)
    torch.cuda.synchronize()
    assert len(dets) == 2


@requires_gpu
def test_fasterrcnn_gpu():
    torch.manual_seed(0)
    m = build_model("fasterrcnn_resnet50_fpn", num_classes=5,
                    min_size=256, max_size=320).cuda()
    imgs = [torch.rand(3, 256, 200, device="cuda")]
    targets = [{"boxes": torch.tensor([[10.0, 10.0, 100.0, 120.0]],
                                      device="cuda"),
                "labels": torch.tensor([1], device="cuda")}]
    m.train()
    losses = m(imgs, targets)
    sum(losses.values()).backward()
    m.eval()
    with torch.no_grad():
        dets = m(imgs)
    torch.cuda.synchronize()
    assert set(dets[0]) == {"boxes", "scores", "labels"}


@requires_gpu
def test_yolox_gpu():
    torch.manual_seed(0)
    m = build_model("yolox_s", num_classes=5).cuda()
    m.train()
    targets = [{"boxes": torch.tensor([[30.0, 30.0, 120.0, 150.0]],
                                      device="cuda"),
                "labels": torch.tensor([1], device="cuda")}]
    losses = m(torch.rand(1, 3, 256, 256, device="cuda"), targets)
    sum(losses.values()).backward()
    torch.cuda.synchronize()
    assert all(torch.isfinite(v) for v in losses.values())


@requires_gpu
def test_yolov5_gpu():
    from deeplearning_amd.models.detection import ComputeLoss
    torch.manual_seed(0)
    m = build_model("yolov5s", num_classes=5).cuda()
    m.train()
    preds = m(torch.rand(2, 3, 256, 256, device="cuda"))
    crit = ComputeLoss(m)
    targets = torch.tensor([[0, 1, 0.5, 0.5, 0.2, 0.3]], device="cuda")
    loss, _ = crit(preds, targets)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@requires_gpu
@pytest.mark.parametrize("name,shape", [
    ("swinv2_t", (2, 3, 256, 256)),
    ("swin_moe_t", (2, 3, 224, 224)),
    ("transfg_b16", (1, 3, 448, 448)),
    ("dpn68", (2, 3, 224, 224)),
    ("inception_v4", (2, 3, 299, 299)),
])
def test_more_classification_gpu(name, shape):
    torch.manual_seed(0)
    m = build_model(name, num_classes=10).cuda()
    m.train()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = m(torch.randn(*shape, device="cuda"))
        loss = out.float().sum()
        if hasattr(m, "aux_loss"):
            loss = loss + 0.01 * m.aux_loss().float()
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@requires_gpu
def test_fcos_gpu():
    torch.manual_seed(0)
    m = build_model("fcos_resnet50_fpn", num_classes=5,
                    min_size=256, max_size=320).cuda()
    imgs = [torch.rand(3, 256, 200, device="cuda")]
    targets = [{"boxes": torch.tensor([[10.0, 10.0, 100.0, 120.0]],
                                      device="cuda"),
                "labels": torch.tensor([1], device="cuda")}]
    m.train()
    losses = m(imgs, targets)
    sum(losses.values()).backward()
    torch.cuda.synchronize()
    assert all(torch.isfinite(v) for v in losses.values())


@requires_gpu
def test_madnet_gpu():
    torch.manual_seed(0)
    m = build_model("madnet").cuda()
    left = torch.rand(1, 3, 128, 256, device="cuda")
    right = torch.roll(left, 4, dims=3)
    disp, _ = m(left, right)
    disp.sum().backward()
    torch.cuda.synchronize()
    assert disp.shape == (1, 1, 128, 256)


@requires_gpu
def test_bdb_and_sspnet_gpu():
    torch.manual_seed(0)
    m = build_model("bdb_resnet50", num_classes=4).cuda()
    m.train()
    out = m(torch.rand(4, 3, 128, 64, device="cuda"))
    sum(v.float().sum() for v in out.values()).backward()

    s = build_model("sspnet").cuda()
    pred = s(torch.rand(1, 3, 96, 96, device="cuda"),
             (torch.rand(1, 96, 96, device="cuda") > 0.5).long(),
             torch.rand(1, 3, 96, 96, device="cuda"))
    pred.sum().backward()
    torch.cuda.synchronize()


@requires_gpu
def test_hrnet_pose_gpu():
    from deeplearning_amd.models.pose import (KeypointToHeatMap,
                                              heatmap_focal_loss)
    torch.manual_seed(0)
    m = build_model("hrnet_w18_pose", num_joints=4).cuda()
    x = torch.rand(2, 3, 128, 128, device="cuda")
    hm = KeypointToHeatMap((32, 32))(
        torch.rand(2, 4, 2) * 128).cuda()
    loss = heatmap_focal_loss(m(x).float(), hm)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
