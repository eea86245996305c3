"""CPU tests for MAE/SupCon (SSL), HRNet pose, MADNet stereo, BDB metric
learning (MAE, SupCon, Insulator, MadNet, BDB,
Happy-Whale losses)."""
import pytest
import torch

from deeplearning_amd.models import build_model


def test_mae_forward_loss_backward():
    torch.manual_seed(0)
    m = build_model("mae_vit_base_patch16")
    imgs = torch.randn(2, 3, 224, 224)
    loss, pred, mask = m(imgs)
    assert pred.shape == (2, 196, 16 * 16 * 3)
    assert mask.shape == (2, 196)
    # 75% masked
    assert mask.sum().item() == pytest.approx(2 * 196 * 0.75, rel=0.01)
    loss.backward()


def test_mae_patchify_roundtrip():
    m = build_model("mae_vit_base_patch16")
    imgs = torch.randn(2, 3, 224, 224)
    assert torch.allclose(m.unpatchify(m.patchify(imgs)), imgs)


def test_supcon_loss_perfect_vs_random():
    from deeplearning_amd.models.ssl import SupConLoss
    torch.manual_seed(0)
    crit = SupConLoss()
    labels = torch.tensor([0, 0, 1, 1])
    # identical features for same class -> lower loss than random
    base = torch.nn.functional.normalize(torch.randn(2, 32), dim=1)
    aligned = base[labels].unsqueeze(1).repeat(1, 2, 1)
    rand = torch.nn.functional.normalize(torch.randn(4, 2, 32), dim=-1)
    assert crit(aligned, labels) < crit(rand, labels)


def test_pose_heatmap_pipeline():
    from deeplearning_amd.models.pose import (KeypointToHeatMap,
                                              decode_heatmaps,
                                              heatmap_focal_loss)
    kp = torch.tensor([[[40.0, 40.0], [80.0, 80.0]]])
    hm = KeypointToHeatMap((32, 32))(kp)
    assert hm.shape == (1, 2, 32, 32)
    assert hm.max().item() == pytest.approx(1.0)
    coords, scores = decode_heatmaps(hm * 10 - 5)
    assert (coords - kp).abs().max().item() <= 4.0  # within stride
    m = build_model("hrnet_w18_pose", num_joints=2)
    out = m(torch.randn(1, 3, 128, 128))
    loss = heatmap_focal_loss(out, hm)
    loss.backward()


def test_madnet_and_online_adaptation():
    from deeplearning_amd.models.stereo import MADAdapter
    torch.manual_seed(0)
    m = build_model("madnet")
    left, right = torch.randn(1, 3, 64, 128), torch.randn(1, 3, 64, 128)
    full, disps = m(left, right)
    assert full.shape == (1, 1, 64, 128)
    assert len(disps) == 6
    adapter = MADAdapter(m)
    l0 = adapter.step(left, right)
    assert l0 > 0


def test_stereo_warp_identity():
    from deeplearning_amd.models.stereo import linear_warp
    f = torch.randn(1, 4, 16, 32)
    assert torch.allclose(linear_warp(f, torch.zeros(1, 1, 16, 32)), f,
                          atol=1e-5)


def test_bdb_triplet_and_retrieval_eval():
    from deeplearning_amd.models.metric import (TripletLoss, cmc_map,
                                                hard_example_mining,
                                                pairwise_dist)
    torch.manual_seed(0)
    m = build_model("bdb_resnet50", num_classes=4)
    m.train()
    out = m(torch.randn(4, 3, 128, 64))
    labels = torch.tensor([0, 0, 1, 1])
    loss = TripletLoss()(out["global_feat"], labels) + \
        torch.nn.functional.cross_entropy(out["global_logits"], labels)
    loss.backward()
    m.eval()
    feats = m(torch.randn(4, 3, 128, 64))
    assert feats.shape[1] == 512 + 1024  # concat normalized branches

    # hard mining picks the farthest positive / closest negative
    f = torch.tensor([[0.0], [1.0], [10.0], [11.0]])
    d = pairwise_dist(f, f)
    ap, an = hard_example_mining(d, torch.tensor([0, 0, 1, 1]))
    assert ap[0].item() == pytest.approx(1.0, abs=1e-3)
    assert an[0].item() == pytest.approx(10.0, abs=1e-3)

    # perfect retrieval -> cmc@1 == 1, mAP == 1
    q = torch.eye(3)
    cmc, mAP = cmc_map(q, torch.arange(3), q, torch.arange(3))
    assert cmc[0].item() == pytest.approx(1.0)
    assert float(mAP) == pytest.approx(1.0)


def test_lovasz_losses():
    from deeplearning_amd.models.metric import lovasz_hinge, lovasz_softmax
    torch.manual_seed(0)
    labels = torch.randint(0, 3, (100,))
    probs = torch.nn.functional.one_hot(labels, 3).float()
    # perfect prediction -> zero loss
    assert lovasz_softmax(probs, labels).item() == pytest.approx(0.0, abs=1e-5)
    logits = torch.randn(50, requires_grad=True)
    loss = lovasz_hinge(logits, torch.randint(0, 2, (50,)))
    loss.backward()
    assert logits.grad is not None


def test_pose_decode_subpixel_refinement():
    """Quarter-offset refinement shifts toward the higher neighbor
    (simple-baselines post-processing in the reference decode)."""
    import torch

    from deeplearning_amd.models.pose.hrnet_pose import decode_heatmaps

    h = torch.full((1, 1, 9, 9), -8.0)
    h[0, 0, 4, 4] = 4.0    # peak at (x=4, y=4)
    h[0, 0, 4, 5] = 2.0    # right neighbor hotter than left
    h[0, 0, 3, 4] = 2.0    # upper neighbor hotter than lower
    coords, scores = decode_heatmaps(h, stride=4, refine=True)
    x, y = coords[0, 0].tolist()
    assert x == (4 + 0.25) * 4
    assert y == (4 - 0.25) * 4
    coords_raw, _ = decode_heatmaps(h, stride=4, refine=False)
    assert coords_raw[0, 0].tolist() == [16.0, 16.0]
