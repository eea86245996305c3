"""Tests: data layer (datasets/transforms/samplers/mosaic), detection mAP
evaluator, retrieval eval, checkpoint layout of the CLI drivers."""
import json
import os
import subprocess
import sys
import tempfile
from pathlib import Path

import numpy as np
import pytest
import torch
from PIL import Image

REPO = Path(__file__).resolve().parents[1]


def test_read_split_and_dataset():
    from deeplearning_amd.data import (ClassificationDataset,
                                       classification_train_transform,
                                       read_split_data)
    with tempfile.TemporaryDirectory() as d:
        for c in ["a", "b"]:
            os.makedirs(f"{d}/{c}")
            for i in range(10):
                Image.fromarray(np.random.randint(
                    0, 255, (40, 50, 3), dtype=np.uint8)).save(
                        f"{d}/{c}/{i}.jpg")
        tp, tl, vp, vl, classes = read_split_data(d, val_rate=0.2)
        assert classes == ["a", "b"]
        assert len(tp) == 16 and len(vp) == 4
        ds = ClassificationDataset(tp, tl,
                                   classification_train_transform(32))
        x, y = ds[0]
        assert x.shape == (3, 32, 32)


def test_voc_dataset_parsing():
    from deeplearning_amd.data import VOCDetectionDataset
    with tempfile.TemporaryDirectory() as d:
        root = Path(d)
        (root / "JPEGImages").mkdir()
        (root / "Annotations").mkdir()
        (root / "ImageSets" / "Main").mkdir(parents=True)
        Image.fromarray(np.zeros((80, 100, 3), dtype=np.uint8)).save(
            root / "JPEGImages" / "im0.jpg")
        (root / "Annotations" / "im0.xml").write_text(
            "<annotation><size><width>100</width><height>80</height></size>"
            "<object><name>cat</name><bndbox>"
            "<xmin>10</xmin><ymin>20</ymin><xmax>50</xmax><ymax>60</ymax>"
            "</bndbox></object></annotation>")
        (root / "ImageSets" / "Main" / "train.txt").write_text("im0\n")
        ds = VOCDetectionDataset(root, "train")
        assert ds.get_height_and_width(0) == (80, 100)
        img, target = ds[0]
        assert img.shape == (3, 80, 100)
        assert target["boxes"].tolist() == [[10.0, 20.0, 50.0, 60.0]]
        assert target["labels"].tolist() == [8]  # 'cat' in VOC_CLASSES


def test_coco_dataset_parsing():
    from deeplearning_amd.data import COCODetectionDataset
    with tempfile.TemporaryDirectory() as d:
        Image.fromarray(np.zeros((60, 60, 3), dtype=np.uint8)).save(
            f"{d}/x.jpg")
        ann = {"images": [{"id": 7, "file_name": "x.jpg",
                           "width": 60, "height": 60}],
               "annotations": [{"id": 1, "image_id": 7, "category_id": 3,
                                "bbox": [5, 5, 20, 30], "iscrowd": 0}],
               "categories": [{"id": 3, "name": "dog"}]}
        with open(f"{d}/ann.json", "w") as f:
            json.dump(ann, f)
        ds = COCODetectionDataset(d, f"{d}/ann.json")
        img, target = ds[0]
        assert target["boxes"].tolist() == [[5.0, 5.0, 25.0, 35.0]]
        assert target["labels"].tolist() == [1]


def test_mixup_soft_targets_sum_to_one():
    from deeplearning_amd.data import Mixup
    torch.manual_seed(0)
    mix = Mixup(num_classes=7)
    x, y = mix(torch.rand(4, 3, 16, 16), torch.tensor([0, 1, 2, 3]))
    assert y.shape == (4, 7)
    assert torch.allclose(y.sum(1), torch.ones(4), atol=1e-5)


def test_mosaic_boxes_stay_inside():
    from deeplearning_amd.data import mosaic4
    torch.manual_seed(0)
    imgs = [torch.rand(3, 80, 90) for _ in range(4)]
    ts = [{"boxes": torch.tensor([[5.0, 5.0, 40.0, 50.0]]),
           "labels": torch.tensor([1])} for _ in range(4)]
    canvas, t = mosaic4(imgs, ts, out_size=160)
    assert canvas.shape == (3, 160, 160)
    if t["boxes"].numel():
        assert t["boxes"].min() >= 0 and t["boxes"].max() <= 160


def test_det_evaluator_localization_quality():
    from deeplearning_amd.engine.det_eval import DetEvaluator
    gt = {"boxes": torch.tensor([[10.0, 10.0, 50.0, 50.0]]),
          "labels": torch.tensor([1]),
          "iscrowd": torch.tensor([0])}
    # slightly offset box: IoU ~0.68 -> counts at 0.5 but not at 0.75
    pred = {"boxes": torch.tensor([[15.0, 15.0, 55.0, 55.0]]),
            "scores": torch.tensor([0.9]),
            "labels": torch.tensor([1])}
    ev = DetEvaluator()
    ev.update([pred], [gt])
    s = ev.summarize()
    assert s["mAP50"] == pytest.approx(1.0)
    assert s["mAP75"] == pytest.approx(0.0)
    assert 0.0 < s["mAP"] < 1.0


def test_det_evaluator_crowd_ignored():
    from deeplearning_amd.engine.det_eval import DetEvaluator
    gt = {"boxes": torch.tensor([[0.0, 0.0, 100.0, 100.0]]),
          "labels": torch.tensor([1]),
          "iscrowd": torch.tensor([1])}
    pred = {"boxes": torch.tensor([[0.0, 0.0, 100.0, 100.0]]),
            "scores": torch.tensor([0.9]), "labels": torch.tensor([1])}
    ev = DetEvaluator()
    ev.update([pred], [gt])
    # only crowd gt -> no countable gt -> empty summary, not FP explosion
    assert ev.summarize()["mAP"] == 0.0


def _run(script, *args, timeout=240):
    return subprocess.run([sys.executable, str(REPO / script), *args],
                          capture_output=True, text=True, timeout=timeout,
                          cwd=REPO)


@pytest.mark.slow
def test_mnist_project_cli_checkpoint_layout(tmp_path):
    r = _run("projects/classification/mnist/train.py", "--epochs", "2",
             "--batch-size", "8", "--synthetic-size", "32", "--workers", "0",
             "--device", "cpu", "--output", str(tmp_path))
    assert r.returncode == 0, r.stderr[-2000:]
    weights = tmp_path / "mnist" / "weights"
    assert (weights / "model_0.pth").exists()
    assert (weights / "model_1.pth").exists()
    assert (weights / "best_model.pth").exists()
    assert (weights / "ckpt_epoch_1.pth").exists()
    ckpt = torch.load(weights / "ckpt_epoch_1.pth", map_location="cpu",
                      weights_only=False)
    assert {"model", "optimizer", "lr_scheduler", "epoch"} <= set(ckpt)


@pytest.mark.slow
def test_unet_project_cli(tmp_path):
    r = _run("projects/Image_segmentation/U-Net/train.py", "--epochs", "1",
             "--img-size", "64", "--batch-size", "2", "--synthetic-size",
             "4", "--num-classes", "3", "--workers", "0", "--device", "cpu",
             "--output", str(tmp_path))
    assert r.returncode == 0, r.stderr[-2000:]
    assert (tmp_path / "U-Net" / "weights" / "model_0.pth").exists()


def test_cocoeval_native_matches_python():
    """C++ fast matcher (csrc/cocoeval.cpp) vs the Python reference."""
    from deeplearning_amd.engine.det_eval import (COCO_IOU_THRS, match_image,
                                                  match_image_native)
    from deeplearning_amd.ops import has_ext
    if not has_ext():
        pytest.skip("extension not built")
    torch.manual_seed(1)
    for _ in range(8):
        D = int(torch.randint(0, 30, (1,)))
        G = int(torch.randint(0, 8, (1,)))
        db = torch.rand(D, 4) * 100
        db[:, 2:] += db[:, :2] + 2
        ds = torch.rand(D)
        gb = torch.rand(G, 4) * 100
        gb[:, 2:] += gb[:, :2] + 2
        gc = torch.rand(G) > 0.8
        m1, i1, s1, n1 = match_image(db, ds, gb, gc, COCO_IOU_THRS)
        m2, i2, s2, n2 = match_image_native(db, ds, gb, gc, COCO_IOU_THRS)
        assert torch.equal(m1, m2) and torch.equal(i1, i2)
        assert torch.allclose(s1, s2) and n1 == n2


@pytest.mark.slow
def test_predict_cli(tmp_path):
    """predict.py surface: saved weights + image -> top-k prediction lines."""
    import torch as _torch
    from deeplearning_amd.core.checkpoint import save_weights
    from deeplearning_amd.models import build_model
    m = build_model("resnet18", num_classes=5)
    save_weights(m, tmp_path / "w.pth")
    Image.fromarray(np.random.randint(0, 255, (64, 64, 3),
                                      dtype=np.uint8)).save(tmp_path / "x.jpg")
    r = _run("projects/classification/resnet/predict.py",
             str(tmp_path / "x.jpg"), "--model", "resnet18",
             "--weights", str(tmp_path / "w.pth"), "--num-classes", "5",
             "--img-size", "64", "--device", "cpu")
    assert r.returncode == 0, r.stderr[-2000:]
    assert "class" in r.stdout


def test_cached_image_folder_modes(tmp_path):
    from deeplearning_amd.data import CachedImageFolder
    for c in ["a", "b"]:
        (tmp_path / c).mkdir()
        for i in range(4):
            Image.fromarray(np.random.randint(
                0, 255, (16, 16, 3), dtype=np.uint8)).save(
                    tmp_path / c / f"{i}.png")
    for mode in ("no", "part", "full"):
        ds = CachedImageFolder(tmp_path, cache_mode=mode)
        x, y = ds[0]
        assert x.shape == (3, 16, 16) and y in (0, 1)
        x2, _ = ds[0]
        assert torch.equal(x, x2)
    full = CachedImageFolder(tmp_path, cache_mode="full")
    assert len(full._cache) == len(full)


def test_zip_image_dataset(tmp_path):
    import zipfile

    from deeplearning_amd.data import ZipImageDataset
    zp = tmp_path / "imgs.zip"
    with zipfile.ZipFile(zp, "w") as zf:
        for i in range(3):
            img_path = tmp_path / f"i{i}.png"
            Image.fromarray(np.random.randint(
                0, 255, (8, 8, 3), dtype=np.uint8)).save(img_path)
            zf.write(img_path, f"imgs/i{i}.png")
    ds = ZipImageDataset(zp, [(f"imgs/i{i}.png", i) for i in range(3)])
    x, y = ds[2]
    assert x.shape == (3, 8, 8) and y == 2
    assert len(ds) == 3


def test_rand_augment_policy():
    """RandAugment preserves size/mode, is deterministic under a seeded
    random module, and every op in the policy runs standalone."""
    import random

    from PIL import Image

    from deeplearning_amd.data.autoaugment import OPS, RandAugment
    from deeplearning_amd.data.transforms import \
        classification_train_transform

    img = Image.new("RGB", (48, 40))
    px = img.load()
    for i in range(48):
        for j in range(40):
            px[i, j] = (5 * i % 256, 6 * j % 256, (i + j) % 256)

    for name, fn in OPS:
        out = fn(img, 9.0)
        assert out.size == img.size and out.mode == "RGB", name

    ra = RandAugment(num_ops=2, magnitude=9, mstd=0.5)
    random.seed(7)
    a = ra(img)
    random.seed(7)
    b = ra(img)
    assert a.tobytes() == b.tobytes()
    assert a.size == img.size

    t = classification_train_transform(32, rand_augment=True)
    x = t(img)
    assert x.shape == (3, 32, 32)
