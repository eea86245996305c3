"""Edge-case and component tests: window-op eager parity, yolov5 yaml loader,
MosaicDetection, Exp file loading, box utilities, detection transform."""
import textwrap

import pytest
import torch

from deeplearning_amd.ops import (batched_nms, box_iou, clip_boxes_to_image,
                                  nms, remove_small_boxes)
from deeplearning_amd.ops.window import (window_partition_eager,
                                         window_reverse_eager)


def test_window_partition_roundtrip():
    torch.manual_seed(0)
    x = torch.randn(2, 56, 56, 96)
    w = window_partition_eager(x, 7)
    assert w.shape == (2 * 8 * 8, 7, 7, 96)
    back = window_reverse_eager(w, 7, 56, 56)
    assert torch.equal(back, x)


def test_roll_partition_eager_path_matches_manual():
    from deeplearning_amd.ops.window import (roll_and_window_partition,
                                             window_merge_and_roll)
    torch.manual_seed(0)
    x = torch.randn(2, 14, 14, 8)
    out = roll_and_window_partition(x, 7, 3)
    ref = window_partition_eager(torch.roll(x, (-3, -3), (1, 2)), 7)
    assert torch.equal(out, ref)
    back = window_merge_and_roll(out, 2, 14, 14, 7, 3)
    assert torch.allclose(back, x)


def test_nms_degenerate_cases():
    assert nms(torch.zeros(0, 4), torch.zeros(0), 0.5).numel() == 0
    # identical boxes: keep exactly one
    b = torch.tensor([[0.0, 0.0, 10.0, 10.0]] * 5)
    s = torch.arange(5.0)
    keep = nms(b, s, 0.5)
    assert keep.tolist() == [4]
    # batched nms never suppresses across classes
    idxs = torch.tensor([0, 1, 0, 1, 0])
    keep2 = batched_nms(b, s, idxs, 0.5)
    assert sorted(idxs[keep2].tolist()) == [0, 1]


def test_box_utils():
    b = torch.tensor([[-5.0, -5.0, 15.0, 15.0], [1.0, 1.0, 1.5, 1.5]])
    clipped = clip_boxes_to_image(b, (10, 12))
    assert clipped[0].tolist() == [0.0, 0.0, 12.0, 10.0]
    keep = remove_small_boxes(b, 1.0)
    assert keep.tolist() == [0]
    iou = box_iou(torch.tensor([[0.0, 0.0, 10.0, 10.0]]),
                  torch.tensor([[0.0, 0.0, 10.0, 10.0],
                                [5.0, 5.0, 15.0, 15.0]]))
    assert iou[0, 0].item() == pytest.approx(1.0)
    assert iou[0, 1].item() == pytest.approx(25.0 / 175.0, rel=1e-4)


def test_detection_transform_resizes_boxes():
    from deeplearning_amd.models.detection import GeneralizedRCNNTransform
    tr = GeneralizedRCNNTransform(min_size=100, max_size=200)
    img = torch.rand(3, 50, 80)
    target = {"boxes": torch.tensor([[10.0, 10.0, 40.0, 30.0]]),
              "labels": torch.tensor([1])}
    il, targets = tr([img], [target])
    h, w = il.image_sizes[0]
    scale = h / 50
    assert torch.allclose(targets[0]["boxes"],
                          target["boxes"] * scale, atol=1.0)
    assert il.tensors.shape[-1] % 32 == 0 and il.tensors.shape[-2] % 32 == 0


def test_yolov5_from_yaml_matches_factory():
    from deeplearning_amd.models import build_model
    from deeplearning_amd.models.detection.yolov5 import yolov5_from_yaml
    torch.manual_seed(0)
    m_yaml = yolov5_from_yaml("configs/yolov5/yolov5s.yaml", nc=5)
    m_fact = build_model("yolov5s", num_classes=5)
    # same architecture: identical parameter shapes
    sy = [tuple(p.shape) for p in m_yaml.parameters()]
    sf = [tuple(p.shape) for p in m_fact.parameters()]
    assert sy == sf


def test_mosaic_detection_dataset():
    from deeplearning_amd.data import MosaicDetection
    from deeplearning_amd.engine.cli_det import SyntheticDetection
    ds = MosaicDetection(SyntheticDetection(8, (3, 96, 96), 4),
                         out_size=192)
    img, t = ds[0]
    assert img.shape == (3, 192, 192)
    img2, _ = ds[(False, 0)]  # YoloBatchSampler pass-through
    assert img2.shape == (3, 96, 96)


def test_exp_file_loading(tmp_path):
    from deeplearning_amd.engine.exp import get_exp
    f = tmp_path / "my_exp.py"
    f.write_text(textwrap.dedent("""
        from deeplearning_amd.engine.exp import YoloxExp

        class Exp(YoloxExp):
            model_name = "yolox_s"
            num_classes = 7
            max_epoch = 42
    """))
    exp = get_exp(exp_file=str(f))
    assert exp.num_classes == 7 and exp.max_epoch == 42


def test_prefetcher_requires_gpu_shapes():
    # CPU-side structural check only: DataPrefetcher is GPU-only
    from deeplearning_amd.data import DataPrefetcher
    assert DataPrefetcher is not None


def test_rocpd_stats_tool(tmp_path):
    """tools/rocpd_stats.py summarizes a rocprofv3-style sqlite db."""
    import sqlite3
    import subprocess
    import sys
    from pathlib import Path

    db = tmp_path / "r.db"
    con = sqlite3.connect(db)
    con.execute("CREATE TABLE kernels (kernel_name TEXT, start INT, end INT)")
    rows = [("slow_kernel", 0, 1000), ("slow_kernel", 2000, 3200),
            ("fast_kernel", 1000, 1100), ("late_kernel", 9000, 9500)]
    con.executemany("INSERT INTO kernels VALUES (?,?,?)", rows)
    con.commit()
    con.close()
    repo = Path(__file__).resolve().parents[1]
    out = tmp_path / "s.csv"
    r = subprocess.run(
        [sys.executable, str(repo / "tools/rocpd_stats.py"), str(db),
         "-o", str(out)], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    lines = out.read_text().strip().splitlines()
    assert lines[0].startswith("kernel,")
    assert lines[1].startswith("slow_kernel,2200,2")
    # tail filter keeps only the late kernel
    r2 = subprocess.run(
        [sys.executable, str(repo / "tools/rocpd_stats.py"), str(db),
         "--tail-frac", "0.2", "-o", str(out)],
        capture_output=True, text=True)
    lines2 = out.read_text().strip().splitlines()
    assert len(lines2) == 2 and lines2[1].startswith("late_kernel")


def test_generic_export_covers_pose_model(tmp_path):
    """others/deploy export handles any registry model (here: pose HRNet)."""
    import subprocess
    import sys
    from pathlib import Path

    repo = Path(__file__).resolve().parents[1]
    r = subprocess.run(
        [sys.executable, str(repo / "projects/others/deploy/export_onnx.py"),
         "--model", "hrnet_w18_pose", "--img-size", "64",
         "--out", str(tmp_path / "pose.onnx")],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]
    assert "exported" in r.stdout


def test_exp_closes_mosaic_for_no_aug_epochs():
    """YOLOX protocol: the last no_aug_epochs train WITHOUT mosaic
    (ref yolox/core/trainer.py before_epoch)."""
    from deeplearning_amd.engine.exp import get_exp

    exp = get_exp(exp_name="yolox_s")
    exp.max_epoch = 20
    exp.no_aug_epochs = 5
    loader = exp.get_data_loader(batch_size=2, synthetic_size=4)
    assert loader.dataset.enabled  # mosaic on initially
    assert not exp.close_mosaic_if_due(loader, epoch=10)
    assert loader.dataset.enabled
    assert exp.close_mosaic_if_due(loader, epoch=15)
    assert not loader.dataset.enabled
