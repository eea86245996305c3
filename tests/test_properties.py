"""Property-based tests (hypothesis) for geometry utilities: invariants that
example-based tests can miss."""
import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from deeplearning_amd.models.detection import BoxCoder
from deeplearning_amd.ops import box_iou, nms


def boxes_strategy(n_min=1, n_max=12):
    def build(draw):
        n = draw(st.integers(n_min, n_max))
        xy = draw(st.lists(
            st.tuples(st.floats(0, 90), st.floats(0, 90),
                      st.floats(1.0, 40), st.floats(1.0, 40)),
            min_size=n, max_size=n))
        b = torch.tensor([[x, y, x + w, y + h] for x, y, w, h in xy],
                         dtype=torch.float32)
        return b
    return st.composite(lambda draw: build(draw))()


@settings(max_examples=40, deadline=None, derandomize=True)
@given(boxes_strategy())
def test_iou_properties(boxes):
    iou = box_iou(boxes, boxes)
    # symmetric, diag == 1, in [0, 1]
    assert torch.allclose(iou, iou.t(), atol=1e-5)
    assert torch.allclose(iou.diag(), torch.ones(len(boxes)), atol=1e-5)
    assert float(iou.min()) >= -1e-6 and float(iou.max()) <= 1 + 1e-6


@settings(max_examples=40, deadline=None, derandomize=True)
@given(boxes_strategy())
def test_box_coder_roundtrip_property(boxes):
    coder = BoxCoder()
    anchors = boxes + torch.tensor([1.0, 2.0, 3.0, 1.5])
    deltas = coder.encode(boxes, anchors)
    decoded = coder.decode(deltas, anchors)
    assert torch.allclose(decoded, boxes, atol=1e-2)


@settings(max_examples=30, deadline=None, derandomize=True)
@given(boxes_strategy(n_min=2, n_max=10),
       st.floats(min_value=0.1, max_value=0.9))
def test_nms_properties(boxes, thr):
    scores = torch.linspace(1.0, 0.1, len(boxes))
    keep = nms(boxes, scores, thr)
    # kept indices are unique and score-sorted descending
    assert len(set(keep.tolist())) == keep.numel()
    kept_scores = scores[keep]
    assert torch.all(kept_scores[:-1] >= kept_scores[1:])
    # no two kept boxes overlap above the threshold
    if keep.numel() > 1:
        kiou = box_iou(boxes[keep], boxes[keep])
        off_diag = kiou - torch.eye(keep.numel())
        assert float(off_diag.max()) <= thr + 1e-5
    # the top-scoring box always survives
    assert 0 in keep.tolist()


@settings(max_examples=30, deadline=None, derandomize=True)
@given(st.integers(0, 64), st.integers(2, 50))
def test_voc_ap_bounds(extra_pos, n_det):
    from deeplearning_amd.engine.det_eval import voc_ap
    rng = np.random.default_rng(0)
    tp = torch.from_numpy(rng.random(n_det) < 0.5)
    # positives >= matched TPs (valid data: recall <= 1)
    n_pos = max(1, int(tp.sum()) + extra_pos)
    tp_cum = tp.float().cumsum(0)
    fp_cum = (~tp).float().cumsum(0)
    recall = tp_cum / n_pos
    precision = tp_cum / (tp_cum + fp_cum)
    ap = voc_ap(recall, precision)
    ap07 = voc_ap(recall, precision, use_07_metric=True)
    assert 0.0 <= ap <= 1.0 + 1e-6
    assert 0.0 <= ap07 <= 1.0 + 1e-6
