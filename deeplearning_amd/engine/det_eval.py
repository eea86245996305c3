"""Detection evaluation: COCO-style mAP@[.5:.95] and VOC mAP, with cross-rank
result merge.

Reference parity: detection/RetinaNet/train_utils/coco_eval.py:15-199
(CocoEvaluator + all_gather merge), YOLOX yolox/evaluators/{coco_evaluator,
voc_eval}.py — re-designed without pycocotools: greedy per-IoU-threshold
matching (highest score first, crowd-aware) + 101-point PR interpolation,
identical protocol to COCOeval's bbox task. The hot matching loop has a C++
fast path (csrc/cocoeval.cpp, ref YOLOX's native cocoeval §2.2) with this
file as the pure-Python reference.
"""
from __future__ import annotations

import torch

from ..core.dist import all_gather_object_list as all_gather_object
from ..core.dist import get_world_size
from ..ops import box_iou
from ..ops._ext import has_ext

COCO_IOU_THRS = [0.5 + 0.05 * i for i in range(10)]


def match_image_native(det_boxes, det_scores, gt_boxes, gt_crowd, iou_thrs,
                       max_dets=100):
    """C++ fast path (csrc/cocoeval.cpp, ref YOLOX native cocoeval)."""
    from ..ops._ext import ext
    matched, ignored, scores, n_gt = ext().cocoeval_match_image(
        det_boxes.float(), det_scores.float(), gt_boxes.float(),
        gt_crowd.to(torch.bool), torch.as_tensor(iou_thrs), max_dets)
    return matched, ignored, scores, int(n_gt)


def match_image_ranged_native(det_boxes, det_scores, gt_boxes, gt_crowd,
                              gt_extra_ignore, iou_thrs, max_dets=100):
    """C++ fast path of match_image_ranged (area-range evaluation)."""
    from ..ops._ext import ext
    matched, ignored, scores, n_gt = ext().cocoeval_match_image(
        det_boxes.float(), det_scores.float(), gt_boxes.float(),
        gt_crowd.to(torch.bool), torch.as_tensor(iou_thrs), max_dets,
        gt_extra_ignore.to(torch.bool))
    return matched, ignored, scores, int(n_gt)


def match_image(det_boxes, det_scores, gt_boxes, gt_crowd, iou_thrs,
                max_dets=100):
    """Greedy COCO matching for one image+class.

    Returns (matched [T, D] bool, ignored [T, D] bool, scores [D], n_gt).
    """
    order = det_scores.argsort(descending=True)[:max_dets]
    det_boxes = det_boxes[order]
    det_scores = det_scores[order]
    D = det_boxes.shape[0]
    G = gt_boxes.shape[0]
    T = len(iou_thrs)
    matched = torch.zeros(T, D, dtype=torch.bool)
    ignored = torch.zeros(T, D, dtype=torch.bool)
    n_gt = int((~gt_crowd).sum()) if G else 0
    if D == 0 or G == 0:
        return matched, ignored, det_scores, n_gt
    ious = box_iou(det_boxes, gt_boxes)  # D, G
    if gt_crowd.any():
        # pycocotools protocol: crowd IoU = intersection / det area
        lt = torch.max(det_boxes[:, None, :2], gt_boxes[None, :, :2])
        rb = torch.min(det_boxes[:, None, 2:], gt_boxes[None, :, 2:])
        inter = (rb - lt).clamp(min=0).prod(-1)
        det_area = ((det_boxes[:, 2] - det_boxes[:, 0]) *
                    (det_boxes[:, 3] - det_boxes[:, 1])).clamp(min=1e-9)
        crowd_iou = inter / det_area[:, None]
        ious = torch.where(gt_crowd[None, :], crowd_iou, ious)
    for t, thr in enumerate(iou_thrs):
        taken = torch.zeros(G, dtype=torch.bool)
        for d in range(D):
            # pycocotools semantics (gts sorted ignore-last + break): ANY
            # non-crowd match beats any crowd match regardless of IoU
            best_nc = best_c = thr
            g_nc = g_c = -1
            for g in range(G):
                v = float(ious[d, g])
                if gt_crowd[g]:
                    if v >= best_c:
                        best_c, g_c = v, g
                else:
                    if taken[g]:
                        continue
                    if v >= best_nc:
                        best_nc, g_nc = v, g
            if g_nc >= 0:
                matched[t, d] = True
                taken[g_nc] = True
            elif g_c >= 0:
                ignored[t, d] = True
    return matched, ignored, det_scores, n_gt


# pycocotools area ranges (pixels^2), bbox task
AREA_RNG = {"small": (0.0, 32.0 ** 2), "medium": (32.0 ** 2, 96.0 ** 2),
            "large": (96.0 ** 2, 1e10)}


def _box_areas(boxes):
    if boxes.numel() == 0:
        return boxes.new_zeros(0)
    return ((boxes[:, 2] - boxes[:, 0]) * (boxes[:, 3] - boxes[:, 1]))


def match_image_ranged(det_boxes, det_scores, gt_boxes, gt_crowd,
                       gt_extra_ignore, iou_thrs, max_dets=100,
                       presorted=False):
    """match_image generalized to a separate extra-ignore flag (area-range
    evaluation): sorting/preference/dtIg treat ign = crowd|extra as
    pycocotools' gtIg, while the crowd-IoU (inter/det-area) applies ONLY to
    actual crowds. Returns (matched, ignored, scores, n_gt) where n_gt
    counts non-ignored gts."""
    order = det_scores.argsort(descending=True)[:max_dets]
    det_boxes = det_boxes[order]
    det_scores = det_scores[order]
    D = det_boxes.shape[0]
    G = gt_boxes.shape[0]
    T = len(iou_thrs)
    matched = torch.zeros(T, D, dtype=torch.bool)
    ignored = torch.zeros(T, D, dtype=torch.bool)
    ign = (gt_crowd | gt_extra_ignore) if G else gt_crowd
    n_gt = int((~ign).sum()) if G else 0
    if D == 0 or G == 0:
        return matched, ignored, det_scores, n_gt
    ious = box_iou(det_boxes, gt_boxes)
    if gt_crowd.any():
        lt = torch.max(det_boxes[:, None, :2], gt_boxes[None, :, :2])
        rb = torch.min(det_boxes[:, None, 2:], gt_boxes[None, :, 2:])
        inter = (rb - lt).clamp(min=0).prod(-1)
        det_area = _box_areas(det_boxes).clamp(min=1e-9)
        ious = torch.where(gt_crowd[None, :], inter / det_area[:, None], ious)
    for t, thr in enumerate(iou_thrs):
        taken = torch.zeros(G, dtype=torch.bool)
        for d in range(D):
            best_ni = best_ig = thr
            g_ni = g_ig = -1
            for g in range(G):
                v = float(ious[d, g])
                if ign[g]:
                    # non-crowd ignored gts can be taken once; crowds never
                    if not gt_crowd[g] and taken[g]:
                        continue
                    if v >= best_ig:
                        best_ig, g_ig = v, g
                else:
                    if taken[g]:
                        continue
                    if v >= best_ni:
                        best_ni, g_ni = v, g
            if g_ni >= 0:
                matched[t, d] = True
                taken[g_ni] = True
            elif g_ig >= 0:
                ignored[t, d] = True
                if not gt_crowd[g_ig]:
                    taken[g_ig] = True
    return matched, ignored, det_scores, n_gt


def _ap_101(recall, precision):
    """COCO 101-point interpolated AP. float64 throughout: float32 recall
    flips searchsorted boundaries vs pycocotools' float64 accumulate (one
    101-pt sample ~ 3e-4 of AP). rec_thrs as i*0.01 matches np.linspace."""
    recall = recall.double()
    precision = precision.double()
    rec_thrs = torch.arange(101, dtype=torch.float64) * 0.01
    # precision envelope
    prec = precision.clone()
    for i in range(prec.numel() - 2, -1, -1):
        prec[i] = max(prec[i], prec[i + 1])
    idx = torch.searchsorted(recall, rec_thrs)
    ap = torch.zeros(101, dtype=torch.float64)
    valid = idx < prec.numel()
    ap[valid] = prec[idx[valid]]
    return float(ap.mean())


class DetEvaluator:
    """Accumulate (pred, gt) pairs per image; summarize COCO-style."""

    def __init__(self, iou_thrs=None, max_dets=100):
        self.iou_thrs = iou_thrs or COCO_IOU_THRS
        self.max_dets = max_dets
        self.items = []  # (cls, matched [T,D], ignored [T,D], scores [D], n_gt)
        # raw per-(image,class) inputs kept for area-range re-matching
        self.raw = []   # (cls, det_boxes, det_scores, gt_boxes, crowd)

    def update(self, predictions, targets):
        """predictions/targets: lists of dicts with boxes/labels(/scores);
        preds may live on GPU while targets stayed host-side — everything is
        matched on CPU."""
        for pred, gt in zip(predictions, targets):
            pred = {k: v.detach().cpu() for k, v in pred.items()}
            gt = {k: (v.cpu() if torch.is_tensor(v) else v)
                  for k, v in gt.items()}
            classes = torch.cat([pred["labels"], gt["labels"]]).unique()
            crowd = gt.get("iscrowd",
                           torch.zeros_like(gt["labels"]))
            match = match_image_native if has_ext() else match_image
            for c in classes.tolist():
                dm = pred["labels"] == c
                gm = gt["labels"] == c
                matched, ignored, scores, n_gt = match(
                    pred["boxes"][dm].cpu(), pred["scores"][dm].cpu(),
                    gt["boxes"][gm].cpu(), crowd[gm].cpu().bool(),
                    self.iou_thrs, self.max_dets)
                self.items.append((c, matched, ignored, scores, n_gt))
                self.raw.append((c, pred["boxes"][dm], pred["scores"][dm],
                                 gt["boxes"][gm], crowd[gm].bool()))

    def synchronize_between_processes(self):
        if get_world_size() > 1:
            merged = all_gather_object(self.items)
            self.items = [it for part in merged for it in part]
            merged_raw = all_gather_object(self.raw)
            self.raw = [it for part in merged_raw for it in part]

    def _class_aps(self, items_by_class):
        """per-class 101-pt APs from (matched, ignored, scores, n_gt) lists."""
        T = len(self.iou_thrs)
        ap_per_class = {}
        for c, items in items_by_class.items():
            total_gt = sum(it[3] for it in items)
            if total_gt == 0:
                continue
            scores = torch.cat([it[2] for it in items])
            order = scores.argsort(descending=True)
            aps = []
            for t in range(T):
                m = torch.cat([it[0][t] for it in items])[order]
                ig = torch.cat([it[1][t] for it in items])[order]
                keep = ~ig
                tp = m[keep].double().cumsum(0)
                fp = (~m[keep]).double().cumsum(0)
                recall = tp / total_gt
                precision = tp / (tp + fp).clamp(min=1e-9)
                aps.append(_ap_101(recall, precision))
            ap_per_class[c] = aps
        return ap_per_class

    def _ranged_items(self, lo, hi):
        """Re-match self.raw with area-range ignores (pycocotools areaRng:
        gts outside [lo,hi] are ignored; unmatched dets outside are too)."""
        by_class = {}
        for c, db, ds, gb, cr in self.raw:
            # float64 like pycocotools: area/IoU boundary comparisons at the
            # range edges must not flip on fp32 rounding
            db, gb = db.double(), gb.double()
            order = ds.argsort(descending=True, stable=True)[:self.max_dets]
            db_s, ds_s = db[order], ds[order]
            ga = _box_areas(gb)
            extra = (ga < lo) | (ga > hi) if gb.numel() else cr.clone()
            if has_ext():
                m, ig, sc, n = match_image_ranged_native(
                    db_s, ds_s, gb, cr, extra, self.iou_thrs, self.max_dets)
            else:
                m, ig, sc, n = match_image_ranged(db_s, ds_s, gb, cr, extra,
                                                  self.iou_thrs,
                                                  self.max_dets,
                                                  presorted=True)
            da = _box_areas(db_s)
            det_out = (da < lo) | (da > hi)
            ig = ig | (~m & det_out[None, :])
            by_class.setdefault(c, []).append((m, ig, sc, n))
        return by_class

    def _avg_recall(self, k):
        """AR@k: per-image top-k dets (pycocotools accumulate dtm[:, :k])."""
        by_class = {}
        for c, matched, ignored, scores, n_gt in self.items:
            by_class.setdefault(c, []).append((matched, ignored, scores, n_gt))
        recs = []
        for c, items in by_class.items():
            total_gt = sum(it[3] for it in items)
            if total_gt == 0:
                continue
            for t in range(len(self.iou_thrs)):
                tp = sum(int(it[0][t, :k].sum()) for it in items)
                recs.append(tp / total_gt)
        return float(torch.tensor(recs).mean()) if recs else 0.0

    def summarize(self):
        """Returns the COCO 12-metric style dict: mAP (IoU .5:.95), mAP50,
        mAP75, mAP_small/medium/large, AR@1/10/100, per-class AP."""
        by_class = {}
        for c, matched, ignored, scores, n_gt in self.items:
            by_class.setdefault(c, []).append((matched, ignored, scores, n_gt))
        ap_per_class = self._class_aps(by_class)
        if not ap_per_class:
            return {"mAP": 0.0, "mAP50": 0.0, "mAP75": 0.0, "per_class": {}}
        all_aps = torch.tensor(list(ap_per_class.values()))  # C, T

        def at_thr(thr):
            try:
                return float(all_aps[:, self.iou_thrs.index(thr)].mean())
            except ValueError:  # threshold not in this evaluator's list
                return 0.0
        out = {
            "mAP": float(all_aps.mean()),
            "mAP50": at_thr(0.5),
            "mAP75": at_thr(0.75),
            "per_class": {c: float(torch.tensor(a).mean())
                          for c, a in ap_per_class.items()},
        }
        for name, (lo, hi) in AREA_RNG.items():
            aps_r = self._class_aps(self._ranged_items(lo, hi))
            out[f"mAP_{name}"] = float(torch.tensor(
                list(aps_r.values())).mean()) if aps_r else 0.0
        for k in (1, 10, 100):
            out[f"AR{k}"] = self._avg_recall(min(k, self.max_dets))
        return out


def voc_ap(recall, precision, use_07_metric=False):
    """VOC AP (ref YOLOX yolox/evaluators/voc_eval.py)."""
    if use_07_metric:
        ap = 0.0
        for t in torch.arange(0.0, 1.1, 0.1):
            p = precision[recall >= t]
            ap += (float(p.max()) if p.numel() else 0.0) / 11
        return ap
    # all-points interpolation
    mrec = torch.cat([torch.zeros(1), recall, torch.ones(1)])
    mpre = torch.cat([torch.zeros(1), precision, torch.zeros(1)])
    for i in range(mpre.numel() - 2, -1, -1):
        mpre[i] = max(mpre[i], mpre[i + 1])
    idx = (mrec[1:] != mrec[:-1]).nonzero().flatten()
    return float(((mrec[idx + 1] - mrec[idx]) * mpre[idx + 1]).sum())
