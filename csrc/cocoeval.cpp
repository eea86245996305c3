// Fast COCO-style detection matching (CPU, C++).
//
// Reference parity: detection/YOLOX/yolox/layers/csrc/cocoeval/cocoeval.{h,cpp}
// (EvaluateImages / Accumulate, ~10x pycocotools) — re-designed to back this
// repo's DetEvaluator (deeplearning_amd/engine/det_eval.py): the greedy
// per-IoU-threshold matcher below is the hot loop; PR accumulation stays in
// Python (vectorized torch). Python match_image is the reference
// implementation the parity test compares against.
#include <torch/extension.h>

#include <algorithm>
#include <cmath>
#include <vector>

namespace {

// pycocotools protocol (maskUtils.iou): for iscrowd gt the denominator is
// the DETECTION's area, not the union.
inline float iou_xyxy(const float* a, const float* b, bool crowd) {
  const float ix1 = std::max(a[0], b[0]);
  const float iy1 = std::max(a[1], b[1]);
  const float ix2 = std::min(a[2], b[2]);
  const float iy2 = std::min(a[3], b[3]);
  const float iw = std::max(0.f, ix2 - ix1);
  const float ih = std::max(0.f, iy2 - iy1);
  const float inter = iw * ih;
  const float area_a = std::max(0.f, a[2] - a[0]) * std::max(0.f, a[3] - a[1]);
  const float area_b = std::max(0.f, b[2] - b[0]) * std::max(0.f, b[3] - b[1]);
  const float uni = crowd ? area_a : area_a + area_b - inter;
  return uni > 0.f ? inter / uni : 0.f;
}

}  // namespace

// Greedy COCO matching for one image+class. Inputs are float32 CPU tensors:
// det_boxes [D,4] xyxy, det_scores [D], gt_boxes [G,4], gt_crowd [G] (bool).
// Optional gt_extra_ignore [G] (bool) marks area-range-ignored gts
// (pycocotools areaRng): preference/dtIg/n_gt treat ign = crowd|extra as
// gtIg, while the crowd-IoU denominator applies only to actual crowds, and
// a non-crowd ignored gt can be taken once (gtm blocks re-matching).
// Returns (matched [T,D] bool, ignored [T,D] bool, sorted_scores [Dk], n_gt)
// with detections sorted score-descending and truncated to max_dets —
// identical protocol to DetEvaluator.match_image(_ranged).
std::vector<torch::Tensor> cocoeval_match_image(
    torch::Tensor det_boxes, torch::Tensor det_scores, torch::Tensor gt_boxes,
    torch::Tensor gt_crowd, torch::Tensor iou_thrs, int64_t max_dets,
    c10::optional<torch::Tensor> gt_extra_ignore) {
  TORCH_CHECK(det_boxes.device().is_cpu(), "cocoeval runs on CPU tensors");
  auto order = det_scores.argsort(0, /*descending=*/true);
  if (order.numel() > max_dets) order = order.slice(0, 0, max_dets);
  auto boxes = det_boxes.index_select(0, order).contiguous();
  auto scores = det_scores.index_select(0, order).contiguous();
  auto gt = gt_boxes.contiguous();
  auto crowd = gt_crowd.to(torch::kBool).contiguous();
  auto thrs = iou_thrs.to(torch::kFloat).contiguous();
  torch::Tensor ign = crowd;
  if (gt_extra_ignore.has_value() && gt_extra_ignore->numel() > 0)
    ign = (crowd | gt_extra_ignore->to(torch::kBool)).contiguous();

  const int64_t D = boxes.size(0);
  const int64_t G = gt.size(0);
  const int64_t T = thrs.size(0);
  auto matched = torch::zeros({T, D}, torch::kBool);
  auto ignored = torch::zeros({T, D}, torch::kBool);
  int64_t n_gt = G > 0 ? (G - ign.sum().item<int64_t>()) : 0;
  if (D == 0 || G == 0)
    return {matched, ignored, scores, torch::tensor(n_gt)};

  // IoU matrix once
  const bool* cp = crowd.data_ptr<bool>();
  std::vector<float> ious((size_t)D * G);
  const float* bp = boxes.data_ptr<float>();
  const float* gp = gt.data_ptr<float>();
  for (int64_t d = 0; d < D; ++d)
    for (int64_t g = 0; g < G; ++g)
      ious[d * G + g] = iou_xyxy(bp + d * 4, gp + g * 4, cp[g]);

  const float* tp = thrs.data_ptr<float>();
  const bool* ip = ign.data_ptr<bool>();
  auto m_acc = matched.accessor<bool, 2>();
  auto i_acc = ignored.accessor<bool, 2>();
  std::vector<char> taken(G);
  for (int64_t t = 0; t < T; ++t) {
    std::fill(taken.begin(), taken.end(), 0);
    for (int64_t d = 0; d < D; ++d) {
      // pycocotools semantics (gts sorted ignore-last + break): ANY
      // non-ignored match beats any ignored match regardless of IoU; ties
      // within a class of gts keep the later one.
      float best_ni = tp[t], best_ig = tp[t];
      int64_t g_ni = -1, g_ig = -1;
      for (int64_t g = 0; g < G; ++g) {
        const float v = ious[d * G + g];
        if (ip[g]) {
          if (!cp[g] && taken[g]) continue;  // taken non-crowd ignored gt
          if (v >= best_ig) { best_ig = v; g_ig = g; }
        } else {
          if (taken[g]) continue;
          if (v >= best_ni) { best_ni = v; g_ni = g; }
        }
      }
      if (g_ni >= 0) {
        m_acc[t][d] = true;
        taken[g_ni] = 1;
      } else if (g_ig >= 0) {
        i_acc[t][d] = true;
        if (!cp[g_ig]) taken[g_ig] = 1;  // gtm blocks re-matching
      }
    }
  }
  return {matched, ignored, scores, torch::tensor(n_gt)};
}
