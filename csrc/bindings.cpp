// Python bindings for the deeplearning_amd HIP kernel library (gfx950-only).
#include <torch/extension.h>

// layernorm.hip
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd);
// elementwise.hip
torch::Tensor gelu_fwd(torch::Tensor x);
torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x);
torch::Tensor silu_fwd(torch::Tensor x);
torch::Tensor silu_bwd(torch::Tensor dy, torch::Tensor x);
torch::Tensor add_relu_fwd(torch::Tensor a, torch::Tensor b);
torch::Tensor relu_mask_bwd(torch::Tensor dy, torch::Tensor y);
// batchnorm.hip
std::vector<torch::Tensor> batchnorm_fwd(torch::Tensor x, torch::Tensor weight,
                                         torch::Tensor bias,
                                         c10::optional<torch::Tensor> running_mean,
                                         c10::optional<torch::Tensor> running_var,
                                         double momentum, double eps, bool relu);
torch::Tensor bn_apply(torch::Tensor x, torch::Tensor scale, torch::Tensor shift,
                       bool relu);
std::vector<torch::Tensor> batchnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         c10::optional<torch::Tensor> y,
                                         torch::Tensor weight, torch::Tensor mean,
                                         torch::Tensor rstd, bool relu);
std::vector<torch::Tensor> batchnorm_fwd_from_sums(
    torch::Tensor x, torch::Tensor weight, torch::Tensor bias,
    torch::Tensor sums, c10::optional<torch::Tensor> running_mean,
    c10::optional<torch::Tensor> running_var, double momentum, double eps,
    bool relu);
// conv1x1.hip
std::vector<torch::Tensor> conv1x1_fwd(torch::Tensor a, torch::Tensor b,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> scale,
                                       c10::optional<torch::Tensor> shift,
                                       c10::optional<torch::Tensor> residual,
                                       bool relu, bool want_stats);
torch::Tensor conv1x1_wgrad(torch::Tensor dy, torch::Tensor x);
std::vector<torch::Tensor> conv3x3_fwd(torch::Tensor a, torch::Tensor w9,
                                       int64_t imgH, int64_t imgW,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> scale,
                                       c10::optional<torch::Tensor> shift,
                                       c10::optional<torch::Tensor> residual,
                                       bool relu, bool want_stats);
torch::Tensor conv3x3_wgrad(torch::Tensor dy, torch::Tensor x, int64_t imgH,
                            int64_t imgW);
// reorder.hip
torch::Tensor transpose2d(torch::Tensor src);
torch::Tensor stride2_gather(torch::Tensor x);
torch::Tensor stride2_scatter(torch::Tensor dy, int64_t IH, int64_t IW);
// softmax_ce.hip
std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          c10::optional<torch::Tensor> target,
                                          c10::optional<torch::Tensor> soft_target,
                                          double smoothing, int64_t ignore_index);
torch::Tensor softmax_ce_bwd(torch::Tensor logits,
                             c10::optional<torch::Tensor> target,
                             c10::optional<torch::Tensor> soft_target,
                             torch::Tensor lse, double smoothing,
                             int64_t ignore_index, torch::Tensor grad_out,
                             torch::Tensor stats);
// focal.hip
torch::Tensor focal_loss_fwd(torch::Tensor logits, torch::Tensor targets,
                             double alpha, double gamma);
torch::Tensor focal_loss_bwd(torch::Tensor dloss, torch::Tensor logits,
                             torch::Tensor targets, double alpha, double gamma);
// boxes.hip
torch::Tensor box_iou_gpu(torch::Tensor a, torch::Tensor b, bool giou);
torch::Tensor nms_gpu(torch::Tensor boxes_sorted, double iou_threshold);
// roialign.hip
torch::Tensor roialign_fwd(torch::Tensor input, torch::Tensor rois, int64_t PH,
                           int64_t PW, double spatial_scale,
                           int64_t sampling_ratio, bool aligned);
torch::Tensor roialign_bwd(torch::Tensor grad_out, torch::Tensor rois, int64_t N,
                           int64_t C, int64_t H, int64_t W, double spatial_scale,
                           int64_t sampling_ratio, bool aligned);
// attention.hip
std::vector<torch::Tensor> attn_fwd(torch::Tensor qkv, int64_t num_heads,
                                    double scale,
                                    c10::optional<torch::Tensor> bias,
                                    c10::optional<torch::Tensor> mask,
                                    bool save_p, bool save_stats);
std::vector<torch::Tensor> attn_bwd(torch::Tensor qkv, torch::Tensor dout,
                                    torch::Tensor stats, torch::Tensor drow,
                                    int64_t num_heads, double scale);
std::vector<torch::Tensor> attn_fwd_cosine(torch::Tensor qkv, int64_t num_heads,
                              torch::Tensor logit_scale,
                              c10::optional<torch::Tensor> bias,
                              c10::optional<torch::Tensor> mask, bool save_p);
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B);
// my_add.cpp (custom-op export tutorial)
torch::Tensor my_add(torch::Tensor a, torch::Tensor b);
// cocoeval.cpp (CPU)
std::vector<torch::Tensor> cocoeval_match_image(
    torch::Tensor det_boxes, torch::Tensor det_scores, torch::Tensor gt_boxes,
    torch::Tensor gt_crowd, torch::Tensor iou_thrs, int64_t max_dets,
    c10::optional<torch::Tensor> gt_extra_ignore);
// window.hip
torch::Tensor window_partition_fwd(torch::Tensor x, int64_t ws, int64_t shift);
torch::Tensor window_partition_bwd(torch::Tensor grad, int64_t B, int64_t H,
                                   int64_t W, int64_t ws, int64_t shift);
torch::Tensor window_merge_fwd(torch::Tensor windows, int64_t B, int64_t H,
                               int64_t W, int64_t ws, int64_t shift);
torch::Tensor window_merge_bwd(torch::Tensor grad, int64_t ws, int64_t shift);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "deeplearning_amd gfx950 HIP kernels";
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("silu_fwd", &silu_fwd);
  m.def("silu_bwd", &silu_bwd);
  m.def("add_relu_fwd", &add_relu_fwd);
  m.def("relu_mask_bwd", &relu_mask_bwd);
  m.def("batchnorm_fwd", &batchnorm_fwd);
  m.def("batchnorm_fwd_from_sums", &batchnorm_fwd_from_sums);
  m.def("bn_apply", &bn_apply);
  m.def("batchnorm_bwd", &batchnorm_bwd);
  m.def("conv1x1_fwd", &conv1x1_fwd, py::arg("a"), py::arg("b"),
        py::arg("bias") = py::none(), py::arg("scale") = py::none(),
        py::arg("shift") = py::none(), py::arg("residual") = py::none(),
        py::arg("relu") = false, py::arg("want_stats") = false);
  m.def("conv1x1_wgrad", &conv1x1_wgrad);
  m.def("conv3x3_fwd", &conv3x3_fwd, py::arg("a"), py::arg("w9"),
        py::arg("imgH"), py::arg("imgW"), py::arg("bias") = py::none(),
        py::arg("scale") = py::none(), py::arg("shift") = py::none(),
        py::arg("residual") = py::none(), py::arg("relu") = false,
        py::arg("want_stats") = false);
  m.def("conv3x3_wgrad", &conv3x3_wgrad);
  m.def("transpose2d", &transpose2d);
  m.def("stride2_gather", &stride2_gather);
  m.def("stride2_scatter", &stride2_scatter);
  m.def("softmax_ce_fwd", &softmax_ce_fwd);
  m.def("softmax_ce_bwd", &softmax_ce_bwd);
  m.def("focal_loss_fwd", &focal_loss_fwd);
  m.def("focal_loss_bwd", &focal_loss_bwd);
  m.def("box_iou", &box_iou_gpu);
  m.def("nms", &nms_gpu);
  m.def("roialign_fwd", &roialign_fwd);
  m.def("roialign_bwd", &roialign_bwd);
  m.def("attn_fwd", &attn_fwd, py::arg("qkv"), py::arg("num_heads"),
        py::arg("scale"), py::arg("bias") = py::none(),
        py::arg("mask") = py::none(), py::arg("save_p") = false,
        py::arg("save_stats") = false);
  m.def("attn_bwd", &attn_bwd);
  m.def("attn_fwd_cosine", &attn_fwd_cosine, py::arg("qkv"),
        py::arg("num_heads"), py::arg("logit_scale"),
        py::arg("bias") = py::none(), py::arg("mask") = py::none(),
        py::arg("save_p") = false);
  m.def("mfma_probe", &mfma_probe);
  m.def("cocoeval_match_image", &cocoeval_match_image, py::arg("det_boxes"),
        py::arg("det_scores"), py::arg("gt_boxes"), py::arg("gt_crowd"),
        py::arg("iou_thrs"), py::arg("max_dets"),
        py::arg("gt_extra_ignore") = py::none());
  m.def("my_add", &my_add);
  m.def("window_partition_fwd", &window_partition_fwd);
  m.def("window_partition_bwd", &window_partition_bwd);
  m.def("window_merge_fwd", &window_merge_fwd);
  m.def("window_merge_bwd", &window_merge_bwd);
}
