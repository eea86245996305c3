#include "hip/hip_runtime.h"
// Sigmoid focal loss, elementwise over logits, fwd + bwd.
// Reference: RetinaNet network_files/losses.py:5-50, FCOS models/loss.py:344-364.
//   p = sigmoid(x); ce = BCEwithlogits(x, t)
//   loss = alpha_t * (1 - p_t)^gamma * ce
#include "common.h"

namespace dla {

template <typename dev_t>
__global__ void focal_fwd_kernel(const dev_t* __restrict__ logits,
                                 const dev_t* __restrict__ targets,
                                 float* __restrict__ loss, int64_t n,
                                 float alpha, float gamma) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float x = to_f32(logits[i]);
    const float t = to_f32(targets[i]);
    // numerically stable BCE-with-logits
    const float ce = fmaxf(x, 0.f) - x * t + logf(1.f + expf(-fabsf(x)));
    const float p = 1.f / (1.f + expf(-x));
    const float pt = p * t + (1.f - p) * (1.f - t);
    float l = ce * powf(1.f - pt, gamma);
    if (alpha >= 0.f) l *= alpha * t + (1.f - alpha) * (1.f - t);
    loss[i] = l;
  }
}

template <typename dev_t>
__global__ void focal_bwd_kernel(const dev_t* __restrict__ logits,
                                 const dev_t* __restrict__ targets,
                                 const float* __restrict__ dloss,
                                 dev_t* __restrict__ dlogits, int64_t n,
                                 float alpha, float gamma) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float x = to_f32(logits[i]);
    const float t = to_f32(targets[i]);
    const float p = 1.f / (1.f + expf(-x));
    const float pt = p * t + (1.f - p) * (1.f - t);
    const float ce = fmaxf(x, 0.f) - x * t + logf(1.f + expf(-fabsf(x)));
    const float one_m_pt = fmaxf(1.f - pt, 1e-12f);
    // d pt/dx = (2t-1) * p * (1-p);   d ce/dx = p - t
    const float dpt_dx = (2.f * t - 1.f) * p * (1.f - p);
    float g = powf(one_m_pt, gamma) * (p - t) -
              gamma * powf(one_m_pt, gamma - 1.f) * dpt_dx * ce;
    if (alpha >= 0.f) g *= alpha * t + (1.f - alpha) * (1.f - t);
    dlogits[i] = from_f32<dev_t>(g * dloss[i]);
  }
}

}  // namespace dla

torch::Tensor focal_loss_fwd(torch::Tensor logits, torch::Tensor targets,
                             double alpha, double gamma) {
  DLA_CHECK_INPUT(logits); DLA_CHECK_INPUT(targets);
  auto loss = torch::empty(logits.sizes(), logits.options().dtype(torch::kFloat));
  const int64_t n = logits.numel();
  DLA_DISPATCH_FLOAT_TYPES(logits.scalar_type(), "focal_fwd", [&] {
    hipLaunchKernelGGL((dla::focal_fwd_kernel<dev_t>), dim3(dla::grid_1d(n, 256)),
                       dim3(256), 0, dla::stream(),
                       (const dev_t*)logits.data_ptr(),
                       (const dev_t*)targets.data_ptr(), loss.data_ptr<float>(), n,
                       (float)alpha, (float)gamma);
  });
  HIP_CHECK_ERR();
  return loss;
}

torch::Tensor focal_loss_bwd(torch::Tensor dloss, torch::Tensor logits,
                             torch::Tensor targets, double alpha, double gamma) {
  DLA_CHECK_INPUT(logits); DLA_CHECK_INPUT(targets);
  auto dl32 = dloss.to(torch::kFloat).contiguous();
  auto dlogits = torch::empty_like(logits);
  const int64_t n = logits.numel();
  DLA_DISPATCH_FLOAT_TYPES(logits.scalar_type(), "focal_bwd", [&] {
    hipLaunchKernelGGL((dla::focal_bwd_kernel<dev_t>), dim3(dla::grid_1d(n, 256)),
                       dim3(256), 0, dla::stream(),
                       (const dev_t*)logits.data_ptr(),
                       (const dev_t*)targets.data_ptr(), dl32.data_ptr<float>(),
                       (dev_t*)dlogits.data_ptr(), n, (float)alpha, (float)gamma);
  });
  HIP_CHECK_ERR();
  return dlogits;
}
