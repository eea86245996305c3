// Fused softmax cross-entropy with label smoothing (B, C) + backward.
// Covers the reference's nn.CrossEntropyLoss / timm LabelSmoothingCrossEntropy /
// SoftTargetCrossEntropy call sites (swin main.py:111-117, everywhere else).
// Hard-label path (int64 targets) and soft-target path (B,C probabilities).
#include "common.h"
#include "vec.h"

namespace dla {

// one block per row; returns per-row loss and saves lse for backward
template <typename dev_t>
__global__ void ce_fwd_kernel(const dev_t* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const dev_t* __restrict__ soft_target,
                              float* __restrict__ loss, float* __restrict__ lse_out,
                              int B, int C, float smoothing, int64_t ignore_index) {
  __shared__ float smem[16];
  for (int row = blockIdx.x; row < B; row += gridDim.x) {
    const dev_t* lr = logits + (int64_t)row * C;
    float mx = -INFINITY;
    for (int i = threadIdx.x; i < C; i += blockDim.x)
      mx = fmaxf(mx, to_f32(lr[i]));
    mx = block_reduce_max(mx, smem);
    float se = 0.f, ssum = 0.f;  // ssum = sum over classes of q_c * z_c
    for (int i = threadIdx.x; i < C; i += blockDim.x) {
      const float z = to_f32(lr[i]);
      se += expf(z - mx);
      if (soft_target != nullptr)
        ssum += to_f32(soft_target[(int64_t)row * C + i]) * z;
      else if (smoothing > 0.f)
        ssum += z;  // uniform part; target part added below
    }
    se = block_reduce_sum(se, smem);
    const float lse = mx + logf(se);
    if (soft_target != nullptr) {
      ssum = block_reduce_sum(ssum, smem);
      if (threadIdx.x == 0) {
        loss[row] = lse - ssum;
        lse_out[row] = lse;
      }
    } else {
      const int64_t t = target[row];
      if (smoothing > 0.f) ssum = block_reduce_sum(ssum, smem);
      if (threadIdx.x == 0) {
        if (t == ignore_index) {
          loss[row] = 0.f;
        } else {
          const float zt = to_f32(lr[t]);
          // (1-eps)*(lse - z_t) + eps * (lse - mean(z))
          loss[row] = (1.f - smoothing) * (lse - zt) +
                      smoothing * (lse - ssum / C);
        }
        lse_out[row] = lse;
      }
    }
  }
}

// dlogits = (softmax - q) * dloss_row ; q = one-hot smoothed or soft target
template <typename dev_t>
__global__ void ce_bwd_kernel(const dev_t* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const dev_t* __restrict__ soft_target,
                              const float* __restrict__ lse,
                              dev_t* __restrict__ dlogits, int B, int C,
                              float smoothing, int64_t ignore_index,
                              float grad_scale) {
  for (int row = blockIdx.x; row < B; row += gridDim.x) {
    const dev_t* lr = logits + (int64_t)row * C;
    dev_t* dr = dlogits + (int64_t)row * C;
    const float l = lse[row];
    const bool ignored =
        (soft_target == nullptr) && (target[row] == ignore_index);
    for (int i = threadIdx.x; i < C; i += blockDim.x) {
      if (ignored) {
        dr[i] = from_f32<dev_t>(0.f);
        continue;
      }
      const float p = expf(to_f32(lr[i]) - l);
      float q;
      if (soft_target != nullptr)
        q = to_f32(soft_target[(int64_t)row * C + i]);
      else
        q = (i == (int)target[row] ? 1.f - smoothing : 0.f) + smoothing / C;
      dr[i] = from_f32<dev_t>((p - q) * grad_scale);
    }
  }
}

}  // namespace dla

// Returns {loss[B] fp32, lse[B] fp32}
std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          c10::optional<torch::Tensor> target,
                                          c10::optional<torch::Tensor> soft_target,
                                          double smoothing, int64_t ignore_index) {
  DLA_CHECK_INPUT(logits);
  const int B = (int)logits.size(0), C = (int)logits.size(1);
  auto loss = torch::empty({B}, logits.options().dtype(torch::kFloat));
  auto lse = torch::empty({B}, logits.options().dtype(torch::kFloat));
  const int grid = (int)std::min<int64_t>(B, dla::kMaxGrid);
  DLA_DISPATCH_FLOAT_TYPES(logits.scalar_type(), "softmax_ce_fwd", [&] {
    hipLaunchKernelGGL(
        (dla::ce_fwd_kernel<dev_t>), dim3(grid), dim3(256), 0, dla::stream(),
        (const dev_t*)logits.data_ptr(),
        target.has_value() ? target->data_ptr<int64_t>() : nullptr,
        soft_target.has_value() ? (const dev_t*)soft_target->data_ptr() : nullptr,
        loss.data_ptr<float>(), lse.data_ptr<float>(), B, C, (float)smoothing,
        ignore_index);
  });
  HIP_CHECK_ERR();
  return {loss, lse};
}

torch::Tensor softmax_ce_bwd(torch::Tensor logits,
                             c10::optional<torch::Tensor> target,
                             c10::optional<torch::Tensor> soft_target,
                             torch::Tensor lse, double smoothing,
                             int64_t ignore_index, double grad_scale) {
  DLA_CHECK_INPUT(logits);
  const int B = (int)logits.size(0), C = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  const int grid = (int)std::min<int64_t>(B, dla::kMaxGrid);
  DLA_DISPATCH_FLOAT_TYPES(logits.scalar_type(), "softmax_ce_bwd", [&] {
    hipLaunchKernelGGL(
        (dla::ce_bwd_kernel<dev_t>), dim3(grid), dim3(256), 0, dla::stream(),
        (const dev_t*)logits.data_ptr(),
        target.has_value() ? target->data_ptr<int64_t>() : nullptr,
        soft_target.has_value() ? (const dev_t*)soft_target->data_ptr() : nullptr,
        lse.data_ptr<float>(), (dev_t*)dlogits.data_ptr(), B, C, (float)smoothing,
        ignore_index, (float)grad_scale);
  });
  HIP_CHECK_ERR();
  return dlogits;
}
