// Fused softmax cross-entropy with label smoothing (B, C) + backward.
// Covers the reference's nn.CrossEntropyLoss / timm LabelSmoothingCrossEntropy /
// SoftTargetCrossEntropy call sites (swin main.py:111-117, everywhere else).
// Hard-label path (int64 targets) and soft-target path (B,C probabilities).
//
// The loss MEAN is folded into the forward kernel: each row's loss is
// atomicAdd-ed into stats[0] and valid rows counted into stats[1], so the
// wrapper does one division instead of a sum + mask + clamp chain, and the
// backward kernel reads its grad scale from device memory (no host sync,
// hipGraph-capturable).
#include <cstdlib>

#include "common.h"
#include "vec.h"

namespace dla {

// stats = {loss_sum, valid_count} (pre-zeroed by the wrapper)
__device__ __forceinline__ void ce_accumulate(float* stats, float row_loss,
                                              bool valid) {
  if (valid) {
    atomicAdd(stats + 0, row_loss);
    atomicAdd(stats + 1, 1.f);
  }
}

// one block per row; accumulates loss into stats and saves lse for backward
template <typename dev_t>
__global__ void ce_fwd_kernel(const dev_t* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const dev_t* __restrict__ soft_target,
                              float* __restrict__ stats, float* __restrict__ lse_out,
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
        ce_accumulate(stats, lse - ssum, true);
        lse_out[row] = lse;
      }
    } else {
      const int64_t t = target[row];
      if (smoothing > 0.f) ssum = block_reduce_sum(ssum, smem);
      if (threadIdx.x == 0) {
        if (t != ignore_index) {
          const float zt = to_f32(lr[t]);
          // (1-eps)*(lse - z_t) + eps * (lse - mean(z))
          ce_accumulate(stats,
                        (1.f - smoothing) * (lse - zt) +
                            smoothing * (lse - ssum / C),
                        true);
        }
        lse_out[row] = lse;
      }
    }
  }
}

// dlogits = (softmax - q) * grad_out/valid ; q = one-hot smoothed or soft
// target. grad_out (upstream grad of the scalar mean) and stats live on
// device so backward never syncs the host.
template <typename dev_t>
__global__ void ce_bwd_kernel(const dev_t* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const dev_t* __restrict__ soft_target,
                              const float* __restrict__ lse,
                              dev_t* __restrict__ dlogits, int B, int C,
                              float smoothing, int64_t ignore_index,
                              const float* __restrict__ grad_out,
                              const float* __restrict__ stats) {
  const float grad_scale = grad_out[0] / fmaxf(stats[1], 1.f);
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

// ---- wave-per-row variants (C % V == 0, C <= 64*V*MAX_PL) ------------------
// One 64-lane wavefront per row: shuffle-only reductions, 16B vector loads,
// no __syncthreads (the block-per-row kernels above stay as the fallback).
template <typename dev_t, int V>
__global__ __launch_bounds__(256)
void ce_fwd_wave_kernel(const dev_t* __restrict__ logits,
                        const int64_t* __restrict__ target,
                        const dev_t* __restrict__ soft_target,
                        float* __restrict__ stats, float* __restrict__ lse_out,
                        int B, int C, float smoothing, int64_t ignore_index) {
  constexpr int MAX_PL = 8;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  Vec<dev_t, V> zv[MAX_PL];
  for (int row = blockIdx.x * nwaves + wave; row < B;
       row += gridDim.x * nwaves) {
    const dev_t* lr = logits + (int64_t)row * C;
    float mx = -INFINITY;
#pragma unroll
    for (int p = 0; p < MAX_PL; ++p) {
      const int i = lane * V + p * 64 * V;
      if (i < C) {
        zv[p] = vload<dev_t, V>(lr + i);
#pragma unroll
        for (int j = 0; j < V; ++j) mx = fmaxf(mx, to_f32(zv[p].v[j]));
      }
    }
    mx = wave_reduce_max(mx);
    float se = 0.f, ssum = 0.f;
#pragma unroll
    for (int p = 0; p < MAX_PL; ++p) {
      const int i = lane * V + p * 64 * V;
      if (i < C) {
        Vec<dev_t, V> qv;
        if (soft_target != nullptr)
          qv = vload<dev_t, V>(soft_target + (int64_t)row * C + i);
#pragma unroll
        for (int j = 0; j < V; ++j) {
          const float z = to_f32(zv[p].v[j]);
          se += __expf(z - mx);
          if (soft_target != nullptr) ssum += to_f32(qv.v[j]) * z;
          else if (smoothing > 0.f) ssum += z;
        }
      }
    }
    se = wave_reduce_sum(se);
    const float lse = mx + __logf(se);
    if (soft_target != nullptr) {
      ssum = wave_reduce_sum(ssum);
      if (lane == 0) {
        ce_accumulate(stats, lse - ssum, true);
        lse_out[row] = lse;
      }
    } else {
      if (smoothing > 0.f) ssum = wave_reduce_sum(ssum);
      if (lane == 0) {
        const int64_t t = target[row];
        if (t != ignore_index) {
          const float zt = to_f32(lr[t]);
          ce_accumulate(stats,
                        (1.f - smoothing) * (lse - zt) +
                            smoothing * (lse - ssum / C),
                        true);
        }
        lse_out[row] = lse;
      }
    }
  }
}

template <typename dev_t, int V>
__global__ __launch_bounds__(256)
void ce_bwd_wave_kernel(const dev_t* __restrict__ logits,
                        const int64_t* __restrict__ target,
                        const dev_t* __restrict__ soft_target,
                        const float* __restrict__ lse,
                        dev_t* __restrict__ dlogits, int B, int C,
                        float smoothing, int64_t ignore_index,
                        const float* __restrict__ grad_out,
                        const float* __restrict__ stats) {
  constexpr int MAX_PL = 8;
  const float grad_scale = grad_out[0] / fmaxf(stats[1], 1.f);
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  for (int row = blockIdx.x * nwaves + wave; row < B;
       row += gridDim.x * nwaves) {
    const dev_t* lr = logits + (int64_t)row * C;
    dev_t* dr = dlogits + (int64_t)row * C;
    const float l = lse[row];
    const int64_t t = soft_target == nullptr ? target[row] : -1;
    const bool ignored = (soft_target == nullptr) && (t == ignore_index);
#pragma unroll
    for (int p = 0; p < MAX_PL; ++p) {
      const int i = lane * V + p * 64 * V;
      if (i < C) {
        Vec<dev_t, V> out;
        if (ignored) {
#pragma unroll
          for (int j = 0; j < V; ++j) out.v[j] = from_f32<dev_t>(0.f);
        } else {
          Vec<dev_t, V> zv = vload<dev_t, V>(lr + i);
          Vec<dev_t, V> qv;
          if (soft_target != nullptr)
            qv = vload<dev_t, V>(soft_target + (int64_t)row * C + i);
#pragma unroll
          for (int j = 0; j < V; ++j) {
            const float prob = __expf(to_f32(zv.v[j]) - l);
            const float q = soft_target != nullptr
                ? to_f32(qv.v[j])
                : ((i + j == (int)t ? 1.f - smoothing : 0.f) + smoothing / C);
            out.v[j] = from_f32<dev_t>((prob - q) * grad_scale);
          }
        }
        vstore<dev_t, V>(dr + i, out);
      }
    }
  }
}

}  // namespace dla

// Returns {stats[2] fp32 = {loss_sum, valid_count}, lse[B] fp32}
std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          c10::optional<torch::Tensor> target,
                                          c10::optional<torch::Tensor> soft_target,
                                          double smoothing, int64_t ignore_index) {
  DLA_CHECK_INPUT(logits);
  const int B = (int)logits.size(0), C = (int)logits.size(1);
  auto stats = torch::zeros({2}, logits.options().dtype(torch::kFloat));
  auto lse = torch::empty({B}, logits.options().dtype(torch::kFloat));
  const int grid = (int)std::min<int64_t>(B, dla::kMaxGrid);
  DLA_DISPATCH_FLOAT_TYPES(logits.scalar_type(), "softmax_ce_fwd", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    if (C % VMAX == 0 && C <= 64 * VMAX * 8) {
      const int g = (int)std::min<int64_t>((B + 3) / 4, dla::kMaxGrid);
      hipLaunchKernelGGL(
          (dla::ce_fwd_wave_kernel<dev_t, VMAX>), dim3(g), dim3(256), 0,
          dla::stream(), (const dev_t*)logits.data_ptr(),
          target.has_value() ? target->data_ptr<int64_t>() : nullptr,
          soft_target.has_value() ? (const dev_t*)soft_target->data_ptr()
                                  : nullptr,
          stats.data_ptr<float>(), lse.data_ptr<float>(), B, C,
          (float)smoothing, ignore_index);
      return;
    }
    hipLaunchKernelGGL(
        (dla::ce_fwd_kernel<dev_t>), dim3(grid), dim3(256), 0, dla::stream(),
        (const dev_t*)logits.data_ptr(),
        target.has_value() ? target->data_ptr<int64_t>() : nullptr,
        soft_target.has_value() ? (const dev_t*)soft_target->data_ptr() : nullptr,
        stats.data_ptr<float>(), lse.data_ptr<float>(), B, C, (float)smoothing,
        ignore_index);
  });
  HIP_CHECK_ERR();
  return {stats, lse};
}

torch::Tensor softmax_ce_bwd(torch::Tensor logits,
                             c10::optional<torch::Tensor> target,
                             c10::optional<torch::Tensor> soft_target,
                             torch::Tensor lse, double smoothing,
                             int64_t ignore_index, torch::Tensor grad_out,
                             torch::Tensor stats) {
  DLA_CHECK_INPUT(logits);
  const int B = (int)logits.size(0), C = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  const int grid = (int)std::min<int64_t>(B, dla::kMaxGrid);
  DLA_DISPATCH_FLOAT_TYPES(logits.scalar_type(), "softmax_ce_bwd", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    if (C % VMAX == 0 && C <= 64 * VMAX * 8) {
      const int g = (int)std::min<int64_t>((B + 3) / 4, dla::kMaxGrid);
      hipLaunchKernelGGL(
          (dla::ce_bwd_wave_kernel<dev_t, VMAX>), dim3(g), dim3(256), 0,
          dla::stream(), (const dev_t*)logits.data_ptr(),
          target.has_value() ? target->data_ptr<int64_t>() : nullptr,
          soft_target.has_value() ? (const dev_t*)soft_target->data_ptr()
                                  : nullptr,
          lse.data_ptr<float>(), (dev_t*)dlogits.data_ptr(), B, C,
          (float)smoothing, ignore_index, grad_out.data_ptr<float>(),
          stats.data_ptr<float>());
      return;
    }
    hipLaunchKernelGGL(
        (dla::ce_bwd_kernel<dev_t>), dim3(grid), dim3(256), 0, dla::stream(),
        (const dev_t*)logits.data_ptr(),
        target.has_value() ? target->data_ptr<int64_t>() : nullptr,
        soft_target.has_value() ? (const dev_t*)soft_target->data_ptr() : nullptr,
        lse.data_ptr<float>(), (dev_t*)dlogits.data_ptr(), B, C, (float)smoothing,
        ignore_index, grad_out.data_ptr<float>(), stats.data_ptr<float>());
  });
  HIP_CHECK_ERR();
  return dlogits;
}
