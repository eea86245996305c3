// Training-mode BatchNorm2d (NCHW) with optional fused ReLU, plus frozen-BN
// fused apply. fp32/bf16 input, fp32 statistics and running stats.
//
// Reference call sites: nn.BatchNorm2d in every conv net
// (classification/resnet/models/networks.py:45), FrozenBatchNorm2d
// (detection/fasterRcnn/models/backbone/resnet50_fpn.py:5, FPN/fpn_model.py:8).
// The fused ReLU covers the conv->bn->relu chain that dominates ResNet.
#include "common.h"
#include "vec.h"

namespace dla {

// ---- pass 1: per-channel sum / sumsq ---------------------------------------
template <typename dev_t>
__global__ void bn_stats_kernel(const dev_t* __restrict__ x, float* __restrict__ sums,
                                int N, int C, int64_t HW) {
  __shared__ float smem[16];
  const int c = blockIdx.x;
  const int64_t total = (int64_t)N * HW;
  float sum = 0.f, sumsq = 0.f;
  for (int64_t m = (int64_t)blockIdx.y * blockDim.x + threadIdx.x; m < total;
       m += (int64_t)gridDim.y * blockDim.x) {
    const int64_t n = m / HW, hw = m % HW;
    const float f = to_f32(x[(n * C + c) * HW + hw]);
    sum += f;
    sumsq += f * f;
  }
  sum = block_reduce_sum(sum, smem);
  sumsq = block_reduce_sum(sumsq, smem);
  if (threadIdx.x == 0) {
    atomicAdd(&sums[c], sum);
    atomicAdd(&sums[C + c], sumsq);
  }
}

// ---- finalize: mean/rstd + running-stat update + scale/shift ---------------
__global__ void bn_finalize_kernel(const float* __restrict__ sums,
                                   const float* __restrict__ weight,
                                   const float* __restrict__ bias,
                                   float* __restrict__ mean, float* __restrict__ rstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float* __restrict__ scale, float* __restrict__ shift,
                                   int C, float count, float eps, float momentum) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float mu = sums[c] / count;
  const float var = fmaxf(sums[C + c] / count - mu * mu, 0.f);
  const float rs = rsqrtf(var + eps);
  mean[c] = mu;
  rstd[c] = rs;
  if (running_mean != nullptr && momentum > 0.f) {
    const float unbiased = var * count / fmaxf(count - 1.f, 1.f);
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
  const float sc = weight[c] * rs;
  scale[c] = sc;
  shift[c] = bias[c] - mu * sc;
}

// ---- pass 2 (also frozen-BN apply): y = x*scale[c] + shift[c] (+ReLU) ------
template <typename dev_t, int V, bool RELU>
__global__ void bn_apply_kernel(const dev_t* __restrict__ x,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                dev_t* __restrict__ y, int C, int64_t HW,
                                int64_t n_total) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * V;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * V;
       i < n_total; i += stride) {
    const int c = (int)((i / HW) % C);  // V divides HW, so one channel per vec
    const float sc = scale[c], sh = shift[c];
    Vec<dev_t, V> xv = vload<dev_t, V>(x + i);
    Vec<dev_t, V> yv;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float f = to_f32(xv.v[j]) * sc + sh;
      yv.v[j] = from_f32<dev_t>(RELU ? fmaxf(f, 0.f) : f);
    }
    vstore<dev_t, V>(y + i, yv);
  }
}

// ---- backward pass 1: per-channel sum(dy), sum(dy * xhat) ------------------
template <typename dev_t, bool RELU>
__global__ void bn_bwd_stats_kernel(const dev_t* __restrict__ dy,
                                    const dev_t* __restrict__ x,
                                    const dev_t* __restrict__ y,  // for ReLU mask
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ sums,
                                    int N, int C, int64_t HW) {
  __shared__ float smem[16];
  const int c = blockIdx.x;
  const float mu = mean[c], rs = rstd[c];
  const int64_t total = (int64_t)N * HW;
  float s_dy = 0.f, s_dyxh = 0.f;
  for (int64_t m = (int64_t)blockIdx.y * blockDim.x + threadIdx.x; m < total;
       m += (int64_t)gridDim.y * blockDim.x) {
    const int64_t n = m / HW, hw = m % HW;
    const int64_t idx = (n * C + c) * HW + hw;
    float g = to_f32(dy[idx]);
    if (RELU && to_f32(y[idx]) <= 0.f) g = 0.f;
    const float xh = (to_f32(x[idx]) - mu) * rs;
    s_dy += g;
    s_dyxh += g * xh;
  }
  s_dy = block_reduce_sum(s_dy, smem);
  s_dyxh = block_reduce_sum(s_dyxh, smem);
  if (threadIdx.x == 0) {
    atomicAdd(&sums[c], s_dy);
    atomicAdd(&sums[C + c], s_dyxh);
  }
}

// ---- backward pass 2: dx -----------------------------------------------------
template <typename dev_t, bool RELU>
__global__ void bn_bwd_dx_kernel(const dev_t* __restrict__ dy,
                                 const dev_t* __restrict__ x,
                                 const dev_t* __restrict__ y,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 const float* __restrict__ weight,
                                 const float* __restrict__ sums,
                                 dev_t* __restrict__ dx, int C, int64_t HW,
                                 int64_t n_total, float inv_count) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_total;
       i += stride) {
    const int c = (int)((i / HW) % C);
    const float mu = mean[c], rs = rstd[c];
    float g = to_f32(dy[i]);
    if (RELU && to_f32(y[i]) <= 0.f) g = 0.f;
    const float xh = (to_f32(x[i]) - mu) * rs;
    const float m_dy = sums[c] * inv_count;
    const float m_dyxh = sums[C + c] * inv_count;
    dx[i] = from_f32<dev_t>(weight[c] * rs * (g - m_dy - xh * m_dyxh));
  }
}

// ======================= NHWC (channels_last) variants =======================
// NHWC is the fast layout on MI355X (MIOpen conv prefers it; BN reductions are
// column sums over a [N*H*W, C] row-major matrix -> fully coalesced).

// NHWC kernels share one thread mapping: flat thread t -> (c0 = (t %
// lanes_per_row)*V, row = t / lanes_per_row), rows strided by
// used_threads/lanes_per_row. Adjacent lanes read adjacent 16B channel chunks
// (fully coalesced); per-channel parameters load ONCE per thread (c0 is
// loop-invariant). Stats kernels block-reduce through LDS (ds_add_f32) and
// issue 2*C global atomics per block.

inline int64_t nhwc_used_threads(int lanes_per_row, int64_t rows,
                                 int64_t target_lanes) {
  int64_t k = target_lanes / lanes_per_row;
  if (k > rows) k = rows;
  if (k < 1) k = 1;
  return (int64_t)lanes_per_row * k;
}

template <typename dev_t, int V>
__global__ void bn_stats_nhwc_kernel(const dev_t* __restrict__ x,
                                     float* __restrict__ sums, int C,
                                     int64_t rows, int64_t used) {
  extern __shared__ float ls[];  // 2*C floats
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) ls[i] = 0.f;
  __syncthreads();
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int lanes_per_row = C / V;
  if (t < used) {
    const int c0 = (int)(t % lanes_per_row) * V;
    const int64_t rstride = used / lanes_per_row;
    float s[V], q[V];
#pragma unroll
    for (int j = 0; j < V; ++j) { s[j] = 0.f; q[j] = 0.f; }
    for (int64_t r = t / lanes_per_row; r < rows; r += rstride) {
      Vec<dev_t, V> xv = vload<dev_t, V>(x + r * C + c0);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float f = to_f32(xv.v[j]);
        s[j] += f;
        q[j] += f * f;
      }
    }
#pragma unroll
    for (int j = 0; j < V; ++j) {
      atomicAdd(&ls[c0 + j], s[j]);
      atomicAdd(&ls[C + c0 + j], q[j]);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x)
    atomicAdd(&sums[i], ls[i]);
}

template <typename dev_t, int V, bool RELU>
__global__ void bn_apply_nhwc_kernel(const dev_t* __restrict__ x,
                                     const float* __restrict__ scale,
                                     const float* __restrict__ shift,
                                     dev_t* __restrict__ y, int C, int64_t rows,
                                     int64_t used) {
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= used) return;
  const int lanes_per_row = C / V;
  const int c0 = (int)(t % lanes_per_row) * V;
  const int64_t rstride = used / lanes_per_row;
  float sc[V], sh[V];
#pragma unroll
  for (int j = 0; j < V; ++j) { sc[j] = scale[c0 + j]; sh[j] = shift[c0 + j]; }
  for (int64_t r = t / lanes_per_row; r < rows; r += rstride) {
    Vec<dev_t, V> xv = vload<dev_t, V>(x + r * C + c0);
    Vec<dev_t, V> yv;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const float f = to_f32(xv.v[j]) * sc[j] + sh[j];
      yv.v[j] = from_f32<dev_t>(RELU ? fmaxf(f, 0.f) : f);
    }
    vstore<dev_t, V>(y + r * C + c0, yv);
  }
}

template <typename dev_t, int V, bool RELU>
__global__ void bn_bwd_stats_nhwc_kernel(const dev_t* __restrict__ dy,
                                         const dev_t* __restrict__ x,
                                         const dev_t* __restrict__ y,
                                         const float* __restrict__ mean,
                                         const float* __restrict__ rstd,
                                         float* __restrict__ sums, int C,
                                         int64_t rows, int64_t used) {
  extern __shared__ float ls[];
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) ls[i] = 0.f;
  __syncthreads();
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int lanes_per_row = C / V;
  if (t < used) {
    const int c0 = (int)(t % lanes_per_row) * V;
    const int64_t rstride = used / lanes_per_row;
    float mu[V], rs[V], s_dy[V], s_dyxh[V];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      mu[j] = mean[c0 + j];
      rs[j] = rstd[c0 + j];
      s_dy[j] = 0.f;
      s_dyxh[j] = 0.f;
    }
    for (int64_t r = t / lanes_per_row; r < rows; r += rstride) {
      const int64_t base = r * C + c0;
      Vec<dev_t, V> dyv = vload<dev_t, V>(dy + base);
      Vec<dev_t, V> xv = vload<dev_t, V>(x + base);
      Vec<dev_t, V> yv;
      if (RELU) yv = vload<dev_t, V>(y + base);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float g = to_f32(dyv.v[j]);
        if (RELU && to_f32(yv.v[j]) <= 0.f) g = 0.f;
        const float xh = (to_f32(xv.v[j]) - mu[j]) * rs[j];
        s_dy[j] += g;
        s_dyxh[j] += g * xh;
      }
    }
#pragma unroll
    for (int j = 0; j < V; ++j) {
      atomicAdd(&ls[c0 + j], s_dy[j]);
      atomicAdd(&ls[C + c0 + j], s_dyxh[j]);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x)
    atomicAdd(&sums[i], ls[i]);
}

template <typename dev_t, int V, bool RELU>
__global__ void bn_bwd_dx_nhwc_kernel(const dev_t* __restrict__ dy,
                                      const dev_t* __restrict__ x,
                                      const dev_t* __restrict__ y,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ rstd,
                                      const float* __restrict__ weight,
                                      const float* __restrict__ sums,
                                      dev_t* __restrict__ dx, int C,
                                      int64_t rows, int64_t used,
                                      float inv_count) {
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= used) return;
  const int lanes_per_row = C / V;
  const int c0 = (int)(t % lanes_per_row) * V;
  const int64_t rstride = used / lanes_per_row;
  float mu[V], rs[V], wr[V], m_dy[V], m_dyxh[V];
#pragma unroll
  for (int j = 0; j < V; ++j) {
    mu[j] = mean[c0 + j];
    rs[j] = rstd[c0 + j];
    wr[j] = weight[c0 + j] * rs[j];
    m_dy[j] = sums[c0 + j] * inv_count;
    m_dyxh[j] = sums[C + c0 + j] * inv_count;
  }
  for (int64_t r = t / lanes_per_row; r < rows; r += rstride) {
    const int64_t base = r * C + c0;
    Vec<dev_t, V> dyv = vload<dev_t, V>(dy + base);
    Vec<dev_t, V> xv = vload<dev_t, V>(x + base);
    Vec<dev_t, V> yv;
    if (RELU) yv = vload<dev_t, V>(y + base);
    Vec<dev_t, V> dxv;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float g = to_f32(dyv.v[j]);
      if (RELU && to_f32(yv.v[j]) <= 0.f) g = 0.f;
      const float xh = (to_f32(xv.v[j]) - mu[j]) * rs[j];
      dxv.v[j] = from_f32<dev_t>(wr[j] * (g - m_dy[j] - xh * m_dyxh[j]));
    }
    vstore<dev_t, V>(dx + r * C + c0, dxv);
  }
}

inline bool is_nhwc(const torch::Tensor& t) {
  return t.dim() == 4 && t.is_contiguous(at::MemoryFormat::ChannelsLast) &&
         !t.is_contiguous();
}

}  // namespace dla

// Returns {y, save_mean, save_rstd}. Updates running stats in-place when given.
std::vector<torch::Tensor> batchnorm_fwd(torch::Tensor x, torch::Tensor weight,
                                         torch::Tensor bias,
                                         c10::optional<torch::Tensor> running_mean,
                                         c10::optional<torch::Tensor> running_var,
                                         double momentum, double eps, bool relu) {
  DLA_CHECK_CUDA(x);
  TORCH_CHECK(x.dim() == 4, "batchnorm_fwd expects a 4D tensor");
  const bool nhwc = dla::is_nhwc(x);
  TORCH_CHECK(nhwc || x.is_contiguous(), "batchnorm_fwd: x must be contiguous "
              "(NCHW or channels_last)");
  const int N = (int)x.size(0), C = (int)x.size(1);
  const int64_t HW = x.size(2) * x.size(3);
  auto opts_f = x.options().dtype(torch::kFloat);
  auto sums = torch::zeros({2 * C}, opts_f);
  auto mean = torch::empty({C}, opts_f);
  auto rstd = torch::empty({C}, opts_f);
  auto scale = torch::empty({C}, opts_f);
  auto shift = torch::empty({C}, opts_f);
  auto y = torch::empty_like(x);
  auto w32 = weight.to(torch::kFloat);
  auto b32 = bias.to(torch::kFloat);

  const int64_t rows = (int64_t)N * HW;
  const int ysplit = (int)std::min<int64_t>((rows + 65535) / 65536 + 1,
                                            std::max(1, 2048 / C));
  DLA_DISPATCH_FLOAT_TYPES(x.scalar_type(), "batchnorm_fwd", [&] {
    if (nhwc) {
      constexpr int VM = 16 / (int)sizeof(dev_t);
      auto stats = [&](auto vtag) {
        constexpr int V = decltype(vtag)::value;
        const int lanes = C / V;
        const int64_t used = dla::nhwc_used_threads(lanes, rows, 262144);
        const int lds = 2 * C * sizeof(float);
        hipLaunchKernelGGL((dla::bn_stats_nhwc_kernel<dev_t, V>),
                           dim3((int)((used + 255) / 256)), dim3(256), lds,
                           dla::stream(), (const dev_t*)x.data_ptr(),
                           sums.data_ptr<float>(), C, rows, used);
      };
      if (C % VM == 0) stats(std::integral_constant<int, VM>{});
      else if (C % 2 == 0) stats(std::integral_constant<int, 2>{});
      else stats(std::integral_constant<int, 1>{});
    } else {
      hipLaunchKernelGGL((dla::bn_stats_kernel<dev_t>), dim3(C, ysplit), dim3(256), 0,
                         dla::stream(), (const dev_t*)x.data_ptr(),
                         sums.data_ptr<float>(), N, C, HW);
    }
    hipLaunchKernelGGL(dla::bn_finalize_kernel, dim3((C + 255) / 256), dim3(256), 0,
                       dla::stream(), sums.data_ptr<float>(), w32.data_ptr<float>(),
                       b32.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(),
                       running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
                       running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
                       scale.data_ptr<float>(), shift.data_ptr<float>(), C,
                       (float)rows, (float)eps, (float)momentum);
    const int64_t n_total = x.numel();
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag, auto rtag) {
      constexpr int V = decltype(vtag)::value;
      constexpr bool R = decltype(rtag)::value;
      if (nhwc) {
        const int lanes = C / V;
        const int64_t used = dla::nhwc_used_threads(lanes, rows, 524288);
        hipLaunchKernelGGL((dla::bn_apply_nhwc_kernel<dev_t, V, R>),
                           dim3((int)((used + 255) / 256)), dim3(256), 0,
                           dla::stream(), (const dev_t*)x.data_ptr(),
                           scale.data_ptr<float>(), shift.data_ptr<float>(),
                           (dev_t*)y.data_ptr(), C, rows, used);
        return;
      }
      const int grid = dla::grid_1d((n_total + V - 1) / V, 256);
      if (false) ;
      else
        hipLaunchKernelGGL((dla::bn_apply_kernel<dev_t, V, R>), dim3(grid), dim3(256),
                           0, dla::stream(), (const dev_t*)x.data_ptr(),
                           scale.data_ptr<float>(), shift.data_ptr<float>(),
                           (dev_t*)y.data_ptr(), C, HW, n_total);
    };
    const bool vec_ok = nhwc ? (C % VMAX == 0) : (HW % VMAX == 0);
    if (vec_ok) {
      if (relu) launch(std::integral_constant<int, VMAX>{}, std::true_type{});
      else launch(std::integral_constant<int, VMAX>{}, std::false_type{});
    } else {
      if (relu) launch(std::integral_constant<int, 1>{}, std::true_type{});
      else launch(std::integral_constant<int, 1>{}, std::false_type{});
    }
  });
  HIP_CHECK_ERR();
  return {y, mean, rstd};
}

// Train-mode BN forward with the stats pass ALREADY DONE by the producing
// kernel (conv1x1_fwd's fused sum/sumsq epilogue): finalize + apply only —
// saves one full read pass over x vs batchnorm_fwd. sums layout matches
// bn_stats: [2C] = {sum[c], sumsq[C+c]}.
std::vector<torch::Tensor> batchnorm_fwd_from_sums(
    torch::Tensor x, torch::Tensor weight, torch::Tensor bias,
    torch::Tensor sums, c10::optional<torch::Tensor> running_mean,
    c10::optional<torch::Tensor> running_var, double momentum, double eps,
    bool relu) {
  DLA_CHECK_CUDA(x);
  TORCH_CHECK(x.dim() == 4, "batchnorm_fwd_from_sums expects a 4D tensor");
  const bool nhwc = dla::is_nhwc(x);
  TORCH_CHECK(nhwc || x.is_contiguous(),
              "batchnorm_fwd_from_sums: x must be contiguous");
  const int N = (int)x.size(0), C = (int)x.size(1);
  const int64_t HW = x.size(2) * x.size(3);
  TORCH_CHECK(sums.numel() == 2 * C, "sums must be [2C]");
  auto opts_f = x.options().dtype(torch::kFloat);
  auto mean = torch::empty({C}, opts_f);
  auto rstd = torch::empty({C}, opts_f);
  auto scale = torch::empty({C}, opts_f);
  auto shift = torch::empty({C}, opts_f);
  auto y = torch::empty_like(x);
  auto w32 = weight.to(torch::kFloat);
  auto b32 = bias.to(torch::kFloat);
  const int64_t rows = (int64_t)N * HW;
  DLA_DISPATCH_FLOAT_TYPES(x.scalar_type(), "batchnorm_fwd_from_sums", [&] {
    hipLaunchKernelGGL(dla::bn_finalize_kernel, dim3((C + 255) / 256), dim3(256),
                       0, dla::stream(), sums.data_ptr<float>(),
                       w32.data_ptr<float>(), b32.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
                       running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
                       scale.data_ptr<float>(), shift.data_ptr<float>(), C,
                       (float)rows, (float)eps, (float)momentum);
    const int64_t n_total = x.numel();
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag, auto rtag) {
      constexpr int V = decltype(vtag)::value;
      constexpr bool R = decltype(rtag)::value;
      if (nhwc) {
        const int lanes = C / V;
        const int64_t used = dla::nhwc_used_threads(lanes, rows, 524288);
        hipLaunchKernelGGL((dla::bn_apply_nhwc_kernel<dev_t, V, R>),
                           dim3((int)((used + 255) / 256)), dim3(256), 0,
                           dla::stream(), (const dev_t*)x.data_ptr(),
                           scale.data_ptr<float>(), shift.data_ptr<float>(),
                           (dev_t*)y.data_ptr(), C, rows, used);
        return;
      }
      const int grid = dla::grid_1d((n_total + V - 1) / V, 256);
      hipLaunchKernelGGL((dla::bn_apply_kernel<dev_t, V, R>), dim3(grid),
                         dim3(256), 0, dla::stream(),
                         (const dev_t*)x.data_ptr(), scale.data_ptr<float>(),
                         shift.data_ptr<float>(), (dev_t*)y.data_ptr(), C, HW,
                         n_total);
    };
    const bool vec_ok = nhwc ? (C % VMAX == 0) : (HW % VMAX == 0);
    if (vec_ok) {
      if (relu) launch(std::integral_constant<int, VMAX>{}, std::true_type{});
      else launch(std::integral_constant<int, VMAX>{}, std::false_type{});
    } else {
      if (relu) launch(std::integral_constant<int, 1>{}, std::true_type{});
      else launch(std::integral_constant<int, 1>{}, std::false_type{});
    }
  });
  HIP_CHECK_ERR();
  return {y, mean, rstd};
}

// Frozen/eval BN apply (+optional ReLU): per-channel scale/shift precomputed host-side.
torch::Tensor bn_apply(torch::Tensor x, torch::Tensor scale, torch::Tensor shift,
                       bool relu) {
  DLA_CHECK_CUDA(x);
  TORCH_CHECK(x.dim() == 4, "bn_apply expects a 4D tensor");
  const bool nhwc = dla::is_nhwc(x);
  TORCH_CHECK(nhwc || x.is_contiguous(), "bn_apply: x must be contiguous");
  const int C = (int)x.size(1);
  const int64_t HW = x.size(2) * x.size(3);
  auto y = torch::empty_like(x);
  auto sc = scale.to(torch::kFloat).contiguous();
  auto sh = shift.to(torch::kFloat).contiguous();
  const int64_t n_total = x.numel();
  DLA_DISPATCH_FLOAT_TYPES(x.scalar_type(), "bn_apply", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag, auto rtag) {
      constexpr int V = decltype(vtag)::value;
      constexpr bool R = decltype(rtag)::value;
      if (nhwc) {
        const int64_t rows = n_total / C;
        const int lanes = C / V;
        const int64_t used = dla::nhwc_used_threads(lanes, rows, 524288);
        hipLaunchKernelGGL((dla::bn_apply_nhwc_kernel<dev_t, V, R>),
                           dim3((int)((used + 255) / 256)), dim3(256), 0,
                           dla::stream(), (const dev_t*)x.data_ptr(),
                           sc.data_ptr<float>(), sh.data_ptr<float>(),
                           (dev_t*)y.data_ptr(), C, rows, used);
        return;
      }
      const int grid = dla::grid_1d((n_total + V - 1) / V, 256);
      if (false) ;
      else
        hipLaunchKernelGGL((dla::bn_apply_kernel<dev_t, V, R>), dim3(grid), dim3(256),
                           0, dla::stream(), (const dev_t*)x.data_ptr(),
                           sc.data_ptr<float>(), sh.data_ptr<float>(),
                           (dev_t*)y.data_ptr(), C, HW, n_total);
    };
    const bool vec_ok = nhwc ? (C % VMAX == 0) : (HW % VMAX == 0);
    if (vec_ok) {
      if (relu) launch(std::integral_constant<int, VMAX>{}, std::true_type{});
      else launch(std::integral_constant<int, VMAX>{}, std::false_type{});
    } else {
      if (relu) launch(std::integral_constant<int, 1>{}, std::true_type{});
      else launch(std::integral_constant<int, 1>{}, std::false_type{});
    }
  });
  HIP_CHECK_ERR();
  return y;
}

// Returns {dx, dweight, dbias}. y is required when relu=true (mask source).
std::vector<torch::Tensor> batchnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         c10::optional<torch::Tensor> y,
                                         torch::Tensor weight, torch::Tensor mean,
                                         torch::Tensor rstd, bool relu) {
  DLA_CHECK_CUDA(dy); DLA_CHECK_CUDA(x);
  const bool nhwc = dla::is_nhwc(x);
  TORCH_CHECK(nhwc || x.is_contiguous(), "batchnorm_bwd: x must be contiguous");
  if (nhwc && !dla::is_nhwc(dy)) dy = dy.contiguous(at::MemoryFormat::ChannelsLast);
  if (!nhwc) dy = dy.contiguous();
  const int N = (int)x.size(0), C = (int)x.size(1);
  const int64_t HW = x.size(2) * x.size(3);
  const int64_t rows = (int64_t)N * HW;
  auto opts_f = x.options().dtype(torch::kFloat);
  auto sums = torch::zeros({2 * C}, opts_f);
  auto dx = torch::empty_like(x);
  auto w32 = weight.to(torch::kFloat);
  const int ysplit = (int)std::min<int64_t>((rows + 65535) / 65536 + 1,
                                            std::max(1, 2048 / C));
  TORCH_CHECK(!relu || y.has_value(), "batchnorm_bwd: relu mask needs y");
  DLA_DISPATCH_FLOAT_TYPES(x.scalar_type(), "batchnorm_bwd", [&] {
    const dev_t* yp = y.has_value() ? (const dev_t*)y->data_ptr() : nullptr;
    auto launch_stats = [&](auto rtag) {
      constexpr bool R = decltype(rtag)::value;
      if (nhwc) {
        constexpr int VM = 16 / (int)sizeof(dev_t);
        auto stats = [&](auto vtag) {
          constexpr int V = decltype(vtag)::value;
          const int lanes = C / V;
          const int64_t used = dla::nhwc_used_threads(lanes, rows, 262144);
          const int lds = 2 * C * sizeof(float);
          hipLaunchKernelGGL((dla::bn_bwd_stats_nhwc_kernel<dev_t, V, R>),
                             dim3((int)((used + 255) / 256)), dim3(256), lds,
                             dla::stream(), (const dev_t*)dy.data_ptr(),
                             (const dev_t*)x.data_ptr(), yp, mean.data_ptr<float>(),
                             rstd.data_ptr<float>(), sums.data_ptr<float>(), C,
                             rows, used);
        };
        if (C % VM == 0) stats(std::integral_constant<int, VM>{});
        else if (C % 2 == 0) stats(std::integral_constant<int, 2>{});
        else stats(std::integral_constant<int, 1>{});
      } else {
        hipLaunchKernelGGL((dla::bn_bwd_stats_kernel<dev_t, R>), dim3(C, ysplit),
                           dim3(256), 0, dla::stream(), (const dev_t*)dy.data_ptr(),
                           (const dev_t*)x.data_ptr(), yp, mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), sums.data_ptr<float>(), N, C, HW);
      }
    };
    if (relu) launch_stats(std::true_type{});
    else launch_stats(std::false_type{});

    const int64_t n_total = x.numel();
    const float inv_count = 1.f / (float)rows;
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch_dx = [&](auto vtag, auto rtag) {
      constexpr int V = decltype(vtag)::value;
      constexpr bool R = decltype(rtag)::value;
      if (nhwc) {
        const int lanes = C / V;
        const int64_t used = dla::nhwc_used_threads(lanes, rows, 524288);
        hipLaunchKernelGGL((dla::bn_bwd_dx_nhwc_kernel<dev_t, V, R>),
                           dim3((int)((used + 255) / 256)), dim3(256), 0,
                           dla::stream(), (const dev_t*)dy.data_ptr(),
                           (const dev_t*)x.data_ptr(), yp, mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), w32.data_ptr<float>(),
                           sums.data_ptr<float>(), (dev_t*)dx.data_ptr(), C,
                           rows, used, inv_count);
      } else {
        hipLaunchKernelGGL((dla::bn_bwd_dx_kernel<dev_t, R>), dim3(dla::grid_1d(n_total, 256)),
                           dim3(256), 0, dla::stream(), (const dev_t*)dy.data_ptr(),
                           (const dev_t*)x.data_ptr(), yp, mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), w32.data_ptr<float>(),
                           sums.data_ptr<float>(), (dev_t*)dx.data_ptr(), C, HW,
                           n_total, inv_count);
      }
    };
    const bool vec_ok = nhwc && (C % VMAX == 0);
    if (vec_ok) {
      if (relu) launch_dx(std::integral_constant<int, VMAX>{}, std::true_type{});
      else launch_dx(std::integral_constant<int, VMAX>{}, std::false_type{});
    } else {
      if (relu) launch_dx(std::integral_constant<int, 1>{}, std::true_type{});
      else launch_dx(std::integral_constant<int, 1>{}, std::false_type{});
    }
  });
  HIP_CHECK_ERR();
  auto dweight = sums.narrow(0, C, C).clone().to(weight.scalar_type());
  auto dbias = sums.narrow(0, 0, C).clone().to(weight.scalar_type());
  return {dx, dweight, dbias};
}
