// Fused LayerNorm forward/backward for (M, C) rows, fp32/bf16/fp16 in, fp32 stats.
//
// Replaces the reference's nn.LayerNorm call sites (ViT vit_model.py:114-135,
// Swin models/swin_transformer.py:19-36 + --fused_layernorm flag main.py:72,
// ConvNeXt channels-first LN) with one hand-written CDNA4 kernel.
// Design: one 256-thread block per row (grid-stride over rows), wave64
// shuffle reductions, 16B vectorized loads when C divides the vector width.
#include "common.h"
#include "vec.h"

namespace dla {

template <typename dev_t, int V>
__global__ void ln_fwd_kernel(const dev_t* __restrict__ x,
                              const dev_t* __restrict__ w,
                              const dev_t* __restrict__ b,
                              dev_t* __restrict__ y,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out,
                              int M, int C, float eps) {
  __shared__ float smem[16];
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const dev_t* xr = x + (int64_t)row * C;
    dev_t* yr = y + (int64_t)row * C;
    float sum = 0.f, sumsq = 0.f;
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> xv = vload<dev_t, V>(xr + i);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float f = to_f32(xv.v[j]);
        sum += f;
        sumsq += f * f;
      }
    }
    sum = block_reduce_sum(sum, smem);
    sumsq = block_reduce_sum(sumsq, smem);
    const float mu = sum / C;
    const float var = fmaxf(sumsq / C - mu * mu, 0.f);
    const float rs = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mu;
      rstd_out[row] = rs;
    }
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> xv = vload<dev_t, V>(xr + i);
      Vec<dev_t, V> wv = vload<dev_t, V>(w + i);
      Vec<dev_t, V> bv = vload<dev_t, V>(b + i);
      Vec<dev_t, V> yv;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float f = (to_f32(xv.v[j]) - mu) * rs;
        yv.v[j] = from_f32<dev_t>(f * to_f32(wv.v[j]) + to_f32(bv.v[j]));
      }
      vstore<dev_t, V>(yr + i, yv);
    }
  }
}

// dx = rs * (dyw - mean(dyw) - xhat * mean(dyw * xhat)), dyw = dy * w
template <typename dev_t, int V>
__global__ void ln_bwd_dx_kernel(const dev_t* __restrict__ dy,
                                 const dev_t* __restrict__ x,
                                 const dev_t* __restrict__ w,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 dev_t* __restrict__ dx,
                                 int M, int C) {
  __shared__ float smem[16];
  for (int row = blockIdx.x; row < M; row += gridDim.x) {
    const dev_t* dyr = dy + (int64_t)row * C;
    const dev_t* xr = x + (int64_t)row * C;
    dev_t* dxr = dx + (int64_t)row * C;
    const float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> dyv = vload<dev_t, V>(dyr + i);
      Vec<dev_t, V> xv = vload<dev_t, V>(xr + i);
      Vec<dev_t, V> wv = vload<dev_t, V>(w + i);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float g = to_f32(dyv.v[j]) * to_f32(wv.v[j]);
        float xh = (to_f32(xv.v[j]) - mu) * rs;
        s1 += g;
        s2 += g * xh;
      }
    }
    s1 = block_reduce_sum(s1, smem) / C;
    s2 = block_reduce_sum(s2, smem) / C;
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> dyv = vload<dev_t, V>(dyr + i);
      Vec<dev_t, V> xv = vload<dev_t, V>(xr + i);
      Vec<dev_t, V> wv = vload<dev_t, V>(w + i);
      Vec<dev_t, V> dxv;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float g = to_f32(dyv.v[j]) * to_f32(wv.v[j]);
        float xh = (to_f32(xv.v[j]) - mu) * rs;
        dxv.v[j] = from_f32<dev_t>(rs * (g - s1 - xh * s2));
      }
      vstore<dev_t, V>(dxr + i, dxv);
    }
  }
}

// dgamma[c] = sum_rows dy*xhat ; dbeta[c] = sum_rows dy.
// 2D grid: x walks C in 256-chunks (coalesced), y splits rows; fp32 atomics.
template <typename dev_t>
__global__ void ln_bwd_dwdb_kernel(const dev_t* __restrict__ dy,
                                   const dev_t* __restrict__ x,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   float* __restrict__ dw,
                                   float* __restrict__ db,
                                   int M, int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float sw = 0.f, sb = 0.f;
  for (int row = blockIdx.y; row < M; row += gridDim.y) {
    const float g = to_f32(dy[(int64_t)row * C + c]);
    const float xh = (to_f32(x[(int64_t)row * C + c]) - mean[row]) * rstd[row];
    sw += g * xh;
    sb += g;
  }
  if (gridDim.y == 1) {
    dw[c] = sw;
    db[c] = sb;
  } else {
    atomicAdd(&dw[c], sw);
    atomicAdd(&db[c], sb);
  }
}

// ---- LDS-staged single-pass variants (C <= 4096) ---------------------------
// fwd: read x once (staged in LDS as f32), one write. bwd: read dy,x once,
// write dx, and accumulate dgamma/dbeta per block in LDS -> one atomicAdd per
// channel per block (removes the separate dw/db tensor passes entirely).

template <typename dev_t, int V>
__global__ void ln_fwd_smem_kernel(const dev_t* __restrict__ x,
                                   const dev_t* __restrict__ w,
                                   const dev_t* __restrict__ b,
                                   dev_t* __restrict__ y,
                                   float* __restrict__ mean_out,
                                   float* __restrict__ rstd_out,
                                   int M, int C, float eps) {
  extern __shared__ __attribute__((aligned(16))) float lds[];
  float* wsh = lds;            // C
  float* bsh = lds + C;        // C
  float* row = lds + 2 * C;    // C
  float* red = lds + 3 * C;    // 16
  for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
    Vec<dev_t, V> wv = vload<dev_t, V>(w + i);
    Vec<dev_t, V> bv = vload<dev_t, V>(b + i);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      wsh[i + j] = to_f32(wv.v[j]);
      bsh[i + j] = to_f32(bv.v[j]);
    }
  }
  __syncthreads();
  for (int r = blockIdx.x; r < M; r += gridDim.x) {
    const dev_t* xr = x + (int64_t)r * C;
    dev_t* yr = y + (int64_t)r * C;
    float sum = 0.f, sumsq = 0.f;
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> xv = vload<dev_t, V>(xr + i);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float f = to_f32(xv.v[j]);
        row[i + j] = f;
        sum += f;
        sumsq += f * f;
      }
    }
    sum = block_reduce_sum(sum, red);
    sumsq = block_reduce_sum(sumsq, red);
    const float mu = sum / C;
    const float rs = rsqrtf(fmaxf(sumsq / C - mu * mu, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[r] = mu;
      rstd_out[r] = rs;
    }
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> yv;
#pragma unroll
      for (int j = 0; j < V; ++j)
        yv.v[j] = from_f32<dev_t>((row[i + j] - mu) * rs * wsh[i + j] + bsh[i + j]);
      vstore<dev_t, V>(yr + i, yv);
    }
    __syncthreads();  // row[] reused next iteration
  }
}

template <typename dev_t, int V>
__global__ void ln_bwd_smem_kernel(const dev_t* __restrict__ dy,
                                   const dev_t* __restrict__ x,
                                   const dev_t* __restrict__ w,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   dev_t* __restrict__ dx,
                                   float* __restrict__ dw,
                                   float* __restrict__ db, int M, int C) {
  extern __shared__ __attribute__((aligned(16))) float lds[];
  float* wsh = lds;            // C
  float* dyrow = lds + C;      // C
  float* xhrow = lds + 2 * C;  // C
  float* dgs = lds + 3 * C;    // C (block dgamma)
  float* dbs = lds + 4 * C;    // C (block dbeta)
  float* red = lds + 5 * C;    // 16
  for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
    Vec<dev_t, V> wv = vload<dev_t, V>(w + i);
#pragma unroll
    for (int j = 0; j < V; ++j) {
      wsh[i + j] = to_f32(wv.v[j]);
      dgs[i + j] = 0.f;
      dbs[i + j] = 0.f;
    }
  }
  __syncthreads();
  for (int r = blockIdx.x; r < M; r += gridDim.x) {
    const dev_t* dyr = dy + (int64_t)r * C;
    const dev_t* xr = x + (int64_t)r * C;
    dev_t* dxr = dx + (int64_t)r * C;
    const float mu = mean[r], rs = rstd[r];
    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> dyv = vload<dev_t, V>(dyr + i);
      Vec<dev_t, V> xv = vload<dev_t, V>(xr + i);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float g = to_f32(dyv.v[j]);
        const float xh = (to_f32(xv.v[j]) - mu) * rs;
        dyrow[i + j] = g;
        xhrow[i + j] = xh;
        const float gw = g * wsh[i + j];
        s1 += gw;
        s2 += gw * xh;
      }
    }
    s1 = block_reduce_sum(s1, red) / C;
    s2 = block_reduce_sum(s2, red) / C;
    for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
      Vec<dev_t, V> dxv;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float g = dyrow[i + j];
        const float xh = xhrow[i + j];
        dxv.v[j] = from_f32<dev_t>(rs * (g * wsh[i + j] - s1 - xh * s2));
        dgs[i + j] += g * xh;  // thread-exclusive slot: i+j depends on tid only
        dbs[i + j] += g;
      }
      vstore<dev_t, V>(dxr + i, dxv);
    }
    __syncthreads();
  }
  for (int i = threadIdx.x * V; i < C; i += blockDim.x * V) {
#pragma unroll
    for (int j = 0; j < V; ++j) {
      atomicAdd(&dw[i + j], dgs[i + j]);
      atomicAdd(&db[i + j], dbs[i + j]);
    }
  }
}

// ---- wave-per-row variants (C <= 64*V*8) -----------------------------------
// One 64-lane wavefront owns one row: shuffle-only stats (no __syncthreads in
// the row loop), full-lane vector loads, 4 rows in flight per 256-thread block.
// bwd accumulates dgamma/dbeta into per-block LDS (atomicAdd to LDS across the
// 4 waves) and flushes once per block. Profiling motivation: the smem variants
// idle 60% of lanes at C=768 and barrier twice per row (rocprof: ln_bwd 295us
// vs ~30us of traffic at ViT-B shapes).

template <typename dev_t, int V>
__global__ __launch_bounds__(256) void ln_fwd_wave_kernel(const dev_t* __restrict__ x,
                                   const dev_t* __restrict__ w,
                                   const dev_t* __restrict__ b,
                                   dev_t* __restrict__ y,
                                   float* __restrict__ mean_out,
                                   float* __restrict__ rstd_out,
                                   int M, int C, float eps) {
  constexpr int MAX_PL = 8;  // max vectors per lane (compile-time bound:
                             // runtime-indexed arrays would spill to scratch)
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  const int rows_per_iter = gridDim.x * nwaves;
  Vec<dev_t, V> xv[MAX_PL];
  for (int r = blockIdx.x * nwaves + wave; r < M; r += rows_per_iter) {
    const dev_t* xr = x + (int64_t)r * C;
    dev_t* yr = y + (int64_t)r * C;
    float sum = 0.f, sumsq = 0.f;
#pragma unroll
    for (int p = 0; p < MAX_PL; ++p) {
      const int i = lane * V + p * 64 * V;
      if (i < C) {
        xv[p] = vload<dev_t, V>(xr + i);
  #pragma unroll
        for (int j = 0; j < V; ++j) {
          const float f = to_f32(xv[p].v[j]);
          sum += f;
          sumsq += f * f;
        }
      }
    }
    sum = wave_reduce_sum(sum);
    sumsq = wave_reduce_sum(sumsq);
    const float mu = sum / C;
    const float rs = rsqrtf(fmaxf(sumsq / C - mu * mu, 0.f) + eps);
    if (lane == 0) {
      mean_out[r] = mu;
      rstd_out[r] = rs;
    }
#pragma unroll
    for (int p = 0; p < MAX_PL; ++p) {
      const int i = lane * V + p * 64 * V;
      if (i < C) {
        Vec<dev_t, V> wv = vload<dev_t, V>(w + i);
        Vec<dev_t, V> bv = vload<dev_t, V>(b + i);
        Vec<dev_t, V> yv;
  #pragma unroll
        for (int j = 0; j < V; ++j)
          yv.v[j] = from_f32<dev_t>((to_f32(xv[p].v[j]) - mu) * rs *
                                        to_f32(wv.v[j]) +
                                    to_f32(bv.v[j]));
        vstore<dev_t, V>(yr + i, yv);
      }
    }
  }
}

template <typename dev_t, int V>
__global__ __launch_bounds__(256) void ln_bwd_wave_kernel(const dev_t* __restrict__ dy,
                                   const dev_t* __restrict__ x,
                                   const dev_t* __restrict__ w,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   dev_t* __restrict__ dx,
                                   float* __restrict__ dw,
                                   float* __restrict__ db, int M, int C) {
  // dgamma/dbeta accumulate in per-lane REGISTERS (each lane owns fixed
  // channels); one partial row per (block, wave) is written at the end and
  // summed by the host (no LDS or global atomics anywhere).
  constexpr int MAX_PL = 4;  // caps C at 64*V*4 (bf16: 2048); keeps the
                             // 2*MAX_PL*V accumulator regs affordable
  extern __shared__ __attribute__((aligned(16))) float lds[];
  float* wsh = lds;  // C (fp32 weight, shared by all waves)
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  for (int i = threadIdx.x; i < C; i += blockDim.x) wsh[i] = to_f32(w[i]);
  __syncthreads();
  Vec<dev_t, V> dyv[MAX_PL], xv[MAX_PL];
  float acc_dg[MAX_PL][V], acc_db[MAX_PL][V];
#pragma unroll
  for (int p = 0; p < MAX_PL; ++p)
#pragma unroll
    for (int j = 0; j < V; ++j) {
      acc_dg[p][j] = 0.f;
      acc_db[p][j] = 0.f;
    }
  for (int r = blockIdx.x * nwaves + wave; r < M;
       r += gridDim.x * nwaves) {
    const dev_t* dyr = dy + (int64_t)r * C;
    const dev_t* xr = x + (int64_t)r * C;
    dev_t* dxr = dx + (int64_t)r * C;
    const float mu = mean[r], rs = rstd[r];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int p = 0; p < MAX_PL; ++p) {
      const int i = lane * V + p * 64 * V;
      if (i < C) {
        dyv[p] = vload<dev_t, V>(dyr + i);
        xv[p] = vload<dev_t, V>(xr + i);
  #pragma unroll
        for (int j = 0; j < V; ++j) {
          const float gw = to_f32(dyv[p].v[j]) * wsh[i + j];
          const float xh = (to_f32(xv[p].v[j]) - mu) * rs;
          s1 += gw;
          s2 += gw * xh;
        }
      }
    }
    s1 = wave_reduce_sum(s1) / C;
    s2 = wave_reduce_sum(s2) / C;
#pragma unroll
    for (int p = 0; p < MAX_PL; ++p) {
      const int i = lane * V + p * 64 * V;
      if (i < C) {
        Vec<dev_t, V> dxv;
  #pragma unroll
        for (int j = 0; j < V; ++j) {
          const float g = to_f32(dyv[p].v[j]);
          const float xh = (to_f32(xv[p].v[j]) - mu) * rs;
          dxv.v[j] = from_f32<dev_t>(rs * (g * wsh[i + j] - s1 - xh * s2));
          acc_dg[p][j] += g * xh;
          acc_db[p][j] += g;
        }
        vstore<dev_t, V>(dxr + i, dxv);
      }
    }
  }
  // flush one partial row per (block, wave)
  const int64_t prow = ((int64_t)blockIdx.x * nwaves + wave) * C;
#pragma unroll
  for (int p = 0; p < MAX_PL; ++p) {
    const int i = lane * V + p * 64 * V;
    if (i < C) {
  #pragma unroll
      for (int j = 0; j < V; ++j) {
        dw[prow + i + j] = acc_dg[p][j];
        db[prow + i + j] = acc_db[p][j];
      }
    }
  }
}

// out[c] = sum_g part[g*C + c], for two buffers at once.
__global__ void col_sum_2_kernel(const float* __restrict__ pa,
                                 const float* __restrict__ pb,
                                 float* __restrict__ oa,
                                 float* __restrict__ ob, int G, int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float sa = 0.f, sb = 0.f;
  for (int g = 0; g < G; ++g) {
    sa += pa[(int64_t)g * C + c];
    sb += pb[(int64_t)g * C + c];
  }
  oa[c] = sa;
  ob[c] = sb;
}

}  // namespace dla

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  DLA_CHECK_INPUT(x); DLA_CHECK_INPUT(w); DLA_CHECK_INPUT(b);
  const int C = (int)x.size(-1);
  const int64_t M = x.numel() / C;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({M}, x.options().dtype(torch::kFloat));
  auto rstd = torch::empty({M}, x.options().dtype(torch::kFloat));
  const int block = 256;
  const int grid = (int)std::min<int64_t>(M, dla::kMaxGrid);
  DLA_DISPATCH_FLOAT_TYPES(x.scalar_type(), "layernorm_fwd", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag) {
      constexpr int V = decltype(vtag)::value;
      if (C <= 64 * V * 8) {  // wave-per-row: 4 rows in flight per block
        const int nwaves = 4;
        const int g = (int)std::min<int64_t>((M + nwaves - 1) / nwaves,
                                             dla::kMaxGrid);
        hipLaunchKernelGGL((dla::ln_fwd_wave_kernel<dev_t, V>), dim3(g),
                           dim3(nwaves * 64), 0, dla::stream(),
                           (const dev_t*)x.data_ptr(), (const dev_t*)w.data_ptr(),
                           (const dev_t*)b.data_ptr(), (dev_t*)y.data_ptr(),
                           mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)M,
                           C, (float)eps);
      } else if (C <= 4096) {
        const int lds = (3 * C + 16) * sizeof(float);
        hipLaunchKernelGGL((dla::ln_fwd_smem_kernel<dev_t, V>), dim3(grid),
                           dim3(block), lds, dla::stream(),
                           (const dev_t*)x.data_ptr(), (const dev_t*)w.data_ptr(),
                           (const dev_t*)b.data_ptr(), (dev_t*)y.data_ptr(),
                           mean.data_ptr<float>(), rstd.data_ptr<float>(), (int)M,
                           C, (float)eps);
      } else {
        hipLaunchKernelGGL((dla::ln_fwd_kernel<dev_t, V>), dim3(grid), dim3(block), 0,
                           dla::stream(), (const dev_t*)x.data_ptr(),
                           (const dev_t*)w.data_ptr(), (const dev_t*)b.data_ptr(),
                           (dev_t*)y.data_ptr(), mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), (int)M, C, (float)eps);
      }
    };
    if (C % VMAX == 0) launch(std::integral_constant<int, VMAX>{});
    else if (C % 4 == 0) launch(std::integral_constant<int, 4>{});
    else launch(std::integral_constant<int, 1>{});
  });
  HIP_CHECK_ERR();
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  DLA_CHECK_INPUT(dy); DLA_CHECK_INPUT(x); DLA_CHECK_INPUT(w);
  const int C = (int)x.size(-1);
  const int64_t M = x.numel() / C;
  auto dx = torch::empty_like(x);
  const int ysplit = (int)std::min<int64_t>((M + 255) / 256 + 1, 64);
  auto dw = torch::zeros({C}, x.options().dtype(torch::kFloat));
  auto db = torch::zeros({C}, x.options().dtype(torch::kFloat));
  const int block = 256;
  const int grid = (int)std::min<int64_t>(M, dla::kMaxGrid);
  DLA_DISPATCH_FLOAT_TYPES(x.scalar_type(), "layernorm_bwd", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag) {
      constexpr int V = decltype(vtag)::value;
      if (C <= 64 * V * 4) {  // wave-per-row fused path
        const int nwaves = 4;
        const int g = (int)std::min<int64_t>((M + nwaves - 1) / nwaves,
                                             dla::kMaxGrid);
        const int lds = C * sizeof(float);
        auto part_dw = torch::empty({g * nwaves, C}, dw.options());
        auto part_db = torch::empty({g * nwaves, C}, db.options());
        hipLaunchKernelGGL((dla::ln_bwd_wave_kernel<dev_t, V>), dim3(g),
                           dim3(nwaves * 64), lds, dla::stream(),
                           (const dev_t*)dy.data_ptr(), (const dev_t*)x.data_ptr(),
                           (const dev_t*)w.data_ptr(), mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), (dev_t*)dx.data_ptr(),
                           part_dw.data_ptr<float>(), part_db.data_ptr<float>(),
                           (int)M, C);
        at::sum_out(dw, part_dw, {0});  // torch's tuned column reduce
        at::sum_out(db, part_db, {0});
        return;
      }
      if (C <= 3072) {  // fused dx + dgamma/dbeta path (5C+16 floats of LDS)
        const int lds = (5 * C + 16) * sizeof(float);
        hipLaunchKernelGGL((dla::ln_bwd_smem_kernel<dev_t, V>), dim3(grid),
                           dim3(block), lds, dla::stream(),
                           (const dev_t*)dy.data_ptr(), (const dev_t*)x.data_ptr(),
                           (const dev_t*)w.data_ptr(), mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), (dev_t*)dx.data_ptr(),
                           dw.data_ptr<float>(), db.data_ptr<float>(), (int)M, C);
        return;
      }
      hipLaunchKernelGGL((dla::ln_bwd_dx_kernel<dev_t, V>), dim3(grid), dim3(block), 0,
                         dla::stream(), (const dev_t*)dy.data_ptr(),
                         (const dev_t*)x.data_ptr(), (const dev_t*)w.data_ptr(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         (dev_t*)dx.data_ptr(), (int)M, C);
      dim3 g2((C + 255) / 256, ysplit);
      hipLaunchKernelGGL((dla::ln_bwd_dwdb_kernel<dev_t>), g2, dim3(256), 0,
                         dla::stream(), (const dev_t*)dy.data_ptr(),
                         (const dev_t*)x.data_ptr(), mean.data_ptr<float>(),
                         rstd.data_ptr<float>(), dw.data_ptr<float>(),
                         db.data_ptr<float>(), (int)M, C);
    };
    if (C % VMAX == 0) launch(std::integral_constant<int, VMAX>{});
    else if (C % 4 == 0) launch(std::integral_constant<int, 4>{});
    else launch(std::integral_constant<int, 1>{});
  });
  HIP_CHECK_ERR();
  return {dx, dw.to(x.scalar_type()), db.to(x.scalar_type())};
}
