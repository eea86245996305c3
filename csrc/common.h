// Common helpers for deeplearning_amd HIP kernels (gfx950 / CDNA4 only).
//
// Design rules (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; block sizes are multiples of 64 (256 default)
//  - bf16 loads vectorized (>=8B per lane) wherever layout permits (G13)
//  - memory-bound grids capped at ~2048 blocks with grid-stride loops (G11)
//  - fp32 accumulation for all reductions
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define DLA_WAVE 64
#define DLA_CHECK_CUDA(x) TORCH_CHECK((x).is_cuda(), #x " must be a GPU tensor")
#define DLA_CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define DLA_CHECK_INPUT(x) DLA_CHECK_CUDA(x); DLA_CHECK_CONTIG(x)
// dense = standard-contiguous OR channels_last-contiguous (flat elementwise ok)
#define DLA_CHECK_DENSE(x)                                                     \
  DLA_CHECK_CUDA(x);                                                           \
  TORCH_CHECK((x).is_contiguous() ||                                           \
                  ((x).dim() == 4 &&                                           \
                   (x).is_contiguous(at::MemoryFormat::ChannelsLast)),         \
              #x " must be dense (contiguous or channels_last)")

#define HIP_CHECK_ERR()                                                        \
  do {                                                                         \
    hipError_t e = hipGetLastError();                                          \
    TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ",                 \
                hipGetErrorString(e));                                         \
  } while (0)

namespace dla {

constexpr int kMaxGrid = 2048;  // ~8 blocks per CU on 256 CUs

inline int grid_1d(int64_t n, int block) {
  int64_t g = (n + block - 1) / block;
  return (int)std::min<int64_t>(g, kMaxGrid);
}

inline hipStream_t stream() { return at::hip::getCurrentHIPStream().stream(); }

// ---- dtype conversion helpers ------------------------------------------------
template <typename T> struct AccT { using type = float; };

__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }
__device__ __forceinline__ float to_f32(__half x) { return __half2float(x); }

template <typename T> __device__ __forceinline__ T from_f32(float x);
template <> __device__ __forceinline__ float from_f32<float>(float x) { return x; }
template <> __device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}
template <> __device__ __forceinline__ __half from_f32<__half>(float x) {
  return __float2half(x);
}

// map at::ScalarType -> device type
template <typename scalar_t> struct DevT { using type = scalar_t; };
template <> struct DevT<at::BFloat16> { using type = __hip_bfloat16; };
template <> struct DevT<at::Half> { using type = __half; };

// ---- wave & block reductions -------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block reduce using LDS; valid for blockDim.x <= 1024, one value per thread.
// smem must have >= blockDim.x/64 floats. Result valid on every thread.
__device__ __forceinline__ float block_reduce_sum(float v, float* smem) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + 63) >> 6;
  float r = (threadIdx.x < nw) ? smem[threadIdx.x] : 0.0f;
  r = wave_reduce_sum(r);  // nw <= 16, fits one wave; all lanes get it
  if (threadIdx.x == 0) smem[0] = r;
  __syncthreads();
  r = smem[0];
  __syncthreads();
  return r;
}

__device__ __forceinline__ float block_reduce_max(float v, float* smem) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  v = wave_reduce_max(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + 63) >> 6;
  float r = (threadIdx.x < nw) ? smem[threadIdx.x] : -INFINITY;
  r = wave_reduce_max(r);
  if (threadIdx.x == 0) smem[0] = r;
  __syncthreads();
  r = smem[0];
  __syncthreads();
  return r;
}

}  // namespace dla

// dispatch fp32/bf16/fp16 with device-native types
#define DLA_DISPATCH_FLOAT_TYPES(TYPE, NAME, ...)                              \
  [&] {                                                                        \
    switch (TYPE) {                                                            \
      case at::ScalarType::Float: {                                            \
        using scalar_t = float; using dev_t = float;                           \
        return __VA_ARGS__();                                                  \
      }                                                                        \
      case at::ScalarType::BFloat16: {                                         \
        using scalar_t = at::BFloat16; using dev_t = __hip_bfloat16;           \
        return __VA_ARGS__();                                                  \
      }                                                                        \
      case at::ScalarType::Half: {                                             \
        using scalar_t = at::Half; using dev_t = __half;                       \
        return __VA_ARGS__();                                                  \
      }                                                                        \
      default:                                                                 \
        TORCH_CHECK(false, #NAME ": unsupported dtype ", TYPE);                \
    }                                                                          \
  }()
