#include "hip/hip_runtime.h"
// Elementwise activation kernels: GELU (erf, as nn.GELU default), SiLU,
// fused add+ReLU (residual join), hardswish. fp32/bf16/fp16, vectorized.
//
// Reference call sites: GELU in ViT/Swin/ConvNeXt MLPs (vit_model.py:136+),
// SiLU in yolov5 Conv blocks (models/common.py:36-44), residual-add+ReLU in
// every ResNet BasicBlock/Bottleneck (classification/resnet/models/networks.py).
#include "common.h"
#include "vec.h"

namespace dla {

__device__ __forceinline__ float gelu_f(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f));
}
__device__ __forceinline__ float gelu_grad_f(float x) {
  const float cdf = 0.5f * (1.0f + erff(x * 0.70710678118654752440f));
  const float pdf = 0.3989422804014327f * expf(-0.5f * x * x);
  return cdf + x * pdf;
}
__device__ __forceinline__ float silu_f(float x) {
  return x / (1.0f + expf(-x));
}
__device__ __forceinline__ float silu_grad_f(float x) {
  const float s = 1.0f / (1.0f + expf(-x));
  return s * (1.0f + x * (1.0f - s));
}

enum class EwOp { kGelu, kGeluGrad, kSilu, kSiluGrad };

template <typename dev_t, int V, EwOp OP>
__global__ void ew_unary_kernel(const dev_t* __restrict__ x,
                                const dev_t* __restrict__ g,  // grad_out (grad ops)
                                dev_t* __restrict__ y, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * V;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * V; i < n;
       i += stride) {
    Vec<dev_t, V> xv = vload<dev_t, V>(x + i);
    Vec<dev_t, V> yv;
    if constexpr (OP == EwOp::kGeluGrad || OP == EwOp::kSiluGrad) {
      Vec<dev_t, V> gv = vload<dev_t, V>(g + i);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float f = to_f32(xv.v[j]);
        float d = (OP == EwOp::kGeluGrad) ? gelu_grad_f(f) : silu_grad_f(f);
        yv.v[j] = from_f32<dev_t>(to_f32(gv.v[j]) * d);
      }
    } else {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float f = to_f32(xv.v[j]);
        yv.v[j] = from_f32<dev_t>(OP == EwOp::kGelu ? gelu_f(f) : silu_f(f));
      }
    }
    vstore<dev_t, V>(y + i, yv);
  }
}

// y = relu(a + b); mask for backward comes from y > 0
template <typename dev_t, int V>
__global__ void add_relu_kernel(const dev_t* __restrict__ a,
                                const dev_t* __restrict__ b,
                                dev_t* __restrict__ y, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * V;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * V; i < n;
       i += stride) {
    Vec<dev_t, V> av = vload<dev_t, V>(a + i);
    Vec<dev_t, V> bv = vload<dev_t, V>(b + i);
    Vec<dev_t, V> yv;
#pragma unroll
    for (int j = 0; j < V; ++j)
      yv.v[j] = from_f32<dev_t>(fmaxf(to_f32(av.v[j]) + to_f32(bv.v[j]), 0.f));
    vstore<dev_t, V>(y + i, yv);
  }
}

template <typename dev_t, int V>
__global__ void relu_mask_grad_kernel(const dev_t* __restrict__ dy,
                                      const dev_t* __restrict__ y,
                                      dev_t* __restrict__ dx, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * V;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * V; i < n;
       i += stride) {
    Vec<dev_t, V> dyv = vload<dev_t, V>(dy + i);
    Vec<dev_t, V> yv = vload<dev_t, V>(y + i);
    Vec<dev_t, V> dxv;
#pragma unroll
    for (int j = 0; j < V; ++j)
      dxv.v[j] = from_f32<dev_t>(to_f32(yv.v[j]) > 0.f ? to_f32(dyv.v[j]) : 0.f);
    vstore<dev_t, V>(dx + i, dxv);
  }
}

template <EwOp OP>
torch::Tensor ew_unary(torch::Tensor x, c10::optional<torch::Tensor> g) {
  DLA_CHECK_DENSE(x);
  auto y = torch::empty_like(x);
  const int64_t n = x.numel();
  DLA_DISPATCH_FLOAT_TYPES(x.scalar_type(), "ew_unary", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    const dev_t* gp = g.has_value() ? (const dev_t*)g->data_ptr() : nullptr;
    auto launch = [&](auto vtag) {
      constexpr int V = decltype(vtag)::value;
      const int grid = dla::grid_1d((n + V - 1) / V, 256);
      hipLaunchKernelGGL((dla::ew_unary_kernel<dev_t, V, OP>), dim3(grid),
                         dim3(256), 0, dla::stream(), (const dev_t*)x.data_ptr(),
                         gp, (dev_t*)y.data_ptr(), n);
    };
    if (n % VMAX == 0) launch(std::integral_constant<int, VMAX>{});
    else launch(std::integral_constant<int, 1>{});
  });
  HIP_CHECK_ERR();
  return y;
}

}  // namespace dla

torch::Tensor gelu_fwd(torch::Tensor x) {
  return dla::ew_unary<dla::EwOp::kGelu>(x, c10::nullopt);
}
torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x) {
  return dla::ew_unary<dla::EwOp::kGeluGrad>(x, dy);
}
torch::Tensor silu_fwd(torch::Tensor x) {
  return dla::ew_unary<dla::EwOp::kSilu>(x, c10::nullopt);
}
torch::Tensor silu_bwd(torch::Tensor dy, torch::Tensor x) {
  return dla::ew_unary<dla::EwOp::kSiluGrad>(x, dy);
}

torch::Tensor add_relu_fwd(torch::Tensor a, torch::Tensor b) {
  DLA_CHECK_DENSE(a); DLA_CHECK_DENSE(b);
  TORCH_CHECK(a.sizes() == b.sizes(), "add_relu: shape mismatch");
  auto y = torch::empty_like(a);
  const int64_t n = a.numel();
  DLA_DISPATCH_FLOAT_TYPES(a.scalar_type(), "add_relu", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag) {
      constexpr int V = decltype(vtag)::value;
      const int grid = dla::grid_1d((n + V - 1) / V, 256);
      hipLaunchKernelGGL((dla::add_relu_kernel<dev_t, V>), dim3(grid), dim3(256),
                         0, dla::stream(), (const dev_t*)a.data_ptr(),
                         (const dev_t*)b.data_ptr(), (dev_t*)y.data_ptr(), n);
    };
    if (n % VMAX == 0) launch(std::integral_constant<int, VMAX>{});
    else launch(std::integral_constant<int, 1>{});
  });
  HIP_CHECK_ERR();
  return y;
}

torch::Tensor relu_mask_bwd(torch::Tensor dy, torch::Tensor y) {
  DLA_CHECK_DENSE(dy); DLA_CHECK_DENSE(y);
  auto dx = torch::empty_like(dy);
  const int64_t n = dy.numel();
  DLA_DISPATCH_FLOAT_TYPES(dy.scalar_type(), "relu_mask_bwd", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag) {
      constexpr int V = decltype(vtag)::value;
      const int grid = dla::grid_1d((n + V - 1) / V, 256);
      hipLaunchKernelGGL((dla::relu_mask_grad_kernel<dev_t, V>), dim3(grid),
                         dim3(256), 0, dla::stream(), (const dev_t*)dy.data_ptr(),
                         (const dev_t*)y.data_ptr(), (dev_t*)dx.data_ptr(), n);
    };
    if (n % VMAX == 0) launch(std::integral_constant<int, VMAX>{});
    else launch(std::integral_constant<int, 1>{});
  });
  HIP_CHECK_ERR();
  return dx;
}
