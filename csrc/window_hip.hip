#include "hip/hip_runtime.h"
// Swin fused (roll +) window partition / merge kernels.
//
// Same semantics as the reference CUDA kernels
// (classification/swin_transformer/kernels/window_process/swin_window_process_kernel.cu:42-323):
//   partition fwd : [B,H,W,C] -> [B*nH*nW, ws, ws, C] with roll(-s) folded in
//   merge fwd     : inverse, with roll(+s) folded in
// but built fresh for CDNA4: both directions are pure gathers (bijective index
// map), vectorized 16B over C, 256-thread wave64 blocks, grid-stride.
// The same two gathers serve as each other's backward with the shift negated:
//   partition_bwd == merge_fwd gather, merge_bwd == partition_fwd gather.
#include "common.h"
#include "vec.h"

namespace dla {

__device__ __forceinline__ int wrap(int v, int m) {
  v %= m;
  return v < 0 ? v + m : v;
}

// out[bw, i, j, :] = in[b, (wi*ws+i+shift) mod H, (wj*ws+j+shift) mod W, :]
template <typename dev_t, int V>
__global__ void to_windows_kernel(const dev_t* __restrict__ in,
                                  dev_t* __restrict__ out, int B, int H, int W,
                                  int C, int ws, int shift) {
  const int nW = W / ws;
  const int nH = H / ws;
  const int Cv = C / V;
  const int64_t total = (int64_t)B * H * W * Cv;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const int cv = (int)(t % Cv);
    const int j = (int)((t / Cv) % ws);
    const int i = (int)((t / Cv / ws) % ws);
    const int64_t bw = t / Cv / ws / ws;  // window index in [0, B*nH*nW)
    const int b = (int)(bw / (nH * nW));
    const int wi = (int)((bw / nW) % nH);
    const int wj = (int)(bw % nW);
    const int h = wrap(wi * ws + i + shift, H);
    const int w = wrap(wj * ws + j + shift, W);
    const int64_t src = (((int64_t)b * H + h) * W + w) * Cv + cv;
    vstore<dev_t, V>(out + t * V, vload<dev_t, V>(in + src * V));
  }
}

// out[b, h, w, :] = in[window index of ((h+shift) mod H, (w+shift) mod W), :]
template <typename dev_t, int V>
__global__ void from_windows_kernel(const dev_t* __restrict__ in,
                                    dev_t* __restrict__ out, int B, int H, int W,
                                    int C, int ws, int shift) {
  const int nW = W / ws;
  const int nH = H / ws;
  const int Cv = C / V;
  const int64_t total = (int64_t)B * H * W * Cv;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += stride) {
    const int cv = (int)(t % Cv);
    const int w = (int)((t / Cv) % W);
    const int h = (int)((t / Cv / W) % H);
    const int b = (int)(t / Cv / W / H);
    const int h2 = wrap(h + shift, H);
    const int w2 = wrap(w + shift, W);
    const int wi = h2 / ws, i = h2 % ws;
    const int wj = w2 / ws, j = w2 % ws;
    const int64_t bw = ((int64_t)b * nH + wi) * nW + wj;
    const int64_t src = (((int64_t)bw * ws + i) * ws + j) * Cv + cv;
    vstore<dev_t, V>(out + t * V, vload<dev_t, V>(in + src * V));
  }
}

template <bool TO_WINDOWS>
torch::Tensor window_gather(torch::Tensor x, int B, int H, int W, int C, int ws,
                            int shift) {
  auto xc = x.contiguous();
  const int nW = W / ws, nH = H / ws;
  torch::Tensor out;
  if (TO_WINDOWS)
    out = torch::empty({(int64_t)B * nH * nW, ws, ws, C}, xc.options());
  else
    out = torch::empty({B, H, W, C}, xc.options());
  const int64_t total = (int64_t)B * H * W * C;
  DLA_DISPATCH_FLOAT_TYPES(xc.scalar_type(), "window_gather", [&] {
    constexpr int VMAX = 16 / (int)sizeof(dev_t);
    auto launch = [&](auto vtag) {
      constexpr int V = decltype(vtag)::value;
      const int grid = dla::grid_1d(total / V, 256);
      if (TO_WINDOWS)
        hipLaunchKernelGGL((dla::to_windows_kernel<dev_t, V>), dim3(grid),
                           dim3(256), 0, dla::stream(),
                           (const dev_t*)xc.data_ptr(), (dev_t*)out.data_ptr(), B,
                           H, W, C, ws, shift);
      else
        hipLaunchKernelGGL((dla::from_windows_kernel<dev_t, V>), dim3(grid),
                           dim3(256), 0, dla::stream(),
                           (const dev_t*)xc.data_ptr(), (dev_t*)out.data_ptr(), B,
                           H, W, C, ws, shift);
    };
    if (C % VMAX == 0) launch(std::integral_constant<int, VMAX>{});
    else if (C % 2 == 0) launch(std::integral_constant<int, 2>{});
    else launch(std::integral_constant<int, 1>{});
  });
  HIP_CHECK_ERR();
  return out;
}

}  // namespace dla

// roll(-shift) + partition: [B,H,W,C] -> [B*nH*nW,ws,ws,C]
torch::Tensor window_partition_fwd(torch::Tensor x, int64_t ws, int64_t shift) {
  DLA_CHECK_CUDA(x);
  TORCH_CHECK(x.dim() == 4, "expect [B,H,W,C]");
  const int B = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
            C = (int)x.size(3);
  return dla::window_gather<true>(x, B, H, W, C, (int)ws, (int)shift);
}

// backward of partition == merge-style gather with shift negated
torch::Tensor window_partition_bwd(torch::Tensor grad, int64_t B, int64_t H,
                                   int64_t W, int64_t ws, int64_t shift) {
  DLA_CHECK_CUDA(grad);
  const int C = (int)grad.size(-1);
  return dla::window_gather<false>(grad, (int)B, (int)H, (int)W, C, (int)ws,
                                   -(int)shift);
}

// merge + roll(+shift): [B*nH*nW,ws,ws,C] -> [B,H,W,C]
torch::Tensor window_merge_fwd(torch::Tensor windows, int64_t B, int64_t H,
                               int64_t W, int64_t ws, int64_t shift) {
  DLA_CHECK_CUDA(windows);
  const int C = (int)windows.size(-1);
  return dla::window_gather<false>(windows, (int)B, (int)H, (int)W, C, (int)ws,
                                   -(int)shift);
}

torch::Tensor window_merge_bwd(torch::Tensor grad, int64_t ws, int64_t shift) {
  DLA_CHECK_CUDA(grad);
  TORCH_CHECK(grad.dim() == 4, "expect [B,H,W,C]");
  const int B = (int)grad.size(0), H = (int)grad.size(1), W = (int)grad.size(2),
            C = (int)grad.size(3);
  return dla::window_gather<true>(grad, B, H, W, C, (int)ws, (int)shift);
}
