#include "hip/hip_runtime.h"
// RoIAlign forward/backward, torchvision semantics (aligned flag, sampling
// ratio, spatial scale). Reference call sites: torchvision.ops.MultiScaleRoIAlign
// in fasterRcnn (models/faster_rcnn.py:8,305-309) — 512 proposals -> 7x7x256.
// One thread per output element; backward scatters with fp32 atomics.
#include "common.h"

namespace dla {

template <typename T>
__device__ __forceinline__ float bilinear_interpolate(const T* data, int H, int W,
                                                      float y, float x) {
  if (y < -1.f || y > H || x < -1.f || x > W) return 0.f;
  y = fmaxf(y, 0.f);
  x = fmaxf(x, 0.f);
  int y0 = (int)y, x0 = (int)x;
  int y1 = y0 + 1, x1 = x0 + 1;
  if (y0 >= H - 1) { y0 = y1 = H - 1; y = (float)y0; }
  if (x0 >= W - 1) { x0 = x1 = W - 1; x = (float)x0; }
  const float ly = y - y0, lx = x - x0;
  const float hy = 1.f - ly, hx = 1.f - lx;
  const float v00 = to_f32(data[y0 * W + x0]), v01 = to_f32(data[y0 * W + x1]);
  const float v10 = to_f32(data[y1 * W + x0]), v11 = to_f32(data[y1 * W + x1]);
  return hy * hx * v00 + hy * lx * v01 + ly * hx * v10 + ly * lx * v11;
}

template <typename dev_t>
__global__ void roialign_fwd_kernel(const dev_t* __restrict__ input,
                                    const float* __restrict__ rois,  // (R,5) b,x1,y1,x2,y2
                                    dev_t* __restrict__ output, int R, int C,
                                    int H, int W, int PH, int PW,
                                    float spatial_scale, int sampling_ratio,
                                    bool aligned) {
  const int64_t total = (int64_t)R * C * PH * PW;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int pw = (int)(i % PW);
    const int ph = (int)((i / PW) % PH);
    const int c = (int)((i / PW / PH) % C);
    const int r = (int)(i / PW / PH / C);
    const float* roi = rois + r * 5;
    const int b = (int)roi[0];
    const float off = aligned ? 0.5f : 0.f;
    const float x1 = roi[1] * spatial_scale - off;
    const float y1 = roi[2] * spatial_scale - off;
    const float x2 = roi[3] * spatial_scale - off;
    const float y2 = roi[4] * spatial_scale - off;
    float rw = x2 - x1, rh = y2 - y1;
    if (!aligned) { rw = fmaxf(rw, 1.f); rh = fmaxf(rh, 1.f); }
    const float bin_h = rh / PH, bin_w = rw / PW;
    const int gh = sampling_ratio > 0 ? sampling_ratio : (int)ceilf(rh / PH);
    const int gw = sampling_ratio > 0 ? sampling_ratio : (int)ceilf(rw / PW);
    const int cnt = max(gh * gw, 1);
    const dev_t* data = input + ((int64_t)b * C + c) * H * W;
    float acc = 0.f;
    for (int iy = 0; iy < gh; ++iy) {
      const float y = y1 + ph * bin_h + (iy + 0.5f) * bin_h / gh;
      for (int ix = 0; ix < gw; ++ix) {
        const float x = x1 + pw * bin_w + (ix + 0.5f) * bin_w / gw;
        acc += bilinear_interpolate(data, H, W, y, x);
      }
    }
    output[i] = from_f32<dev_t>(acc / cnt);
  }
}

template <typename dev_t>
__global__ void roialign_bwd_kernel(const dev_t* __restrict__ grad_out,
                                    const float* __restrict__ rois,
                                    float* __restrict__ grad_in, int R, int C,
                                    int H, int W, int PH, int PW,
                                    float spatial_scale, int sampling_ratio,
                                    bool aligned) {
  const int64_t total = (int64_t)R * C * PH * PW;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int pw = (int)(i % PW);
    const int ph = (int)((i / PW) % PH);
    const int c = (int)((i / PW / PH) % C);
    const int r = (int)(i / PW / PH / C);
    const float* roi = rois + r * 5;
    const int b = (int)roi[0];
    const float off = aligned ? 0.5f : 0.f;
    const float x1 = roi[1] * spatial_scale - off;
    const float y1 = roi[2] * spatial_scale - off;
    const float x2 = roi[3] * spatial_scale - off;
    const float y2 = roi[4] * spatial_scale - off;
    float rw = x2 - x1, rh = y2 - y1;
    if (!aligned) { rw = fmaxf(rw, 1.f); rh = fmaxf(rh, 1.f); }
    const float bin_h = rh / PH, bin_w = rw / PW;
    const int gh = sampling_ratio > 0 ? sampling_ratio : (int)ceilf(rh / PH);
    const int gw = sampling_ratio > 0 ? sampling_ratio : (int)ceilf(rw / PW);
    const int cnt = max(gh * gw, 1);
    const float g = to_f32(grad_out[i]) / cnt;
    float* gdata = grad_in + ((int64_t)b * C + c) * H * W;
    for (int iy = 0; iy < gh; ++iy) {
      float y = y1 + ph * bin_h + (iy + 0.5f) * bin_h / gh;
      for (int ix = 0; ix < gw; ++ix) {
        float x = x1 + pw * bin_w + (ix + 0.5f) * bin_w / gw;
        if (y < -1.f || y > H || x < -1.f || x > W) continue;
        float yy = fmaxf(y, 0.f), xx = fmaxf(x, 0.f);
        int y0 = (int)yy, x0 = (int)xx;
        int y1i = y0 + 1, x1i = x0 + 1;
        if (y0 >= H - 1) { y0 = y1i = H - 1; yy = (float)y0; }
        if (x0 >= W - 1) { x0 = x1i = W - 1; xx = (float)x0; }
        const float ly = yy - y0, lx = xx - x0;
        const float hy = 1.f - ly, hx = 1.f - lx;
        atomicAdd(&gdata[y0 * W + x0], g * hy * hx);
        atomicAdd(&gdata[y0 * W + x1i], g * hy * lx);
        atomicAdd(&gdata[y1i * W + x0], g * ly * hx);
        atomicAdd(&gdata[y1i * W + x1i], g * ly * lx);
      }
    }
  }
}

}  // namespace dla

torch::Tensor roialign_fwd(torch::Tensor input, torch::Tensor rois, int64_t PH,
                           int64_t PW, double spatial_scale,
                           int64_t sampling_ratio, bool aligned) {
  DLA_CHECK_INPUT(input);
  auto rf = rois.to(torch::kFloat).contiguous();
  const int R = (int)rf.size(0), C = (int)input.size(1);
  const int H = (int)input.size(2), W = (int)input.size(3);
  auto out = torch::empty({R, C, PH, PW}, input.options());
  if (R == 0) return out;
  const int64_t total = (int64_t)R * C * PH * PW;
  DLA_DISPATCH_FLOAT_TYPES(input.scalar_type(), "roialign_fwd", [&] {
    hipLaunchKernelGGL((dla::roialign_fwd_kernel<dev_t>),
                       dim3(dla::grid_1d(total, 256)), dim3(256), 0, dla::stream(),
                       (const dev_t*)input.data_ptr(), rf.data_ptr<float>(),
                       (dev_t*)out.data_ptr(), R, C, H, W, (int)PH, (int)PW,
                       (float)spatial_scale, (int)sampling_ratio, aligned);
  });
  HIP_CHECK_ERR();
  return out;
}

torch::Tensor roialign_bwd(torch::Tensor grad_out, torch::Tensor rois, int64_t N,
                           int64_t C, int64_t H, int64_t W, double spatial_scale,
                           int64_t sampling_ratio, bool aligned) {
  auto go = grad_out.contiguous();
  auto rf = rois.to(torch::kFloat).contiguous();
  const int R = (int)rf.size(0);
  const int PH = (int)go.size(2), PW = (int)go.size(3);
  auto grad_in = torch::zeros({N, C, H, W}, go.options().dtype(torch::kFloat));
  if (R > 0) {
    const int64_t total = (int64_t)R * C * PH * PW;
    DLA_DISPATCH_FLOAT_TYPES(go.scalar_type(), "roialign_bwd", [&] {
      hipLaunchKernelGGL((dla::roialign_bwd_kernel<dev_t>),
                         dim3(dla::grid_1d(total, 256)), dim3(256), 0,
                         dla::stream(), (const dev_t*)go.data_ptr(),
                         rf.data_ptr<float>(), grad_in.data_ptr<float>(), R,
                         (int)C, (int)H, (int)W, PH, PW, (float)spatial_scale,
                         (int)sampling_ratio, aligned);
    });
    HIP_CHECK_ERR();
  }
  return grad_in.to(go.scalar_type());
}
