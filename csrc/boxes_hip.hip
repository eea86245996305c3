#include "hip/hip_runtime.h"
// Box ops: pairwise IoU/GIoU matrices and NMS (torchvision-semantics).
// Reference: box_iou (RetinaNet network_files/boxes.py:154), GIoU
// (FCOS models/loss.py:388), torchvision.ops.nms call sites
// (yolov5 utils/general.py:694, batched_nms via class-offset trick).
//
// NMS design (wave64-native): boxes pre-sorted by score on the host; kernel
// fills a suppression bitmask with 64 boxes per block column (one ull per
// wave), final sequential sweep on CPU — identical output order to
// torchvision.ops.nms.
#include "common.h"

namespace dla {

__device__ __forceinline__ float box_area(float x1, float y1, float x2, float y2) {
  return fmaxf(x2 - x1, 0.f) * fmaxf(y2 - y1, 0.f);
}

__device__ __forceinline__ float iou_pair(const float* a, const float* b) {
  const float ix1 = fmaxf(a[0], b[0]), iy1 = fmaxf(a[1], b[1]);
  const float ix2 = fminf(a[2], b[2]), iy2 = fminf(a[3], b[3]);
  const float inter = box_area(ix1, iy1, ix2, iy2);
  const float ua = box_area(a[0], a[1], a[2], a[3]) +
                   box_area(b[0], b[1], b[2], b[3]) - inter;
  return ua > 0.f ? inter / ua : 0.f;
}

__global__ void box_iou_kernel(const float* __restrict__ a,
                               const float* __restrict__ b,
                               float* __restrict__ out, int N, int M, bool giou) {
  const int64_t total = (int64_t)N * M;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int n = (int)(i / M), m = (int)(i % M);
    const float* pa = a + 4 * n;
    const float* pb = b + 4 * m;
    float v = iou_pair(pa, pb);
    if (giou) {
      const float cx1 = fminf(pa[0], pb[0]), cy1 = fminf(pa[1], pb[1]);
      const float cx2 = fmaxf(pa[2], pb[2]), cy2 = fmaxf(pa[3], pb[3]);
      const float carea = box_area(cx1, cy1, cx2, cy2);
      if (carea > 0.f) {
        const float ix1 = fmaxf(pa[0], pb[0]), iy1 = fmaxf(pa[1], pb[1]);
        const float ix2 = fminf(pa[2], pb[2]), iy2 = fminf(pa[3], pb[3]);
        const float inter = box_area(ix1, iy1, ix2, iy2);
        const float uni = box_area(pa[0], pa[1], pa[2], pa[3]) +
                          box_area(pb[0], pb[1], pb[2], pb[3]) - inter;
        v = v - (carea - uni) / carea;
      }
    }
    out[i] = v;
  }
}

// NMS mask: block (i_wave, j_block): rows = 64 boxes, cols = 64 boxes.
// mask[i][jb] bit k set => box jb*64+k suppressed by box i (iou > thr, i < j).
constexpr int kNmsChunk = 64;

__global__ void nms_mask_kernel(const float* __restrict__ boxes,
                                unsigned long long* __restrict__ mask,
                                int n, int n_chunks, float thr) {
  const int row_chunk = blockIdx.y;
  const int col_chunk = blockIdx.x;
  if (col_chunk < row_chunk) return;  // only j >= i blocks matter
  const int row_size = min(n - row_chunk * kNmsChunk, kNmsChunk);
  const int col_size = min(n - col_chunk * kNmsChunk, kNmsChunk);

  __shared__ float col_boxes[kNmsChunk * 4];
  if (threadIdx.x < col_size) {
    const int j = col_chunk * kNmsChunk + threadIdx.x;
#pragma unroll
    for (int k = 0; k < 4; ++k) col_boxes[threadIdx.x * 4 + k] = boxes[j * 4 + k];
  }
  __syncthreads();
  if (threadIdx.x >= row_size) return;
  const int i = row_chunk * kNmsChunk + threadIdx.x;
  const float* pa = boxes + i * 4;
  unsigned long long bits = 0ull;
  const int start = (row_chunk == col_chunk) ? threadIdx.x + 1 : 0;
  for (int k = start; k < col_size; ++k) {
    if (iou_pair(pa, col_boxes + k * 4) > thr) bits |= (1ull << k);
  }
  mask[(int64_t)i * n_chunks + col_chunk] = bits;
}

}  // namespace dla

torch::Tensor box_iou_gpu(torch::Tensor a, torch::Tensor b, bool giou) {
  DLA_CHECK_CUDA(a); DLA_CHECK_CUDA(b);
  auto af = a.to(torch::kFloat).contiguous();
  auto bf = b.to(torch::kFloat).contiguous();
  const int N = (int)af.size(0), M = (int)bf.size(0);
  auto out = torch::empty({N, M}, af.options());
  if (N == 0 || M == 0) return out;
  const int grid = dla::grid_1d((int64_t)N * M, 256);
  hipLaunchKernelGGL(dla::box_iou_kernel, dim3(grid), dim3(256), 0, dla::stream(),
                     af.data_ptr<float>(), bf.data_ptr<float>(),
                     out.data_ptr<float>(), N, M, giou);
  HIP_CHECK_ERR();
  return out;
}

// boxes must already be sorted by score desc; returns keep indices into that order.
torch::Tensor nms_gpu(torch::Tensor boxes_sorted, double iou_threshold) {
  DLA_CHECK_CUDA(boxes_sorted);
  auto bf = boxes_sorted.to(torch::kFloat).contiguous();
  const int n = (int)bf.size(0);
  if (n == 0) return torch::empty({0}, bf.options().dtype(torch::kLong));
  const int n_chunks = (n + dla::kNmsChunk - 1) / dla::kNmsChunk;
  auto mask = torch::zeros({(int64_t)n * n_chunks},
                           bf.options().dtype(torch::kLong));
  dim3 grid(n_chunks, n_chunks);
  hipLaunchKernelGGL(dla::nms_mask_kernel, grid, dim3(dla::kNmsChunk), 0,
                     dla::stream(), bf.data_ptr<float>(),
                     (unsigned long long*)mask.data_ptr<int64_t>(), n, n_chunks,
                     (float)iou_threshold);
  HIP_CHECK_ERR();
  auto mask_cpu = mask.cpu();
  const unsigned long long* mp = (const unsigned long long*)mask_cpu.data_ptr<int64_t>();
  std::vector<unsigned long long> removed(n_chunks, 0ull);
  std::vector<int64_t> keep;
  keep.reserve(n);
  for (int i = 0; i < n; ++i) {
    const int chunk = i / dla::kNmsChunk, bit = i % dla::kNmsChunk;
    if (removed[chunk] & (1ull << bit)) continue;
    keep.push_back(i);
    const unsigned long long* row = mp + (int64_t)i * n_chunks;
    for (int c = chunk; c < n_chunks; ++c) removed[c] |= row[c];
  }
  return torch::tensor(keep, torch::dtype(torch::kLong))
      .to(boxes_sorted.device());
}
