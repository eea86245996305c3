// Fused multi-head attention forward for CDNA4 (gfx950).
//
// Replaces the eager softmax(Q K^T / sqrt(d) + bias + mask) V chain in
// ViT (classification/vision_transformer/vit_model.py:88-113), Swin window
// attention (classification/swin_transformer/models/swin_transformer.py:118-151),
// MAE and TransFG with ONE kernel launch per tensor:
//  - consumes the packed qkv projection [B, N, 3, H, d] directly (no permute
//    copies, no [B,H,N,N] fp32 score tensor, no separate softmax kernels);
//  - writes O in [B, N, H*d] (the layout the output projection wants);
//  - optionally writes P = softmax(S) in bf16 for the backward pass.
//
// Shape domain: d in {32, 64}, N <= 256 (ViT 197, Swin windows 49, MAE 50).
// Design: one 256-thread workgroup (4 waves) per (batch, head). K is staged
// in LDS [N][d] (row stride padded to dodge bank conflicts), V transposed in
// LDS [d][Npad]. Each wave owns 16 query rows at a time:
//   S(16xN) via mfma_f32_16x16x32_bf16 (A = Q rows, B = K^T from LDS),
//   row softmax with shuffle reductions over the 16-lane C/D columns,
//   P -> LDS bf16, O(16xd) via mfma (A = P, B = V^T from LDS).
#include "common.h"
#include "vec.h"

namespace dla {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// K LDS row padding (bf16 elements) so B-fragment reads (16 lanes x 16B,
// row stride apart) hit distinct banks: stride 72*2B=144B -> 36 dwords,
// 36*k mod 64 distinct for k in [0,16).
constexpr int KPAD = 72;

__device__ __forceinline__ float wave16_max(float v) {
  // max over the 16-lane group (lane bits 0..3)
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

__device__ __forceinline__ float wave16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// One workgroup per (b, h); template D in {32, 64}.
template <int D, bool HAS_BIAS, bool HAS_MASK, bool SAVE_P>
__global__ __launch_bounds__(256)
void attn_fwd_kernel(const __hip_bfloat16* __restrict__ qkv,  // [B,N,3,H,D]
                     const float* __restrict__ bias,          // [H,N,N] or null
                     const float* __restrict__ mask,          // [nW,N,N] or null
                     __hip_bfloat16* __restrict__ out,        // [B,N,H*D]
                     __hip_bfloat16* __restrict__ p_out,      // [B,H,N,N] or null
                     float* __restrict__ stats,               // [2,B,H,N] or null
                     int B, int N, int H, int n_win, float scale) {
  constexpr int KSLICES = D / 32;        // mfma K-steps over head dim
  const int b = blockIdx.x / H;
  const int h = blockIdx.x % H;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int Npad = (N + 15) & ~15;       // key tiles padded to 16
  const int n_ktiles = Npad / 16;

  extern __shared__ __attribute__((aligned(16))) char lds_raw[];
  __hip_bfloat16* k_lds = (__hip_bfloat16*)lds_raw;            // [Npad][KPAD]
  __hip_bfloat16* vt_lds = k_lds + Npad * KPAD;                // [D][Npad+8]
  const int VROW = Npad + 8;
  const int PROW = Npad + 8;  // P row stride (Npad keys + bank-dodge pad)
  __hip_bfloat16* p_lds = vt_lds + D * VROW;                   // [4][16][PROW]

  // ---- stage K ([N][D] rows) and V transposed ([D][Npad]) -------------------
  // qkv element (b, n, c, h, d) at (((b*N + n)*3 + c)*H + h)*D + d
  const int64_t bh_stride = (int64_t)3 * H * D;
  const __hip_bfloat16* k_src = qkv + ((int64_t)b * N * 3 + 1) * H * D +
                                (int64_t)h * D;
  const __hip_bfloat16* v_src = qkv + ((int64_t)b * N * 3 + 2) * H * D +
                                (int64_t)h * D;
  for (int n = threadIdx.x / (D / 4); n < Npad; n += 256 / (D / 4)) {
    const int d0 = (threadIdx.x % (D / 4)) * 4;
    Vec<__hip_bfloat16, 4> kv, vv;
    if (n < N) {
      kv = vload<__hip_bfloat16, 4>(k_src + n * bh_stride + d0);
      vv = vload<__hip_bfloat16, 4>(v_src + n * bh_stride + d0);
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        kv.v[j] = __hip_bfloat16(0.f);
        vv.v[j] = __hip_bfloat16(0.f);
      }
    }
    vstore<__hip_bfloat16, 4>(&k_lds[n * KPAD + d0], kv);
#pragma unroll
    for (int j = 0; j < 4; ++j) vt_lds[(d0 + j) * VROW + n] = vv.v[j];
  }
  __syncthreads();

  const __hip_bfloat16* q_src = qkv + (int64_t)b * N * 3 * H * D +
                                (int64_t)h * D;
  const int n_qblocks = (N + 15) / 16;

  for (int qb = wave; qb < n_qblocks; qb += 4) {
    const int q0 = qb * 16;
    // ---- load Q fragment: A of mfma = Q[16 rows][32 k], row=lane&15,
    // k=(lane>>4)*8+j  -> per lane 8 consecutive d -> one 16B load per slice
    bf16x8 q_frag[KSLICES];
    {
      const int row = q0 + (lane & 15);
      const int srow = row < N ? row : N - 1;
#pragma unroll
      for (int s = 0; s < KSLICES; ++s) {
        const int d0 = s * 32 + (lane >> 4) * 8;
        q_frag[s] = *(const bf16x8*)(q_src + srow * bh_stride + d0);
      }
    }

    // ---- S = scale * Q K^T over all key tiles -------------------------------
    // NOTE: every loop over s_acc is fully unrolled with a compile-time bound
    // so the accumulator array stays in VGPRs (a runtime-indexed array goes
    // to scratch = global memory; measured 60x slowdown).
    f32x4 s_acc[16];  // up to 16 key tiles (N<=256)
#pragma unroll
    for (int kt = 0; kt < 16; ++kt) {
      if (kt < n_ktiles) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  #pragma unroll
        for (int s = 0; s < KSLICES; ++s) {
          // B fragment: col=lane&15 (key), k=(lane>>4)*8+j
          const int key = kt * 16 + (lane & 15);
          const int d0 = s * 32 + (lane >> 4) * 8;
          bf16x8 k_frag = *(const bf16x8*)(&k_lds[key * KPAD + d0]);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[s], k_frag, acc,
                                                        0, 0, 0);
        }
        s_acc[kt] = acc;
      }
    }

    // ---- softmax over each of the 16 rows this lane-group covers ------------
    // C/D: col = lane&15 (key within tile), row = (lane>>4)*4 + reg
    const int col_in_tile = lane & 15;
    float row_max[4], row_sum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m = -INFINITY;
      const int qrow = q0 + (lane >> 4) * 4 + r;
#pragma unroll
      for (int kt = 0; kt < 16; ++kt) {
        if (kt < n_ktiles) {
          const int key = kt * 16 + col_in_tile;
          float s = s_acc[kt][r] * scale;
          if ((HAS_BIAS || HAS_MASK) && key < N && qrow < N) {
            if (HAS_BIAS) s += bias[((int64_t)h * N + qrow) * N + key];
            if (HAS_MASK) s += mask[((int64_t)(b % n_win) * N + qrow) * N + key];
          }
          if (key >= N) s = -INFINITY;
          s_acc[kt][r] = s;
          m = fmaxf(m, s);
        }
      }
      m = wave16_max(m);
      float sum = 0.f;
#pragma unroll
      for (int kt = 0; kt < 16; ++kt) {
        if (kt < n_ktiles) {
          const float p = __expf(s_acc[kt][r] - m);
          s_acc[kt][r] = p;
          sum += p;
        }
      }
      row_max[r] = m;
      row_sum[r] = wave16_sum(sum);
      if (stats != nullptr && (lane & 15) == 0) {
        const int qrow = q0 + (lane >> 4) * 4 + r;
        if (qrow < N) {
          stats[((int64_t)b * H + h) * N + qrow] = m;
          stats[((int64_t)(B + b) * H + h) * N + qrow] = row_sum[r];
        }
      }
    }

    // ---- P -> LDS (bf16), per-wave buffer [16 rows][KPAD row stride] --------
    __hip_bfloat16* p_buf = p_lds + wave * 16 * PROW;
    // rows are interleaved across lanes; each lane writes its 4 elements/tile
#pragma unroll
    for (int kt = 0; kt < 16; ++kt) {
      if (kt < n_ktiles) {
  #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = (lane >> 4) * 4 + r;
          p_buf[row * PROW + kt * 16 + col_in_tile] =
              __hip_bfloat16(s_acc[kt][r]);
        }
      }
    }
    if (SAVE_P) {
      const int qrow_base = q0;
#pragma unroll
      for (int kt = 0; kt < 16; ++kt) {
        if (kt < n_ktiles) {
  #pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int qrow = qrow_base + (lane >> 4) * 4 + r;
            const int key = kt * 16 + col_in_tile;
            if (qrow < N && key < N)
              p_out[(((int64_t)b * H + h) * N + qrow) * N + key] =
                  __hip_bfloat16(s_acc[kt][r] / row_sum[r]);
          }
        }
      }
    }
    // no cross-wave LDS sharing of p_buf: same wave writes then reads.
    // s_waitcnt lgkmcnt is enough; compiler inserts it for LDS dependences.

    // ---- O = P V : A = P[16 rows][32 keys], B = V^T[32 keys][16 d] ----------
    f32x4 o_acc[D / 16];
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) o_acc[dt] = {0.f, 0.f, 0.f, 0.f};
    for (int kt2 = 0; kt2 < Npad / 32; ++kt2) {
      // A fragment: row=lane&15, k=(lane>>4)*8+j  (keys)
      const int row = lane & 15;
      const int k0 = kt2 * 32 + (lane >> 4) * 8;
      bf16x8 p_frag = *(const bf16x8*)(&p_buf[row * PROW + k0]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        // B fragment: col=lane&15 (d), k=(lane>>4)*8+j (keys) from vt_lds
        const int d = dt * 16 + (lane & 15);
        bf16x8 v_frag = *(const bf16x8*)(&vt_lds[d * VROW + k0]);
        o_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag, v_frag,
                                                            o_acc[dt], 0, 0, 0);
      }
    }
    // Npad may have a 16-key tail not covered by the 32-key loop
    if (Npad % 32) {
      const int row = lane & 15;
      const int k0 = (Npad / 32) * 32 + (lane >> 4) * 8;
      bf16x8 p_frag{};
      if ((lane >> 4) * 8 < 16)  // only first 16 keys valid in this half tile
        p_frag = *(const bf16x8*)(&p_buf[row * PROW + k0]);
      else
#pragma unroll
        for (int j = 0; j < 8; ++j) p_frag[j] = 0;
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int d = dt * 16 + (lane & 15);
        bf16x8 v_frag{};
        if ((lane >> 4) * 8 < 16)
          v_frag = *(const bf16x8*)(&vt_lds[d * VROW + k0]);
        else
#pragma unroll
          for (int j = 0; j < 8; ++j) v_frag[j] = 0;
        o_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag, v_frag,
                                                            o_acc[dt], 0, 0, 0);
      }
    }

    // ---- write O / row_sum normalize: out[b][qrow][h*D + d] -----------------
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + (lane >> 4) * 4 + r;
        if (qrow >= N) continue;
        const int d = dt * 16 + (lane & 15);
        out[((int64_t)b * N + qrow) * H * D + h * D + d] =
            __hip_bfloat16(o_acc[dt][r] / row_sum[r]);
      }
    }
  }
}

// ---- fused attention backward (no bias/mask; flash-style recompute) --------
// Saved from fwd: row max m and row sum l ([B,H,N] f32 each). D_row =
// rowsum(dO * O) is computed by the host in one fused reduce.
//   P = exp(scale*QK^T - m)/l
//   dV = P^T dO ; dP = dO V^T ; dS = P*(dP - D_row) ; dQ = scale*dS K ;
//   dK = scale*dS^T Q
// Kernel A (dK/dV): per (b,h) workgroup; Q and dO staged in LDS both
// row-major (B-fragments of S^T / dP^T) and transposed (B-fragments of the
// dK / dV MFMAs); each wave owns 16 keys per iteration, K/V fragments read
// straight from global.
// Kernel B (dQ): K and V row-major + K transposed in LDS; each wave owns 16
// queries.

template <int D>
__global__ __launch_bounds__(256)
void attn_bwd_kv_kernel(const __hip_bfloat16* __restrict__ qkv,
                        const __hip_bfloat16* __restrict__ dout,  // [B,N,H*D]
                        const float* __restrict__ stats,          // [2,B,H,N]
                        const float* __restrict__ drow,           // [B,H,N]
                        __hip_bfloat16* __restrict__ dqkv,
                        int B, int N, int H, float scale) {
  constexpr int KSLICES = D / 32;
  const int b = blockIdx.x / H;
  const int h = blockIdx.x % H;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int Npad = (N + 15) & ~15;
  const int n_qtiles = Npad / 16;

  extern __shared__ __attribute__((aligned(16))) char lds_raw[];
  __hip_bfloat16* q_lds = (__hip_bfloat16*)lds_raw;        // [Npad][KPAD]
  __hip_bfloat16* do_lds = q_lds + Npad * KPAD;            // [Npad][KPAD]
  const int VROW = Npad + 8;
  __hip_bfloat16* qt_lds = do_lds + Npad * KPAD;           // [D][VROW]
  __hip_bfloat16* dot_lds = qt_lds + D * VROW;             // [D][VROW]
  const int PROW = Npad + 8;
  __hip_bfloat16* p_lds = dot_lds + D * VROW;              // [4][16][PROW]

  const int64_t bh_stride = (int64_t)3 * H * D;
  const __hip_bfloat16* q_src = qkv + (int64_t)b * N * 3 * H * D +
                                (int64_t)h * D;
  const __hip_bfloat16* do_src = dout + (int64_t)b * N * H * D +
                                 (int64_t)h * D;
  for (int n = threadIdx.x / (D / 4); n < Npad; n += 256 / (D / 4)) {
    const int d0 = (threadIdx.x % (D / 4)) * 4;
    Vec<__hip_bfloat16, 4> qv, dv;
    if (n < N) {
      qv = vload<__hip_bfloat16, 4>(q_src + n * bh_stride + d0);
      dv = vload<__hip_bfloat16, 4>(do_src + (int64_t)n * H * D + d0);
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        qv.v[j] = __hip_bfloat16(0.f);
        dv.v[j] = __hip_bfloat16(0.f);
      }
    }
    vstore<__hip_bfloat16, 4>(&q_lds[n * KPAD + d0], qv);
    vstore<__hip_bfloat16, 4>(&do_lds[n * KPAD + d0], dv);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      qt_lds[(d0 + j) * VROW + n] = qv.v[j];
      dot_lds[(d0 + j) * VROW + n] = dv.v[j];
    }
  }
  __syncthreads();

  const float* m_arr = stats + ((int64_t)b * H + h) * N;
  const float* l_arr = stats + ((int64_t)(B + b) * H + h) * N;
  const float* d_arr = drow + ((int64_t)b * H + h) * N;
  const __hip_bfloat16* k_src = qkv + ((int64_t)b * N * 3 + 1) * H * D +
                                (int64_t)h * D;
  const __hip_bfloat16* v_src = qkv + ((int64_t)b * N * 3 + 2) * H * D +
                                (int64_t)h * D;
  __hip_bfloat16* p_buf = p_lds + wave * 16 * PROW;
  const int n_ktiles_own = (N + 15) / 16;

  for (int kt = wave; kt < n_ktiles_own; kt += 4) {
    const int key0 = kt * 16;
    // K_tile / V_tile A-fragments from global: row=key, k=d
    bf16x8 k_frag[KSLICES], v_frag[KSLICES];
    {
      const int key = key0 + (lane & 15);
      const int skey = key < N ? key : N - 1;
#pragma unroll
      for (int sl = 0; sl < KSLICES; ++sl) {
        const int d0 = sl * 32 + (lane >> 4) * 8;
        k_frag[sl] = *(const bf16x8*)(k_src + skey * bh_stride + d0);
        v_frag[sl] = *(const bf16x8*)(v_src + skey * bh_stride + d0);
      }
    }
    // per q-tile: S^T and dP^T accumulators (full unroll: keep in VGPRs)
    f32x4 st_acc[16], dpt_acc[16];
#pragma unroll
    for (int qt = 0; qt < 16; ++qt) {
      if (qt < n_qtiles) {
        f32x4 sacc = {0.f, 0.f, 0.f, 0.f}, dacc = {0.f, 0.f, 0.f, 0.f};
  #pragma unroll
        for (int sl = 0; sl < KSLICES; ++sl) {
          const int q = qt * 16 + (lane & 15);
          const int d0 = sl * 32 + (lane >> 4) * 8;
          bf16x8 qb = *(const bf16x8*)(&q_lds[q * KPAD + d0]);
          bf16x8 db = *(const bf16x8*)(&do_lds[q * KPAD + d0]);
          sacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(k_frag[sl], qb, sacc,
                                                         0, 0, 0);
          dacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(v_frag[sl], db, dacc,
                                                         0, 0, 0);
        }
        st_acc[qt] = sacc;
        dpt_acc[qt] = dacc;
      }
    }
    // P^T and dS^T: C/D col = q (lane&15), row = key ((lane>>4)*4+reg)
    const int qcol_base = lane & 15;
#pragma unroll
    for (int qt = 0; qt < 16; ++qt) {
      if (qt < n_qtiles) {
        const int q = qt * 16 + qcol_base;
        const float m = q < N ? m_arr[q] : 0.f;
        const float li = q < N ? 1.f / l_arr[q] : 0.f;
        const float dr = q < N ? d_arr[q] : 0.f;
  #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float pt = __expf(st_acc[qt][r] * scale - m) * li;
          st_acc[qt][r] = pt;                                   // P^T
          dpt_acc[qt][r] = pt * (dpt_acc[qt][r] - dr) * scale;  // scale*dS^T
        }
      }
    }
    // ---- dV = P^T dO : stage P^T, MFMA over q ------------------------------
#pragma unroll
    for (int qt = 0; qt < 16; ++qt) {
      if (qt < n_qtiles) {
  #pragma unroll
        for (int r = 0; r < 4; ++r)
          p_buf[((lane >> 4) * 4 + r) * PROW + qt * 16 + qcol_base] =
              __hip_bfloat16(st_acc[qt][r]);
      }
    }
    f32x4 dv_acc[D / 16], dk_acc[D / 16];
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
      dv_acc[dt] = {0.f, 0.f, 0.f, 0.f};
      dk_acc[dt] = {0.f, 0.f, 0.f, 0.f};
    }
    for (int q2 = 0; q2 + 32 <= Npad; q2 += 32) {
      const int k0 = q2 + (lane >> 4) * 8;
      bf16x8 pt_frag = *(const bf16x8*)(&p_buf[(lane & 15) * PROW + k0]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int d = dt * 16 + (lane & 15);
        bf16x8 dob = *(const bf16x8*)(&dot_lds[d * VROW + k0]);
        dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pt_frag, dob,
                                                             dv_acc[dt], 0, 0, 0);
      }
    }
    if (Npad % 32) {  // 16-query half tile: upper k-halves masked to zero
      const int k0 = (Npad & ~31) + (lane >> 4) * 8;
      const bool lo = (lane >> 4) * 8 < 16;
      bf16x8 pt_frag{};
      if (lo) pt_frag = *(const bf16x8*)(&p_buf[(lane & 15) * PROW + k0]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int d = dt * 16 + (lane & 15);
        bf16x8 dob{};
        if (lo) dob = *(const bf16x8*)(&dot_lds[d * VROW + k0]);
        dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pt_frag, dob,
                                                             dv_acc[dt], 0, 0, 0);
      }
    }
    // ---- dK = (scale dS)^T Q : stage dS^T, MFMA over q ---------------------
    // p_buf rewrite is wave-local; same-wave LDS ops stay ordered (no block
    // barrier here - waves have different kt trip counts).
#pragma unroll
    for (int qt = 0; qt < 16; ++qt) {
      if (qt < n_qtiles) {
  #pragma unroll
        for (int r = 0; r < 4; ++r)
          p_buf[((lane >> 4) * 4 + r) * PROW + qt * 16 + qcol_base] =
              __hip_bfloat16(dpt_acc[qt][r]);
      }
    }
    for (int q2 = 0; q2 + 32 <= Npad; q2 += 32) {
      const int k0 = q2 + (lane >> 4) * 8;
      bf16x8 ds_frag = *(const bf16x8*)(&p_buf[(lane & 15) * PROW + k0]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int d = dt * 16 + (lane & 15);
        bf16x8 qb = *(const bf16x8*)(&qt_lds[d * VROW + k0]);
        dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ds_frag, qb,
                                                             dk_acc[dt], 0, 0, 0);
      }
    }
    if (Npad % 32) {
      const int k0 = (Npad & ~31) + (lane >> 4) * 8;
      const bool lo = (lane >> 4) * 8 < 16;
      bf16x8 ds_frag{};
      if (lo) ds_frag = *(const bf16x8*)(&p_buf[(lane & 15) * PROW + k0]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int d = dt * 16 + (lane & 15);
        bf16x8 qb{};
        if (lo) qb = *(const bf16x8*)(&qt_lds[d * VROW + k0]);
        dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ds_frag, qb,
                                                             dk_acc[dt], 0, 0, 0);
      }
    }
    // ---- write dK (comp 1) and dV (comp 2): C/D row=key, col=d -------------
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = key0 + (lane >> 4) * 4 + r;
        if (key >= N) continue;
        const int d = dt * 16 + (lane & 15);
        const int64_t base = ((int64_t)(b * N + key) * 3) * H * D +
                             (int64_t)h * D + d;
        dqkv[base + (int64_t)H * D] = __hip_bfloat16(dk_acc[dt][r]);
        dqkv[base + (int64_t)2 * H * D] = __hip_bfloat16(dv_acc[dt][r]);
      }
    }
  }
}

template <int D>
__global__ __launch_bounds__(256)
void attn_bwd_q_kernel(const __hip_bfloat16* __restrict__ qkv,
                       const __hip_bfloat16* __restrict__ dout,
                       const float* __restrict__ stats,
                       const float* __restrict__ drow,
                       __hip_bfloat16* __restrict__ dqkv,
                       int B, int N, int H, float scale) {
  constexpr int KSLICES = D / 32;
  const int b = blockIdx.x / H;
  const int h = blockIdx.x % H;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int Npad = (N + 15) & ~15;
  const int n_ktiles = Npad / 16;

  extern __shared__ __attribute__((aligned(16))) char lds_raw[];
  __hip_bfloat16* k_lds = (__hip_bfloat16*)lds_raw;        // [Npad][KPAD]
  __hip_bfloat16* v_lds = k_lds + Npad * KPAD;             // [Npad][KPAD]
  const int VROW = Npad + 8;
  __hip_bfloat16* kt_lds = v_lds + Npad * KPAD;            // [D][VROW]
  const int PROW = Npad + 8;
  __hip_bfloat16* p_lds = kt_lds + D * VROW;               // [4][16][PROW]

  const int64_t bh_stride = (int64_t)3 * H * D;
  const __hip_bfloat16* k_src = qkv + ((int64_t)b * N * 3 + 1) * H * D +
                                (int64_t)h * D;
  const __hip_bfloat16* v_src = qkv + ((int64_t)b * N * 3 + 2) * H * D +
                                (int64_t)h * D;
  for (int n = threadIdx.x / (D / 4); n < Npad; n += 256 / (D / 4)) {
    const int d0 = (threadIdx.x % (D / 4)) * 4;
    Vec<__hip_bfloat16, 4> kv, vv;
    if (n < N) {
      kv = vload<__hip_bfloat16, 4>(k_src + n * bh_stride + d0);
      vv = vload<__hip_bfloat16, 4>(v_src + n * bh_stride + d0);
    } else {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        kv.v[j] = __hip_bfloat16(0.f);
        vv.v[j] = __hip_bfloat16(0.f);
      }
    }
    vstore<__hip_bfloat16, 4>(&k_lds[n * KPAD + d0], kv);
    vstore<__hip_bfloat16, 4>(&v_lds[n * KPAD + d0], vv);
#pragma unroll
    for (int j = 0; j < 4; ++j) kt_lds[(d0 + j) * VROW + n] = kv.v[j];
  }
  __syncthreads();

  const float* m_arr = stats + ((int64_t)b * H + h) * N;
  const float* l_arr = stats + ((int64_t)(B + b) * H + h) * N;
  const float* d_arr = drow + ((int64_t)b * H + h) * N;
  const __hip_bfloat16* q_src = qkv + (int64_t)b * N * 3 * H * D +
                                (int64_t)h * D;
  const __hip_bfloat16* do_src = dout + (int64_t)b * N * H * D +
                                 (int64_t)h * D;
  __hip_bfloat16* p_buf = p_lds + wave * 16 * PROW;
  const int n_qblocks = (N + 15) / 16;

  for (int qb = wave; qb < n_qblocks; qb += 4) {
    const int q0 = qb * 16;
    bf16x8 q_frag[KSLICES], do_frag[KSLICES];
    {
      const int row = q0 + (lane & 15);
      const int srow = row < N ? row : N - 1;
#pragma unroll
      for (int sl = 0; sl < KSLICES; ++sl) {
        const int d0 = sl * 32 + (lane >> 4) * 8;
        q_frag[sl] = *(const bf16x8*)(q_src + srow * bh_stride + d0);
        do_frag[sl] = *(const bf16x8*)(do_src + (int64_t)srow * H * D + d0);
      }
    }
    f32x4 s_acc[16], dp_acc[16];
#pragma unroll
    for (int kt = 0; kt < 16; ++kt) {
      if (kt < n_ktiles) {
        f32x4 sacc = {0.f, 0.f, 0.f, 0.f}, dacc = {0.f, 0.f, 0.f, 0.f};
  #pragma unroll
        for (int sl = 0; sl < KSLICES; ++sl) {
          const int key = kt * 16 + (lane & 15);
          const int d0 = sl * 32 + (lane >> 4) * 8;
          bf16x8 kb = *(const bf16x8*)(&k_lds[key * KPAD + d0]);
          bf16x8 vb = *(const bf16x8*)(&v_lds[key * KPAD + d0]);
          sacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[sl], kb, sacc,
                                                         0, 0, 0);
          dacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(do_frag[sl], vb, dacc,
                                                         0, 0, 0);
        }
        s_acc[kt] = sacc;
        dp_acc[kt] = dacc;
      }
    }
    // ds = scale * P * (dP - Drow); C/D row = (lane>>4)*4+reg (query)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + (lane >> 4) * 4 + r;
      const float m = qrow < N ? m_arr[qrow] : 0.f;
      const float li = qrow < N ? 1.f / l_arr[qrow] : 0.f;
      const float dr = qrow < N ? d_arr[qrow] : 0.f;
#pragma unroll
      for (int kt = 0; kt < 16; ++kt) {
        if (kt < n_ktiles) {
          const int key = kt * 16 + (lane & 15);
          float pv = (key < N) ? __expf(s_acc[kt][r] * scale - m) * li : 0.f;
          s_acc[kt][r] = pv * (dp_acc[kt][r] - dr) * scale;
        }
      }
    }
#pragma unroll
    for (int kt = 0; kt < 16; ++kt) {
      if (kt < n_ktiles) {
  #pragma unroll
        for (int r = 0; r < 4; ++r)
          p_buf[((lane >> 4) * 4 + r) * PROW + kt * 16 + (lane & 15)] =
              __hip_bfloat16(s_acc[kt][r]);
      }
    }
    f32x4 dq_acc[D / 16];
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) dq_acc[dt] = {0.f, 0.f, 0.f, 0.f};
    for (int k2 = 0; k2 + 32 <= Npad; k2 += 32) {
      const int k0 = k2 + (lane >> 4) * 8;
      bf16x8 ds_frag = *(const bf16x8*)(&p_buf[(lane & 15) * PROW + k0]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int d = dt * 16 + (lane & 15);
        bf16x8 kb = *(const bf16x8*)(&kt_lds[d * VROW + k0]);
        dq_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ds_frag, kb,
                                                             dq_acc[dt], 0, 0, 0);
      }
    }
    if (Npad % 32) {
      const int k0 = (Npad & ~31) + (lane >> 4) * 8;
      const bool lo = (lane >> 4) * 8 < 16;
      bf16x8 ds_frag{};
      if (lo) ds_frag = *(const bf16x8*)(&p_buf[(lane & 15) * PROW + k0]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int d = dt * 16 + (lane & 15);
        bf16x8 kb{};
        if (lo) kb = *(const bf16x8*)(&kt_lds[d * VROW + k0]);
        dq_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ds_frag, kb,
                                                             dq_acc[dt], 0, 0, 0);
      }
    }
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + (lane >> 4) * 4 + r;
        if (qrow >= N) continue;
        const int d = dt * 16 + (lane & 15);
        dqkv[((int64_t)(b * N + qrow) * 3) * H * D + (int64_t)h * D + d] =
            __hip_bfloat16(dq_acc[dt][r]);
      }
    }
  }
}

// numerics probe: D[16,16] = A[16,32] @ B[32,16] via one mfma, to pin the
// fragment layout on real hardware (guide: A=I with asymmetric B).
__global__ void mfma_probe_kernel(const __hip_bfloat16* A,
                                  const __hip_bfloat16* B, float* D_out) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int row = lane & 15, k = (lane >> 4) * 8 + j;
    a[j] = ((const short*)A)[row * 32 + k];
    const int col = lane & 15;
    b[j] = ((const short*)B)[k * 16 + col];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = (lane >> 4) * 4 + r, col = lane & 15;
    D_out[row * 16 + col] = acc[r];
  }
}

}  // namespace dla

std::vector<torch::Tensor> attn_fwd(torch::Tensor qkv, int64_t num_heads,
                                    double scale,
                                    c10::optional<torch::Tensor> bias,
                                    c10::optional<torch::Tensor> mask,
                                    bool save_p, bool save_stats) {
  // qkv: [B, N, 3*H*D] or [B, N, 3, H, D] contiguous bf16
  TORCH_CHECK(qkv.scalar_type() == at::kBFloat16, "attn_fwd wants bf16 qkv");
  TORCH_CHECK(qkv.is_contiguous(), "attn_fwd wants contiguous qkv");
  const int B = (int)qkv.size(0);
  const int N = (int)qkv.size(1);
  const int H = (int)num_heads;
  const int64_t inner = qkv.numel() / ((int64_t)B * N);
  const int D = (int)(inner / (3 * H));
  TORCH_CHECK(inner == (int64_t)3 * H * D, "qkv inner dim mismatch");
  TORCH_CHECK(D == 32 || D == 64, "attn_fwd supports head_dim 32/64, got ", D);
  TORCH_CHECK(N <= 256, "attn_fwd supports N <= 256, got ", N);

  auto out = torch::empty({B, N, (int64_t)H * D}, qkv.options());
  torch::Tensor p, stats;
  if (save_p) p = torch::empty({B, H, N, N}, qkv.options());
  if (save_stats)
    stats = torch::empty({2, B, H, N}, qkv.options().dtype(torch::kFloat));

  int n_win = 1;
  const float* bias_ptr = nullptr;
  const float* mask_ptr = nullptr;
  torch::Tensor bias_f, mask_f;
  if (bias.has_value()) {
    bias_f = bias->to(torch::kFloat).contiguous();
    TORCH_CHECK(bias_f.dim() == 3 && bias_f.size(1) == N, "bias must be [H,N,N]");
    bias_ptr = bias_f.data_ptr<float>();
  }
  if (mask.has_value()) {
    mask_f = mask->to(torch::kFloat).contiguous();
    TORCH_CHECK(mask_f.dim() == 3 && mask_f.size(1) == N, "mask must be [nW,N,N]");
    n_win = (int)mask_f.size(0);
    mask_ptr = mask_f.data_ptr<float>();
  }

  const int Npad = (N + 15) & ~15;
  const int lds = (Npad * dla::KPAD + D * (Npad + 8) + 4 * 16 * (Npad + 8)) *
                  sizeof(__hip_bfloat16);
  dim3 grid(B * H), block(256);

  auto launch = [&](auto dtag, auto btag, auto mtag, auto ptag) {
    constexpr int DD = decltype(dtag)::value;
    constexpr bool BB = decltype(btag)::value;
    constexpr bool MM = decltype(mtag)::value;
    constexpr bool PP = decltype(ptag)::value;
    hipLaunchKernelGGL((dla::attn_fwd_kernel<DD, BB, MM, PP>), grid, block,
                       lds, dla::stream(),
                       (const __hip_bfloat16*)qkv.data_ptr(), bias_ptr,
                       mask_ptr, (__hip_bfloat16*)out.data_ptr(),
                       save_p ? (__hip_bfloat16*)p.data_ptr() : nullptr,
                       save_stats ? stats.data_ptr<float>() : nullptr,
                       B, N, H, n_win, (float)scale);
  };
  auto d3 = [&](auto dtag) {
    constexpr int DD = decltype(dtag)::value;
    using T = std::true_type;
    using F = std::false_type;
    const bool bb = bias_ptr != nullptr, mm = mask_ptr != nullptr;
    if (bb && mm) save_p ? launch(dtag, T{}, T{}, T{})
                          : launch(dtag, T{}, T{}, F{});
    else if (bb) save_p ? launch(dtag, T{}, F{}, T{})
                         : launch(dtag, T{}, F{}, F{});
    else if (mm) save_p ? launch(dtag, F{}, T{}, T{})
                         : launch(dtag, F{}, T{}, F{});
    else save_p ? launch(dtag, F{}, F{}, T{})
                 : launch(dtag, F{}, F{}, F{});
  };
  if (D == 64) d3(std::integral_constant<int, 64>{});
  else d3(std::integral_constant<int, 32>{});
  HIP_CHECK_ERR();
  std::vector<torch::Tensor> res = {out};
  if (save_p) res.push_back(p);
  if (save_stats) res.push_back(stats);
  return res;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 && B.scalar_type() == at::kBFloat16);
  auto D = torch::zeros({16, 16}, A.options().dtype(torch::kFloat));
  hipLaunchKernelGGL(dla::mfma_probe_kernel, dim3(1), dim3(64), 0,
                     dla::stream(), (const __hip_bfloat16*)A.data_ptr(),
                     (const __hip_bfloat16*)B.data_ptr(),
                     D.data_ptr<float>());
  HIP_CHECK_ERR();
  return D;
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor qkv, torch::Tensor dout,
                                    torch::Tensor stats, torch::Tensor drow,
                                    int64_t num_heads, double scale) {
  TORCH_CHECK(qkv.is_contiguous() && dout.is_contiguous());
  const int B = (int)qkv.size(0);
  const int N = (int)qkv.size(1);
  const int H = (int)num_heads;
  const int D = (int)(qkv.numel() / ((int64_t)B * N * 3 * H));
  TORCH_CHECK(D == 32 || D == 64);
  auto dqkv = torch::empty_like(qkv);
  const int Npad = (N + 15) & ~15;
  dim3 grid(B * H), block(256);
  const int lds_kv = (2 * Npad * dla::KPAD + 2 * D * (Npad + 8) +
                      4 * 16 * (Npad + 8)) * 2;
  const int lds_q = (2 * Npad * dla::KPAD + D * (Npad + 8) +
                     4 * 16 * (Npad + 8)) * 2;
  auto run = [&](auto dtag) {
    constexpr int DD = decltype(dtag)::value;
    hipLaunchKernelGGL((dla::attn_bwd_kv_kernel<DD>), grid, block, lds_kv,
                       dla::stream(), (const __hip_bfloat16*)qkv.data_ptr(),
                       (const __hip_bfloat16*)dout.data_ptr(),
                       stats.data_ptr<float>(), drow.data_ptr<float>(),
                       (__hip_bfloat16*)dqkv.data_ptr(), B, N, H,
                       (float)scale);
    hipLaunchKernelGGL((dla::attn_bwd_q_kernel<DD>), grid, block, lds_q,
                       dla::stream(), (const __hip_bfloat16*)qkv.data_ptr(),
                       (const __hip_bfloat16*)dout.data_ptr(),
                       stats.data_ptr<float>(), drow.data_ptr<float>(),
                       (__hip_bfloat16*)dqkv.data_ptr(), B, N, H,
                       (float)scale);
  };
  if (D == 64) run(std::integral_constant<int, 64>{});
  else run(std::integral_constant<int, 32>{});
  HIP_CHECK_ERR();
  return {dqkv};
}
