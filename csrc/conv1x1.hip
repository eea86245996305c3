// 1x1 convolution as an implicit-GEMM MFMA kernel (gfx950 / CDNA4), the
// north-star conv path of BASELINE.json: NHWC conv1x1 IS the GEMM
//   C[M,N] = A[M,K] @ B[N,K]^T,  M = batch*H*W, K = Cin, N = Cout,
// with the weight consumed in its native [Cout][Cin] layout (B^T), so the
// SAME kernel serves forward (A=x, B=W) and data-grad (A=dy, B=W^T).
//
// Replaces the reference's nn.Conv2d 1x1 call sites
// (classification/resnet/models/networks.py:27-35 Bottleneck conv1/conv3 and
// downsample) with a hand-written CDNA4 kernel:
//  - 128xBN tile (BN 64/128), BK=64, 4 waves, mfma_f32_16x16x32_bf16
//  - global->LDS staging via __builtin_amdgcn_global_load_lds (16B pieces),
//    double-buffered, one barrier per K-step (guide §5 step-3 structure)
//  - XOR-swizzled LDS image (guide T2 / G4: byte ^= (row&7)<<4 applied on the
//    glds SOURCE address and again on the ds_read_b128 fragment reads)
//  - fused epilogue: optional per-channel bias / scale+shift (eval-mode BN) /
//    residual add / ReLU, and optional per-channel sum+sumsq accumulation so
//    train-mode BatchNorm needs NO separate stats pass over the output
//    (sums layout matches batchnorm.hip: sums[c], sums[C+c])
//  - epilogue stores go through an LDS f32 transpose so global writes are
//    row-major 8B/lane coalesced (fragment layout is column-per-lane)
//
// wgrad: dW[N][K] = sum_m dy[m][n] * x[m][k] — separate kernel contracting
// over M with transposed LDS images (scatter ds_write staging, m-contiguous
// fragment reads), fp32 atomicAdd accumulation over M-split blocks.
#include <cstdlib>

#include "common.h"
#include "vec.h"

namespace dla {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

using bf16 = __hip_bfloat16;

// global->LDS direct copy, 16 bytes per lane (guide §5: the 16B width is the
// fast variant; dest must be wave-uniform base + lane*16).
__device__ __forceinline__ void glds16(const bf16* gsrc, bf16* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds_dst, 16, 0, 0);
}

// ---------------------------------------------------------------------------
// conv1x1 forward / dgrad kernel: C[M,N] = A[M,K] @ B[N,K]^T (bf16 in/out,
// fp32 accumulate). K % 64 == 0, N % BN == 0.
// ---------------------------------------------------------------------------
// TAPS=1: plain 1x1 GEMM. TAPS=9: 3x3 stride-1 pad-1 implicit GEMM —
// K = 9*Cin tap-major, each BK k-step lies inside ONE (ky,kx) tap plane
// (Cin % 64 == 0), the A row for output position m is the tap-shifted
// input row with zero predication at the image boundary; everything else
// (B staging from the wrapper-pre-permuted [Co][9Ci] weight, MFMA loop,
// stats/bias/scale/relu/residual epilogue, store transpose) is shared.
template <int BN, bool STATS, int BK = 64, int TAPS = 1, int RING = 0>
__global__ __launch_bounds__((BN == 256) ? 512 : 256) void conv1x1_nt_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    bf16* __restrict__ C, const bf16* __restrict__ residual,  // [M,N] | null
    const float* __restrict__ bias,                           // [N] | null
    const float* __restrict__ scale,                          // [N] | null
    const float* __restrict__ shift,                          // [N] | null
    float* __restrict__ sums,  // [gridDim.x, 2N] per-block partials | null
    int64_t M, int K, int N, bool relu, int imgH = 0, int imgW = 0) {
  constexpr int BM = 128;
  constexpr int NWAVES = (BN == 256) ? 8 : 4;
  constexpr int WAVES_M = (BN == 64) ? 4 : 2;
  constexpr int WAVES_N = NWAVES / WAVES_M;  // 4 / 2 / 1
  constexpr int WM = BM / WAVES_M;  // 64 or 32
  constexpr int WN = BN / WAVES_N;  // 64
  constexpr int MFR = WM / 16, NFR = WN / 16;
  constexpr int ROWB = BK * 2;  // LDS row bytes (128)

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  // grid (gy=n, gx=m) with n fastest: the N/BN blocks of one M-strip are
  // dispatched together, so they share the strip's A tile through L2/L3
  // instead of re-fetching it a full grid-pass later
  const int64_t m0 = (int64_t)blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;

  extern __shared__ __attribute__((aligned(16))) char lds_raw[];
  bf16* const lds16 = (bf16*)lds_raw;
  const int nk = K / BK;
  // direct-global A fragments measured SLOWER than LDS staging at nk==1
  // (latency-exposed loads, no glds prefetch): keep the staged path.
  const bool direct_a = false;
  // RING>0: R-slot glds ring, stage 2 ahead (guide "pipelining across
  // barriers": raw s_barrier + counted vmcnt leaves 2 stages in flight)
  const int nbuf = RING > 0 ? RING : (nk > 1 ? 2 : 1);
  auto a_lds = [&](int buf) { return lds16 + buf * (BM * BK); };
  auto b_lds = [&](int buf) {
    return (direct_a ? lds16 : lds16 + nbuf * BM * BK) + buf * (BN * BK);
  };

  // stage one K-step of A[BM][BK] and B[BN][BK] into buffer `buf`.
  // LDS image is row-major with the read-side XOR swizzle baked into the
  // SOURCE address (rule 21: linear dest + inverse-swizzled source).
  // A 1KB piece covers 1024/ROWB rows; swizzle granularity shrinks with the
  // row size (BK=64: byte ^= (row&7)<<4 over 16B slots; BK=32: (row&3)<<4).
  constexpr int RPP = 1024 / (BK * 2);       // rows per 1KB piece (8 or 16)
  constexpr int LPR = 64 / RPP;              // lanes per row (8 or 4)
  // BK=64 (128B rows): byte ^= (row&7)<<4 spreads a 16-lane column read
  // over 8 slots. BK=32 (64B rows): rows alias every 4, so fold row>>2 in:
  // ((row ^ row>>2)&3)<<4 gives 4 distinct slots per alias class.
  auto swz = [&](int row) {
    return BK == 64 ? (row & 7) << 4 : ((row ^ (row >> 2)) & 3) << 4;
  };
  const int CinK = K / TAPS;  // physical A row length (elements)
  auto stage = [&](int buf, int kt) {
    const int k0b_all = kt * BK * 2;
    // tap plane + byte offset within the tap's Cin slice
    const int tap = TAPS == 1 ? 0 : (kt * BK) / CinK;
    const int k0b = TAPS == 1 ? k0b_all : ((kt * BK) % CinK) * 2;
    const int ky = tap / 3 - 1, kx = tap % 3 - 1;  // -1..1 (TAPS==9)
    if (!direct_a) {
#pragma unroll
      for (int pp = 0; pp < (BM / RPP) / NWAVES; ++pp) {
        const int p = wave + pp * NWAVES;
        const int row = p * RPP + lane / LPR;
        const int b = (lane % LPR) * 16;
        if (TAPS == 1) {
          const int64_t rg = m0 + row < M ? m0 + row : M - 1;
          const char* src = (const char*)A + rg * (int64_t)CinK * 2 + k0b +
                            (b ^ swz(row));
          glds16((const bf16*)src, a_lds(buf) + p * 512);
        } else {
          // per-row tap shift + boundary predication: a 128-row strip
          // nearly always crosses a W boundary for kx != 0, so this path
          // stages by predicated 16B loads + LDS vector stores
          const int64_t m = m0 + row < M ? m0 + row : M - 1;
          const int64_t img = m / ((int64_t)imgH * imgW);
          const int r = (int)(m - img * imgH * imgW);
          const int oh = r / imgW, ow = r % imgW;
          const int ih = oh + ky, iw = ow + kx;
          const bool valid = (unsigned)ih < (unsigned)imgH &&
                             (unsigned)iw < (unsigned)imgW &&
                             m0 + row < M;
          Vec<bf16, 8> v;
          if (valid) {
            const int64_t in_row = (img * imgH + ih) * imgW + iw;
            v = *(const Vec<bf16, 8>*)((const char*)A +
                                       in_row * (int64_t)CinK * 2 + k0b +
                                       (b ^ swz(row)));
          } else {
#pragma unroll
            for (int j = 0; j < 8; ++j) v.v[j] = from_f32<bf16>(0.f);
          }
          vstore<bf16, 8>(a_lds(buf) + p * 512 + (lane % LPR) * 8 +
                              (lane / LPR) * (BK),
                          v);
        }
      }
    }
#pragma unroll
    for (int pp = 0; pp < (BN / RPP) / NWAVES; ++pp) {
      const int p = wave + pp * NWAVES;
      const int row = p * RPP + lane / LPR;
      const int b = (lane % LPR) * 16;
      const char* src = (const char*)B + (int64_t)(n0 + row) * K * 2 +
                        k0b_all + (b ^ swz(row));
      glds16((const bf16*)src, b_lds(buf) + p * 512);
    }
  };

  const int wm_off = (wave / WAVES_N) * WM;
  const int wn_off = (wave % WAVES_N) * WN;

  f32x4 acc[MFR][NFR];
#pragma unroll
  for (int mi = 0; mi < MFR; ++mi)
#pragma unroll
    for (int ni = 0; ni < NFR; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  auto compute_step = [&](const char* ab, const char* bb) {
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      const int kbyte = (ks * 32 + (lane >> 4) * 8) * 2;
      bf16x8 af[MFR], bfr[NFR];
#pragma unroll
      for (int mi = 0; mi < MFR; ++mi) {
        const int row = wm_off + mi * 16 + (lane & 15);
        if (direct_a) {
          const int64_t rg = m0 + row < M ? m0 + row : M - 1;
          af[mi] = *(const bf16x8*)((const char*)A + rg * (int64_t)K * 2 +
                                    kbyte);
        } else {
          af[mi] =
              *(const bf16x8*)(ab + row * ROWB + (kbyte ^ swz(row)));
        }
      }
#pragma unroll
      for (int ni = 0; ni < NFR; ++ni) {
        const int row = wn_off + ni * 16 + (lane & 15);
        bfr[ni] = *(const bf16x8*)(bb + row * ROWB + (kbyte ^ swz(row)));
      }
#pragma unroll
      for (int mi = 0; mi < MFR; ++mi)
#pragma unroll
        for (int ni = 0; ni < NFR; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  if constexpr (RING == 0) {
    stage(0, 0);
    __syncthreads();  // drains the glds (vmcnt0 inside the barrier)
    int cur = 0;
    for (int kt = 0; kt < nk; ++kt) {
      if (kt + 1 < nk) stage(cur ^ 1, kt + 1);
      compute_step((const char*)a_lds(cur), (const char*)b_lds(cur));
      __syncthreads();
      cur ^= 1;
    }
  } else {
    // 4-slot ring, stage 2 ahead. Slot s is overwritten by stage(s+RING)
    // issued 2 iterations after its readers' barrier (R >= D+2 bounds the
    // wave skew: issue at iteration j implies all computes <= j-2 done).
    // Counted vmcnt before a RAW barrier per the guide: __syncthreads()
    // would drain every in-flight glds to vmcnt(0).
    static_assert(RING == 4, "ring depth fixed at 4 (stage 2 ahead)");
    constexpr int G = ((BM / RPP) / NWAVES) + ((BN / RPP) / NWAVES);
    stage(0, 0);
    if (nk > 1) stage(1, 1);
    for (int kt = 0; kt < nk; ++kt) {
      if (kt + 2 < nk) {
        stage((kt + 2) % RING, kt + 2);
        asm volatile("s_waitcnt vmcnt(%0)" ::"n"(2 * G) : "memory");
      } else if (kt + 1 < nk) {
        asm volatile("s_waitcnt vmcnt(%0)" ::"n"(G) : "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      compute_step((const char*)a_lds(kt % RING),
                   (const char*)b_lds(kt % RING));
    }
    __syncthreads();  // all glds drained: bare barrier before LDS reuse
  }

  // ---- stats from registers (fp32-exact, before the bf16 rounding) -------
  // fragment layout: lane's channel for n-tile ni is col = ni*16 + (lane&15);
  // its 4*MFR values per ni are rows of that channel. Padded rows (gm >= M)
  // must not contribute.
  if (STATS) {
    // per-block partial sums -> slab row sums[m-block * 2N + ...], NO
    // global atomics (a 6k-deep atomic chain per channel measured 6x the
    // whole GEMM); the [2C] reduction happens in one torch sum downstream.
    float* sbuf = (float*)lds_raw;  // [WAVES_M][2*BN] (staging LDS is free)
    const int wm = wave / WAVES_N;
#pragma unroll
    for (int ni = 0; ni < NFR; ++ni) {
      const int c = wn_off + ni * 16 + (lane & 15);  // channel within block
      const float b = bias != nullptr ? bias[n0 + c] : 0.f;
      float s = 0.f, q = 0.f;
#pragma unroll
      for (int mi = 0; mi < MFR; ++mi)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int64_t gm = m0 + wm_off + mi * 16 + (lane >> 4) * 4 + r;
          if (gm < M) {
            const float raw = acc[mi][ni][r] + b;
            s += raw;
            q += raw * raw;
          }
        }
      // lanes l, l+16, l+32, l+48 hold the same channel (different rows)
      s += __shfl_xor(s, 16, 64);
      s += __shfl_xor(s, 32, 64);
      q += __shfl_xor(q, 16, 64);
      q += __shfl_xor(q, 32, 64);
      if (lane < 16) {
        sbuf[wm * (2 * BN) + c] = s;
        sbuf[wm * (2 * BN) + BN + c] = q;
      }
    }
    __syncthreads();
    float* slab = sums + (int64_t)blockIdx.y * (2 * N);  // m-block row
    for (int t = threadIdx.x; t < 2 * BN; t += NWAVES * 64) {
      float tot = 0.f;
#pragma unroll
      for (int wmi = 0; wmi < WAVES_M; ++wmi) tot += sbuf[wmi * (2 * BN) + t];
      const int c = t < BN ? t : t - BN;
      slab[(t < BN ? 0 : N) + n0 + c] = tot;
    }
    __syncthreads();  // sbuf is about to be reused by the store transpose
  }

  // ---- epilogue: LDS bf16 transpose -> row-major 16B stores --------------
  // chunked over m-halves so the scratch stays at half a wave-tile (keeps
  // the K=64 single-buffer path at 3 blocks/CU)
  constexpr int EROW = WN + 8;  // +8 bf16 pad keeps b128 reads 16B-aligned
  constexpr int EPC = (MFR >= 4) ? 2 : 1;   // epilogue chunks
  constexpr int MFC = MFR / EPC;            // m-frags per chunk
  bf16* ep = (bf16*)lds_raw + wave * (MFC * 16) * EROW;
  const int c0 = (lane & 7) * 8;           // channel chunk in wave tile
  const int ng0 = n0 + wn_off + c0;        // global channel of chunk start
  const bool has_ss = scale != nullptr;
  float sc[8], sh[8];
  if (has_ss) {
#pragma unroll
    for (int j = 0; j < 8; ++j) { sc[j] = scale[ng0 + j]; sh[j] = shift[ng0 + j]; }
  }
  float bcol[NFR];
#pragma unroll
  for (int ni = 0; ni < NFR; ++ni)
    bcol[ni] = bias != nullptr ? bias[n0 + wn_off + ni * 16 + (lane & 15)]
                               : 0.f;
#pragma unroll
  for (int h = 0; h < EPC; ++h) {
#pragma unroll
    for (int mc = 0; mc < MFC; ++mc) {
      const int mi = h * MFC + mc;
#pragma unroll
      for (int ni = 0; ni < NFR; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = mc * 16 + (lane >> 4) * 4 + r;
          const int col = ni * 16 + (lane & 15);
          ep[row * EROW + col] = from_f32<bf16>(acc[mi][ni][r] + bcol[ni]);
        }
    }
    __syncthreads();
#pragma unroll
    for (int i = 0; i < (MFC * 16) / 8; ++i) {
      const int r = (lane >> 3) + 8 * i;
      const int64_t gm = m0 + wm_off + h * (MFC * 16) + r;
      if (gm < M) {
        Vec<bf16, 8> v = vload<bf16, 8>(ep + r * EROW + c0);
        Vec<bf16, 8> res;
        if (residual != nullptr)
          res = vload<bf16, 8>(residual + gm * N + ng0);
        Vec<bf16, 8> out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float y = to_f32(v.v[j]);
          if (has_ss) y = y * sc[j] + sh[j];
          if (residual != nullptr) y += to_f32(res.v[j]);
          if (relu) y = fmaxf(y, 0.f);
          out.v[j] = from_f32<bf16>(y);
        }
        vstore<bf16, 8>(C + gm * N + ng0, out);
      }
    }
    if (h + 1 < EPC) __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// wgrad kernel: dW[N][K] += sum_m dy[m][n] * x[m][k] (fp32 atomics).
// Both operands staged TRANSPOSED in LDS ([ch][m+pad], scatter b16 writes
// with an m-XOR so the channel-group writes spread over banks), fragments
// then read m-contiguous b128. 64x64 output tile, 4 waves (32x32 each).
// ---------------------------------------------------------------------------
// TAPS=9: dW[co][tap*Ci+ci] = sum_m dy[m][co] * x[shift_tap(m)][ci]
// (3x3 s1 p1 wgrad) — the x tile is staged from tap-shifted rows with
// zero predication at the image boundary, everything else unchanged.
template <int BCO, int BCI, int TAPS = 1>
__global__ __launch_bounds__((BCO * BCI >= 8192) ? 512 : 256) void conv1x1_wgrad_kernel(
    const bf16* __restrict__ dy,  // [M,N]
    const bf16* __restrict__ x,   // [M,Cin] (K = TAPS*Cin)
    float* __restrict__ dW,       // [N,K] pre-zeroed
    int64_t M, int K, int N, int64_t chunk, int imgH = 0, int imgW = 0) {
  constexpr int KM = 64;   // m per step
  constexpr int TR = 72;   // LDS row length (elements) for [ch][KM] image
  constexpr int NW = (BCO * BCI >= 8192) ? 8 : 4;
  constexpr int WGC = (BCO >= 128 && BCI == 64) ? 4 : 2;  // wave grid over co
  constexpr int WGI = NW / WGC;        // wave grid over ci
  constexpr int WCO = BCO / WGC;       // 64 or 32
  constexpr int WCI = BCI / WGI;       // 32
  constexpr int FCO = WCO / 16, FCI = WCI / 16;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n0 = blockIdx.x * BCO;  // co tile
  const int k0 = blockIdx.y * BCI;  // ci tile
  const int64_t m_begin = (int64_t)blockIdx.z * chunk;
  const int64_t m_end = m_begin + chunk < M ? m_begin + chunk : M;

  extern __shared__ __attribute__((aligned(16))) char lds_raw[];
  bf16* const lds16 = (bf16*)lds_raw;
  auto dyt = [&](int buf) { return lds16 + buf * (BCO * TR); };
  auto xt = [&](int buf) { return lds16 + 2 * BCO * TR + buf * (BCI * TR); };

  // transposed scatter-stage of a [KM m][BCH ch] global chunk into
  // lds[ch][m ^ mswz(ch)] (mswz spreads the 8 channel-groups over banks).
  // tap >= 0 (TAPS==9): rows are tap-shifted with boundary zeros.
  auto stage_t = [&](bf16* lds, const bf16* g, int stride_elems, int ch0g,
                     int64_t mt, int tap, auto bch_tag) {
    constexpr int BCH = decltype(bch_tag)::value;
#pragma unroll
    for (int it = 0; it < (KM * BCH / 8) / (NW * 64); ++it) {
      const int idx = (int)threadIdx.x + it * (NW * 64);
      const int m = idx / (BCH / 8);            // 0..63
      const int ch = (idx % (BCH / 8)) * 8;
      Vec<bf16, 8> v;
      bool valid = mt + m < m_end;
      int64_t src_row = mt + m;
      if (TAPS != 1 && tap >= 0 && valid) {
        const int ky = tap / 3 - 1, kx = tap % 3 - 1;
        const int64_t img = src_row / ((int64_t)imgH * imgW);
        const int r = (int)(src_row - img * imgH * imgW);
        const int oh = r / imgW, ow = r % imgW;
        const int ih = oh + ky, iw = ow + kx;
        valid = (unsigned)ih < (unsigned)imgH &&
                (unsigned)iw < (unsigned)imgW;
        src_row = (img * imgH + ih) * imgW + iw;
      }
      if (valid) {
        v = vload<bf16, 8>(g + src_row * (int64_t)stride_elems + ch0g + ch);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) v.v[j] = from_f32<bf16>(0.f);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int c = ch + j;
        lds[c * TR + (m ^ (((c >> 3) & 7) << 3))] = v.v[j];
      }
    }
  };

  // TAPS==9: one block owns a ky ROW of taps (3 of them) for its
  // (co, ci) tile, so dy is staged ONCE per m-step instead of once per
  // tap (the all-tap-tiles grid re-read dy ~9x: ~1.8 GB/call at the
  // 56x56x64 shape). The kx taps loop inside over alternating x buffers.
  constexpr int KXN = (TAPS == 1) ? 1 : 3;
  f32x4 acc[KXN][FCO][FCI];
#pragma unroll
  for (int kx = 0; kx < KXN; ++kx)
#pragma unroll
    for (int mi = 0; mi < FCO; ++mi)
#pragma unroll
      for (int ni = 0; ni < FCI; ++ni)
        acc[kx][mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int wco = (wave / WGI) * WCO;
  const int wci = (wave % WGI) * WCI;

  const int CinK = K / TAPS;
  const int ky_idx = TAPS == 1 ? 0 : k0 / CinK;     // ky row of this block
  const int ci0 = TAPS == 1 ? k0 : k0 % CinK;       // offset within x row
  auto tap_of = [&](int kx) { return TAPS == 1 ? -1 : ky_idx * 3 + kx; };

  stage_t(dyt(0), dy, N, n0, m_begin, -1, std::integral_constant<int, BCO>{});
  stage_t(xt(0), x, CinK, ci0, m_begin, tap_of(0),
          std::integral_constant<int, BCI>{});
  __syncthreads();
  int xbuf = 0, dybuf = 0;
  for (int64_t mt = m_begin; mt < m_end; mt += KM) {
#pragma unroll
    for (int kx = 0; kx < KXN; ++kx) {
      // prefetch the next x tile (next tap, or next m-step's first tap)
      if (kx + 1 < KXN) {
        stage_t(xt(xbuf ^ 1), x, CinK, ci0, mt, tap_of(kx + 1),
                std::integral_constant<int, BCI>{});
      } else if (mt + KM < m_end) {
        stage_t(xt(xbuf ^ 1), x, CinK, ci0, mt + KM, tap_of(0),
                std::integral_constant<int, BCI>{});
        stage_t(dyt(dybuf ^ 1), dy, N, n0, mt + KM, -1,
                std::integral_constant<int, BCO>{});
      }
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int mm = ks * 32 + (lane >> 4) * 8;
        bf16x8 af[FCO], bfr[FCI];
#pragma unroll
        for (int mi = 0; mi < FCO; ++mi) {
          const int c = wco + mi * 16 + (lane & 15);
          af[mi] = *(const bf16x8*)(dyt(dybuf) + c * TR +
                                    (mm ^ (((c >> 3) & 7) << 3)));
        }
#pragma unroll
        for (int ni = 0; ni < FCI; ++ni) {
          const int c = wci + ni * 16 + (lane & 15);
          bfr[ni] = *(const bf16x8*)(xt(xbuf) + c * TR +
                                     (mm ^ (((c >> 3) & 7) << 3)));
        }
#pragma unroll
        for (int mi = 0; mi < FCO; ++mi)
#pragma unroll
          for (int ni = 0; ni < FCI; ++ni)
            acc[kx][mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[mi], bfr[ni], acc[kx][mi][ni], 0, 0, 0);
      }
      __syncthreads();
      xbuf ^= 1;
    }
    dybuf ^= 1;
  }

#pragma unroll
  for (int kx = 0; kx < KXN; ++kx)
#pragma unroll
    for (int mi = 0; mi < FCO; ++mi)
#pragma unroll
      for (int ni = 0; ni < FCI; ++ni)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int co = n0 + wco + mi * 16 + (lane >> 4) * 4 + r;
          const int ci_local = wci + ni * 16 + (lane & 15);
          const int kg = TAPS == 1
              ? k0 + ci_local
              : (ky_idx * 3 + kx) * CinK + ci0 + ci_local;
          atomicAdd(&dW[(int64_t)co * K + kg], acc[kx][mi][ni][r]);
        }
}

}  // namespace dla

// A [M,K] bf16, B [N,K] bf16 -> C = A @ B^T [M,N] bf16, with fused epilogue.
// Returns {C, sums[2N] fp32} (sums undefined unless want_stats).
std::vector<torch::Tensor> conv1x1_fwd(torch::Tensor a, torch::Tensor b,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> scale,
                                       c10::optional<torch::Tensor> shift,
                                       c10::optional<torch::Tensor> residual,
                                       bool relu, bool want_stats) {
  DLA_CHECK_INPUT(a);
  DLA_CHECK_INPUT(b);
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  b.scalar_type() == torch::kBFloat16,
              "conv1x1_fwd: bf16 only");
  const int64_t M = a.size(0);
  const int K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK((int)b.size(1) == K, "conv1x1_fwd: K mismatch");
  TORCH_CHECK(K % 64 == 0 && N % 64 == 0, "conv1x1_fwd: K,N must be %64");
  auto C = torch::empty({M, N}, a.options());
  auto opts_f = a.options().dtype(torch::kFloat);
  const int gx = (int)((M + 127) / 128);
  // per-block stats partials (fully written by the kernel; reduced to [2N]
  // by the caller with one torch sum — no global atomics)
  auto sums = want_stats ? torch::empty({gx, 2 * N}, opts_f)
                         : torch::empty({0}, opts_f);
  const dla::bf16* res_p =
      residual.has_value() ? (const dla::bf16*)residual->data_ptr() : nullptr;
  const float* bias_p = bias.has_value() ? bias->data_ptr<float>() : nullptr;
  const float* scale_p = scale.has_value() ? scale->data_ptr<float>() : nullptr;
  const float* shift_p = shift.has_value() ? shift->data_ptr<float>() : nullptr;
  float* sums_p = want_stats ? sums.data_ptr<float>() : nullptr;

  const int nk = K / 64;
  auto launch = [&](auto bntag, auto stag, auto bktag) {
    constexpr int BN = decltype(bntag)::value;
    constexpr bool ST = decltype(stag)::value;
    constexpr int BKT = decltype(bktag)::value;
    constexpr int NW = (BN == 256) ? 8 : 4;
    constexpr int WM = (BN == 64) ? 32 : 64;
    // single staging buffer suffices when there is only one K-step; with one
    // K-step A skips LDS entirely (direct-global fragments)
    const int nbuf = K / BKT > 1 ? 2 : 1;
    const int lds_stage = (128 * BKT + BN * BKT) * 2 * nbuf;
    // chunked transpose scratch: half a wave-tile when MFR >= 4
    const int lds_ep = NW * (WM >= 64 ? WM / 2 : WM) * (64 + 8) * 2;
    const int lds_st = ST ? ((BN == 64) ? 4 : 2) * 2 * BN * 4 : 0;
    const int lds = std::max(std::max(lds_stage, lds_ep), lds_st);
    if (lds > 65536) {
      static bool done[2] = {false, false};
      if (!done[ST]) {
        (void)hipFuncSetAttribute(
            (const void*)&dla::conv1x1_nt_kernel<BN, ST, BKT>,
            hipFuncAttributeMaxDynamicSharedMemorySize, 163840);
        done[ST] = true;
      }
    }
    hipLaunchKernelGGL((dla::conv1x1_nt_kernel<BN, ST, BKT>), dim3(N / BN, gx),
                       dim3(NW * 64), lds, dla::stream(),
                       (const dla::bf16*)a.data_ptr(),
                       (const dla::bf16*)b.data_ptr(),
                       (dla::bf16*)C.data_ptr(), res_p, bias_p, scale_p,
                       shift_p, sums_p, M, K, N, relu);
  };
  // BK=32 halves the staging LDS (4 blocks/CU): measured +5% at K=128
  // (nk==2, the exposed-drain case), neutral/negative deeper — default it
  // only there (DLA_C1X1_BK32=1 forces it for all BN=128 shapes).
  static const bool bk32_force = []{
    const char* e = std::getenv("DLA_C1X1_BK32");
    return e != nullptr && e[0] == '1';
  }();
  // 4-slot glds ring (BK=32, 64KB LDS, 2 blocks/CU) — opt-in pending A/B
  // on the wait-bound fat-N shapes (ROADMAP item 2; guide "3-buf span +83%")
  static const bool ring_on = []{
    const char* e = std::getenv("DLA_C1X1_RING");
    return e != nullptr && e[0] == '1';
  }();
  const bool bk32 = bk32_force || nk == 2;
  auto launch_ring = [&](auto stag) {
    constexpr bool ST = decltype(stag)::value;
    constexpr int lds = 4 * (128 * 32 + 128 * 32) * 2;  // 64 KB
    hipLaunchKernelGGL((dla::conv1x1_nt_kernel<128, ST, 32, 1, 4>),
                       dim3(N / 128, gx), dim3(256), lds, dla::stream(),
                       (const dla::bf16*)a.data_ptr(),
                       (const dla::bf16*)b.data_ptr(),
                       (dla::bf16*)C.data_ptr(), res_p, bias_p, scale_p,
                       shift_p, sums_p, M, K, N, relu);
  };
  auto pick = [&](auto bntag) {
    constexpr int BNv = decltype(bntag)::value;
    if (BNv == 128 && ring_on && K % 32 == 0 && K / 32 >= 4) {
      if (want_stats) launch_ring(std::true_type{});
      else launch_ring(std::false_type{});
      return;
    }
    if (BNv == 128 && bk32 && nk >= 2) {
      if (want_stats) launch(bntag, std::true_type{},
                             std::integral_constant<int, 32>{});
      else launch(bntag, std::false_type{}, std::integral_constant<int, 32>{});
      return;
    }
    if (want_stats) launch(bntag, std::true_type{},
                           std::integral_constant<int, 64>{});
    else launch(bntag, std::false_type{}, std::integral_constant<int, 64>{});
  };
  // BN=256 (512 threads, single A pass) wins only at nk==1 where the LDS
  // fits 3 blocks/CU; at nk>=2 its 96KB double-buffer drops occupancy to
  // 1 block/CU and BN=128 (2 blocks/CU) measures faster.
  if (N % 256 == 0 && nk == 1) pick(std::integral_constant<int, 256>{});
  else if (N % 128 == 0) pick(std::integral_constant<int, 128>{});
  else pick(std::integral_constant<int, 64>{});
  HIP_CHECK_ERR();
  return {C, sums};
}

// 3x3 stride-1 pad-1 conv as tap-major implicit GEMM (TAPS=9 variant of
// the kernel above). a: x flattened NHWC [M=B*H*W, Cin]; w9: weight
// pre-permuted to [Co, 9*Cin] (tap-major: permute(0,2,3,1) of the torch
// [Co,Ci,3,3] layout). Returns {y [M,Co], stats partials}.
std::vector<torch::Tensor> conv3x3_fwd(torch::Tensor a, torch::Tensor w9,
                                       int64_t imgH, int64_t imgW,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> scale,
                                       c10::optional<torch::Tensor> shift,
                                       c10::optional<torch::Tensor> residual,
                                       bool relu, bool want_stats) {
  DLA_CHECK_INPUT(a);
  DLA_CHECK_INPUT(w9);
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
                  w9.scalar_type() == torch::kBFloat16,
              "conv3x3_fwd: bf16 only");
  const int64_t M = a.size(0);
  const int Cin = (int)a.size(1), N = (int)w9.size(0);
  const int K = 9 * Cin;
  TORCH_CHECK((int)w9.size(1) == K, "conv3x3_fwd: weight must be [Co, 9*Ci]");
  TORCH_CHECK(Cin % 64 == 0 && N % 64 == 0, "conv3x3_fwd: C must be %64");
  TORCH_CHECK(M % (imgH * imgW) == 0, "conv3x3_fwd: M != B*H*W");
  auto C = torch::empty({M, N}, a.options());
  auto opts_f = a.options().dtype(torch::kFloat);
  const int gx = (int)((M + 127) / 128);
  auto sums = want_stats ? torch::empty({gx, 2 * N}, opts_f)
                         : torch::empty({0}, opts_f);
  const dla::bf16* res_p =
      residual.has_value() ? (const dla::bf16*)residual->data_ptr() : nullptr;
  const float* bias_p = bias.has_value() ? bias->data_ptr<float>() : nullptr;
  const float* scale_p = scale.has_value() ? scale->data_ptr<float>() : nullptr;
  const float* shift_p = shift.has_value() ? shift->data_ptr<float>() : nullptr;
  float* sums_p = want_stats ? sums.data_ptr<float>() : nullptr;
  auto launch = [&](auto bntag, auto stag) {
    constexpr int BN = decltype(bntag)::value;
    constexpr bool ST = decltype(stag)::value;
    const int lds_stage = (128 * 64 + BN * 64) * 2 * 2;
    const int lds_ep = 4 * ((BN == 64) ? 32 : 64) / ((BN == 64) ? 1 : 2) *
                       (64 + 8) * 2;
    const int lds_st = ST ? ((BN == 64) ? 4 : 2) * 2 * BN * 4 : 0;
    const int lds = std::max(std::max(lds_stage, lds_ep), lds_st);
    hipLaunchKernelGGL((dla::conv1x1_nt_kernel<BN, ST, 64, 9>),
                       dim3(N / BN, gx), dim3(256), lds, dla::stream(),
                       (const dla::bf16*)a.data_ptr(),
                       (const dla::bf16*)w9.data_ptr(),
                       (dla::bf16*)C.data_ptr(), res_p, bias_p, scale_p,
                       shift_p, sums_p, M, K, N, relu, (int)imgH, (int)imgW);
  };
  auto pick = [&](auto bntag) {
    if (want_stats) launch(bntag, std::true_type{});
    else launch(bntag, std::false_type{});
  };
  if (N % 128 == 0) pick(std::integral_constant<int, 128>{});
  else pick(std::integral_constant<int, 64>{});
  HIP_CHECK_ERR();
  return {C, sums};
}

// dW[N,K] fp32 = dy[M,N]^T @ x[M,K]
torch::Tensor conv1x1_wgrad(torch::Tensor dy, torch::Tensor x) {
  DLA_CHECK_INPUT(dy);
  DLA_CHECK_INPUT(x);
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16 &&
                  x.scalar_type() == torch::kBFloat16,
              "conv1x1_wgrad: bf16 only");
  const int64_t M = x.size(0);
  const int K = (int)x.size(1), N = (int)dy.size(1);
  TORCH_CHECK(dy.size(0) == M, "conv1x1_wgrad: M mismatch");
  TORCH_CHECK(K % 64 == 0 && N % 64 == 0, "conv1x1_wgrad: K,N must be %64");
  auto dW = torch::zeros({N, K}, x.options().dtype(torch::kFloat));
  auto launch = [&](auto cotag, auto citag) {
    constexpr int BCO = decltype(cotag)::value;
    constexpr int BCI = decltype(citag)::value;
    const int tiles = (N / BCO) * (K / BCI);
    int splits = (int)std::min<int64_t>(
        std::max<int64_t>(1, 1024 / tiles),
        (M + 2047) / 2048);
    const int64_t chunk0 = (M + splits - 1) / splits;
    const int64_t chunk = ((chunk0 + 63) / 64) * 64;  // multiple of KM
    splits = (int)((M + chunk - 1) / chunk);
    const int lds = 2 * (BCO + BCI) * 72 * 2;
    if (lds > 65536) {
      static bool done = false;
      if (!done) {
        (void)hipFuncSetAttribute(
            (const void*)&dla::conv1x1_wgrad_kernel<BCO, BCI>,
            hipFuncAttributeMaxDynamicSharedMemorySize, 163840);
        done = true;
      }
    }
    hipLaunchKernelGGL((dla::conv1x1_wgrad_kernel<BCO, BCI>),
                       dim3(N / BCO, K / BCI, splits),
                       dim3(BCO * BCI >= 8192 ? 512 : 256), lds,
                       dla::stream(), (const dla::bf16*)dy.data_ptr(),
                       (const dla::bf16*)x.data_ptr(), dW.data_ptr<float>(),
                       M, K, N, chunk);
  };
  // pick the instantiated tile minimizing re-read traffic; (256,64) measured
  // slower than (128,64) (1 block/CU, staging-bound) so 128 caps BCO
  const bool n128 = N % 128 == 0, k128 = K % 128 == 0;
  if (n128 && k128)
    launch(std::integral_constant<int, 128>{}, std::integral_constant<int, 128>{});
  else if (k128)
    launch(std::integral_constant<int, 64>{}, std::integral_constant<int, 128>{});
  else if (n128)
    launch(std::integral_constant<int, 128>{}, std::integral_constant<int, 64>{});
  else
    launch(std::integral_constant<int, 64>{}, std::integral_constant<int, 64>{});
  HIP_CHECK_ERR();
  return dW;
}

// dW9 [Co, 9*Ci] fp32 = 3x3 s1 p1 wgrad (tap-major k, same order the
// conv3x3_fwd weight uses)
torch::Tensor conv3x3_wgrad(torch::Tensor dy, torch::Tensor x, int64_t imgH,
                            int64_t imgW) {
  DLA_CHECK_INPUT(dy);
  DLA_CHECK_INPUT(x);
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16 &&
                  x.scalar_type() == torch::kBFloat16,
              "conv3x3_wgrad: bf16 only");
  const int64_t M = x.size(0);
  const int Cin = (int)x.size(1), N = (int)dy.size(1);
  const int K = 9 * Cin;
  TORCH_CHECK(dy.size(0) == M, "conv3x3_wgrad: M mismatch");
  TORCH_CHECK(Cin % 64 == 0 && N % 64 == 0, "conv3x3_wgrad: C must be %64");
  auto dW = torch::zeros({N, K}, x.options().dtype(torch::kFloat));
  auto launch = [&](auto cotag, auto citag) {
    constexpr int BCO = decltype(cotag)::value;
    constexpr int BCI = decltype(citag)::value;
    const int tiles = (N / BCO) * (3 * Cin / BCI);  // ky-row-merged grid
    int splits = (int)std::min<int64_t>(
        std::max<int64_t>(1, 1024 / tiles),
        (M + 2047) / 2048);
    const int64_t chunk0 = (M + splits - 1) / splits;
    const int64_t chunk = ((chunk0 + 63) / 64) * 64;
    splits = (int)((M + chunk - 1) / chunk);
    const int lds = 2 * (BCO + BCI) * 72 * 2;
    if (lds > 65536) {
      static bool done = false;
      if (!done) {
        (void)hipFuncSetAttribute(
            (const void*)&dla::conv1x1_wgrad_kernel<BCO, BCI, 9>,
            hipFuncAttributeMaxDynamicSharedMemorySize, 163840);
        done = true;
      }
    }
    hipLaunchKernelGGL((dla::conv1x1_wgrad_kernel<BCO, BCI, 9>),
                       dim3(N / BCO, 3 * Cin / BCI, splits),
                       dim3(BCO * BCI >= 8192 ? 512 : 256), lds,
                       dla::stream(), (const dla::bf16*)dy.data_ptr(),
                       (const dla::bf16*)x.data_ptr(), dW.data_ptr<float>(),
                       M, K, N, chunk, (int)imgH, (int)imgW);
  };
  // BCI must divide Cin so a ci tile stays inside one tap plane
  const bool n128 = N % 128 == 0, ci128 = Cin % 128 == 0;
  if (n128 && ci128)
    launch(std::integral_constant<int, 128>{}, std::integral_constant<int, 128>{});
  else if (ci128)
    launch(std::integral_constant<int, 64>{}, std::integral_constant<int, 128>{});
  else if (n128)
    launch(std::integral_constant<int, 128>{}, std::integral_constant<int, 64>{});
  else
    launch(std::integral_constant<int, 64>{}, std::integral_constant<int, 64>{});
  HIP_CHECK_ERR();
  return dW;
}
