// MFMA GEMM kernels for the C=3 stem convs — the 7x7/stride-2 ImageNet
// stem (BASELINE config 4's first layer) and the 3x3/stride-1 CIFAR stem
// (ResNet-18/CIFAR's first layer; reference conv surface
// /root/reference/cifar_example.py:20-29 generalized to ResNet).
//
// The dot2 stem kernels (conv.hip) are VALU/LDS-bound at ~30-64 TF/s;
// these route the same im2col GEMM through the matrix cores. The im2col
// k-axis uses a ROW-PADDED layout: kg = r*SROW + (s*3 + c), rows padded
// 3R -> SROW (a multiple of 8) so every 8-element k-group stays inside
// one filter row (a group is then 8 CONTIGUOUS elements of one input
// image row, loadable as one 16-byte read away from the edges), total
// padded K = SKG (R=7: 24-rows -> 192 = 3 BK=64 chunks, 147/192 used;
// R=3: 16-rows -> 64 = 1 chunk, 27/64 used). The weight is re-laid
// [KO, SKG] on the host per call (<= 27 KB).
//
// Load discipline (v2): every k-group issues ONE unconditional 16-byte
// read from a selected address (real row or a zero page) and fixes up
// masks/edges afterwards — v1's per-group branches serialized the gather
// into dependent round trips (same lesson as conv_gather_gemm's
// "address select, not a branch"). The fwd kernel stages per-64-chunk
// (27.6 KB LDS, next chunk's loads ride under the MFMA phase) instead of
// v1's full-K 77 KB tile that capped occupancy at 2 blocks/CU.
//
//   fwd  : y[m, ko]   = sum_kg A_im2col[m, kg] * w24[ko, kg]
//   wgrad: dw[ko, kg] = sum_m dy[m, ko] * A_im2col[m, kg]
//          (transposed LDS staging, split-M chunk slabs, wgrad_reduce)
#include "common.h"

typedef __attribute__((ext_vector_type(8))) _Float16 half8_s;
typedef __attribute__((ext_vector_type(16))) float f32x16_s;

namespace {

// per-filter-size geometry: R=7 (ImageNet, stride 2) and R=3 (CIFAR,
// stride 1). SROW = 3R padded to 8; SKG = R*SROW padded to 64.
template <int R_>
struct StemGeo {};
template <>
struct StemGeo<7> {
  static constexpr int SROW = 24, SKG = 192, VALID = 21;
};
template <>
struct StemGeo<3> {
  static constexpr int SROW = 16, SKG = 64, VALID = 9;
};

constexpr int SLDK = 64 + 8;
constexpr int SLDM = 72;   // wgrad m-minor row length

template <typename T16>
struct SMfma {};
template <>
struct SMfma<__hip_bfloat16> {
  static DEV_INLINE f32x16_s run(short8 a, short8 b, f32x16_s c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
};
template <>
struct SMfma<__half> {
  static DEV_INLINE f32x16_s run(short8 a, short8 b, f32x16_s c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16((half8_s)a, (half8_s)b, c,
                                                  0, 0, 0);
  }
};

// per-group gather state: phase 1 picks an always-loadable address,
// phase 2 issues the (batched) loads, phase 3 fixes up pads/edges
template <typename T16>
struct G8 {
  const T16* src;
  long base;
  int e0, jj;
  unsigned char fast, row_ok;
};

template <typename T16, int R_, int STRIDE_>
DEV_INLINE void g8_plan(G8<T16>& g, const T16* __restrict__ x,
                        const T16* __restrict__ zpage, int n, int p, int q,
                        int kg0, int H, int W3, int pad, bool m_ok) {
  constexpr int SROW = StemGeo<R_>::SROW;
  const int r = kg0 / SROW;
  g.jj = kg0 - r * SROW;
  const int ih = STRIDE_ * p - pad + r;
  g.e0 = (STRIDE_ * q - pad) * 3 + g.jj;
  g.row_ok = m_ok && r < R_ && (unsigned)ih < (unsigned)H;
  g.fast = g.row_ok && g.e0 >= 0 && g.e0 + 8 <= W3;
  g.base = ((long)n * H + ih) * (long)W3;
  g.src = g.fast ? x + g.base + g.e0 : zpage;
}

template <typename T16, int R_>
DEV_INLINE short8 g8_fix(const G8<T16>& g, short8 v,
                         const T16* __restrict__ x, int W3) {
  constexpr int VALID = StemGeo<R_>::VALID;
  if (g.fast) {
    if (g.jj > VALID - 8) {
#pragma unroll
      for (int u = 0; u < 8; ++u)
        if (g.jj + u >= VALID) v[u] = 0;
    }
    return v;
  }
  short8 w = {};
  if (g.row_ok) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int e = g.e0 + u;
      if (g.jj + u < VALID && (unsigned)e < (unsigned)W3)
        w[u] = *reinterpret_cast<const short*>(x + g.base + e);
    }
  }
  return w;
}

// ---- forward ----------------------------------------------------------
// conv_gather_gemm anatomy: 128m x 64ko block, 3 k-chunks of 64, next
// chunk's gather rides under the MFMA phase, single-buffered LDS.
template <typename T16, int R_, int STRIDE_>
__global__ __launch_bounds__(256, 2) void conv_fwd_stem_gemm(
    const T16* __restrict__ x,    // [N, H, W, 3]
    const T16* __restrict__ w24,  // [KO, SKG] row-padded
    const float* __restrict__ bias, const T16* __restrict__ zpage,
    T16* __restrict__ y, const int N, const int H, const int W,
    const int KO, const int Ho, const int Wo, const int pad, const int act,
    const int has_bias) {
  constexpr int SKG = StemGeo<R_>::SKG;
  __shared__ T16 lds[128 * SLDK + 64 * SLDK];
  const int tid = threadIdx.x;
  const long Mtot = (long)N * Ho * Wo;
  const long bm0 = (long)blockIdx.x * 128;
  const int k0 = blockIdx.y * 64;
  const int W3 = 3 * W;

  // A staging: 2 threads per m-row, 32 k-elements (4 groups) each
  const int sa_m = tid >> 1;
  const int sa_c = (tid & 1) * 32;
  const long m_a = bm0 + sa_m;
  const bool m_ok = m_a < Mtot;
  int n_ = 0, p_ = 0, q_ = 0;
  if (m_ok) {
    n_ = (int)(m_a / ((long)Ho * Wo));
    const int pq = (int)(m_a % ((long)Ho * Wo));
    p_ = pq / Wo;
    q_ = pq % Wo;
  }
  // B staging: 4 threads per ko-row, 16 k-elements each
  const int sb_n = tid >> 2;
  const int sb_c = (tid & 3) * 16;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int wm = wave * 32;
  f32x16_s acc[2] = {};

  short8 sa[4], sb[2];
  auto load_step = [&](int j) {
    G8<T16> g[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      g8_plan<T16, R_, STRIDE_>(g[i], x, zpage, n_, p_, q_,
                                j * 64 + sa_c + 8 * i, H, W3, pad, m_ok);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      sa[i] = *reinterpret_cast<const short8*>(g[i].src);
#pragma unroll
    for (int i = 0; i < 4; ++i) sa[i] = g8_fix<T16, R_>(g[i], sa[i], x, W3);
    const T16* wp = w24 + (long)(k0 + sb_n) * SKG + j * 64 + sb_c;
    sb[0] = *reinterpret_cast<const short8*>(wp);
    sb[1] = *reinterpret_cast<const short8*>(wp + 8);
  };
  auto stage = [&]() {
    short* pa = reinterpret_cast<short*>(lds + sa_m * SLDK + sa_c);
#pragma unroll
    for (int i = 0; i < 4; ++i) *reinterpret_cast<short8*>(pa + 8 * i) = sa[i];
    short* pb =
        reinterpret_cast<short*>(lds + 128 * SLDK + sb_n * SLDK + sb_c);
    *reinterpret_cast<short8*>(pb) = sb[0];
    *reinterpret_cast<short8*>(pb + 8) = sb[1];
  };

  load_step(0);
  for (int j = 0; j < SKG / 64; ++j) {
    __syncthreads();
    stage();
    __syncthreads();
    if (j + 1 < SKG / 64) load_step(j + 1);
    const T16* ldsA = lds;
    const T16* ldsB = lds + 128 * SLDK;
#pragma unroll
    for (int kk = 0; kk < 64; kk += 16) {
      const short8 af = *reinterpret_cast<const short8*>(
          ldsA + (wm + li) * SLDK + kk + kh * 8);
      const short8 b0 = *reinterpret_cast<const short8*>(
          ldsB + li * SLDK + kk + kh * 8);
      const short8 b1 = *reinterpret_cast<const short8*>(
          ldsB + (32 + li) * SLDK + kk + kh * 8);
      acc[0] = SMfma<T16>::run(af, b0, acc[0]);
      acc[1] = SMfma<T16>::run(af, b1, acc[1]);
    }
  }

  float bv[2];
  bv[0] = has_bias ? bias[k0 + li] : 0.f;
  bv[1] = has_bias ? bias[k0 + 32 + li] : 0.f;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    const long m_out = bm0 + wm + row;
    if (m_out < Mtot) {
#pragma unroll
      for (int t2 = 0; t2 < 2; ++t2) {
        float v = acc[t2][reg] + bv[t2];
        if (act == 1) v = fmaxf(v, 0.f);
        y[m_out * KO + k0 + t2 * 32 + li] = F16<T16>::from_f32(v);
      }
    }
  }
}

// ---- forward v3: strip-staged -----------------------------------------
// One block = 128 consecutive outputs of ONE image (requires Ho*Wo % 128
// == 0 so blocks never straddle images). The nrows input rows the block's
// receptive fields span are staged ONCE into an LDS strip (coalesced row
// copies with zero pads); the MFMA A-fragments then read 8-element tap
// spans DIRECTLY from the strip (strip idx 3*STRIDE*q + jj + u —
// contiguous, 4B-aligned) instead of v2's per-m scattered 16-byte global
// gathers (~45 KB of uncoalesced reads per block for the 7x7 stem).
// Weights [64, SKG] live in LDS.
template <typename T16, int R_, int STRIDE_>
__global__ __launch_bounds__(256, 2) void conv_fwd_stem_strip(
    const T16* __restrict__ x,    // [N, H, W, 3]
    const T16* __restrict__ w24,  // [KO, SKG] row-padded
    const float* __restrict__ bias,
    float* __restrict__ stats_slab,  // null, or per-block (sum,sumsq) rows
    T16* __restrict__ y, const int N,
    const int H, const int W, const int KO, const int Ho, const int Wo,
    const int pad, const int act, const int has_bias, const int nrows,
    const int sstride) {
  constexpr int SROW = StemGeo<R_>::SROW;
  constexpr int SKG = StemGeo<R_>::SKG;
  constexpr int VALID = StemGeo<R_>::VALID;
  extern __shared__ __attribute__((aligned(16))) char ssmem[];
  T16* strip = reinterpret_cast<T16*>(ssmem);  // [nrows+1][sstride]
  T16* ldsB = strip + (long)(nrows + 1) * sstride;  // [64][200]
  const int tid = threadIdx.x;
  const int W3 = 3 * W;
  const long Mimg = (long)Ho * Wo;
  const long bm0 = (long)blockIdx.x * 128;
  const int n = (int)(bm0 / Mimg);
  const long bmi = bm0 - (long)n * Mimg;
  const int p0 = (int)(bmi / Wo);
  const int ih0 = STRIDE_ * p0 - pad;  // strip row 0 = input row ih0
  const int k0 = blockIdx.y * 64;

  // ---- strip fill: rows ih0 .. ih0+nrows-1, each 3*pad zeros | row |
  // zeros; row nrows is the all-zero stub for r >= R_ k-pad reads ----
  const int off3 = 3 * pad;
  const int nv8 = (W3 + 7) / 8;
  for (int t = tid; t < nrows * nv8; t += 256) {
    const int rr = t / nv8;
    const int j8 = (t - rr * nv8) * 8;
    const int ih = ih0 + rr;
    T16* dst = strip + (long)rr * sstride + off3 + j8;
    if ((unsigned)ih < (unsigned)H && j8 + 8 <= W3) {
      *reinterpret_cast<short8*>(dst) =
          *reinterpret_cast<const short8*>(x + ((long)n * H + ih) * W3 + j8);
    } else {
      const T16 z{};
#pragma unroll
      for (int u = 0; u < 8; ++u)
        dst[u] = (j8 + u < W3 && (unsigned)ih < (unsigned)H)
                     ? x[((long)n * H + ih) * W3 + j8 + u]
                     : z;
    }
  }
  {  // pad columns of every row + the zero stub row
    const int nz = sstride - W3;
    const T16 z{};
    for (int t = tid; t < nrows * nz; t += 256) {
      const int rr = t / nz;
      const int e = t - rr * nz;
      strip[(long)rr * sstride + (e < off3 ? e : W3 + e)] = z;
    }
    for (int t = tid; t < sstride; t += 256)
      strip[(long)nrows * sstride + t] = z;
  }
  // ---- B fill: [64][SKG] at row stride SKG+8 ----
  for (int e = tid * 8; e < 64 * SKG; e += 256 * 8) {
    const int row = e / SKG, col = e - row * SKG;
    *reinterpret_cast<short8*>(
        reinterpret_cast<short*>(ldsB + row * (SKG + 8) + col)) =
        *reinterpret_cast<const short8*>(w24 + (long)(k0 + row) * SKG + col);
  }
  __syncthreads();

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int wm = wave * 32;
  // per-lane output coords (fixed): m = bm0 + wm + li
  const long mi = bmi + wm + li;
  const int p_lane = (int)(mi / Wo);
  const int q_lane = (int)(mi - (long)p_lane * Wo);
  const int rr_base = STRIDE_ * (p_lane - p0);  // strip row of tap r=0
  const int ebase = 3 * STRIDE_ * q_lane;  // strip elem of (jj=0) tap col
  f32x16_s acc[2] = {};

#pragma unroll
  for (int slice = 0; slice < SKG / 16; ++slice) {
    const int kg0 = slice * 16 + kh * 8;  // wave-uniform per kh
    const int r = kg0 / SROW;
    const int jj = kg0 - r * SROW;
    const int rr = r < R_ ? rr_base + r : nrows;  // stub row for k-pad
    const short* ap = reinterpret_cast<const short*>(
        strip + (long)rr * sstride + ebase + jj);
    short8 af;
#pragma unroll
    for (int u2 = 0; u2 < 4; ++u2) {
      int v32;
      __builtin_memcpy(&v32, ap + 2 * u2, 4);
      __builtin_memcpy(reinterpret_cast<char*>(&af) + 4 * u2, &v32, 4);
    }
    // zero the row-pad lanes (jj+u >= VALID) — weights there are zero
    // too, but the strip span can run past the zero pads at the row end
    if (jj > VALID - 8) {
#pragma unroll
      for (int u = 0; u < 8; ++u)
        if (jj + u >= VALID) af[u] = 0;
    }
    const short8 b0 = *reinterpret_cast<const short8*>(
        ldsB + li * (SKG + 8) + slice * 16 + kh * 8);
    const short8 b1 = *reinterpret_cast<const short8*>(
        ldsB + (32 + li) * (SKG + 8) + slice * 16 + kh * 8);
    acc[0] = SMfma<T16>::run(af, b0, acc[0]);
    acc[1] = SMfma<T16>::run(af, b1, acc[1]);
  }

  float bv[2];
  bv[0] = has_bias ? bias[k0 + li] : 0.f;
  bv[1] = has_bias ? bias[k0 + 32 + li] : 0.f;
  float ssum[2] = {}, ssq[2] = {};
  const long Mtot = (long)N * Mimg;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    const long m_out = bm0 + wm + row;
    if (m_out < Mtot) {
#pragma unroll
      for (int t2 = 0; t2 < 2; ++t2) {
        float v = acc[t2][reg] + bv[t2];
        if (act == 1) v = fmaxf(v, 0.f);
        if (stats_slab) {
          ssum[t2] += v;
          ssq[t2] += v * v;
        }
        y[m_out * KO + k0 + t2 * 32 + li] = F16<T16>::from_f32(v);
      }
    }
  }
  if (stats_slab) {
    // same fold as conv_gather_gemm: kh halves by xor-shuffle, waves by
    // per-wave LDS rows (strip LDS reused after a barrier)
#pragma unroll
    for (int t2 = 0; t2 < 2; ++t2) {
      ssum[t2] += __shfl_xor(ssum[t2], 32, 64);
      ssq[t2] += __shfl_xor(ssq[t2], 32, 64);
    }
    float* lsum = reinterpret_cast<float*>(ssmem);  // [4 waves][128]
    __syncthreads();
    if (kh == 0) {
#pragma unroll
      for (int t2 = 0; t2 < 2; ++t2) {
        lsum[wave * 128 + t2 * 32 + li] = ssum[t2];
        lsum[wave * 128 + 64 + t2 * 32 + li] = ssq[t2];
      }
    }
    __syncthreads();
    float* slab =
        stats_slab + ((long)blockIdx.y * gridDim.x + blockIdx.x) * 2 * 64;
    for (int t = tid; t < 2 * 64; t += 256)
      slab[t] = lsum[t] + lsum[128 + t] + lsum[256 + t] + lsum[384 + t];
  }
}

// ---- wgrad ------------------------------------------------------------
// dw slab scatter target layout: [KO, 3, R, R] (parameter layout), one
// slab per m-chunk; grid = (KO/64, m-chunks). ALL SKG kg in one block:
// dy is staged once for NJ x the MFMA work (the kg-chunked version staged
// the same dy per chunk and its 4-MFMA inner loop couldn't cover the
// scattered x-gather latency).
template <typename T16, int R_, int STRIDE_>
__global__ __launch_bounds__(256, 2) void conv_wgrad_stem_gemm(
    const T16* __restrict__ x,   // [N, H, W, 3]
    const T16* __restrict__ dy,  // [M, KO]
    const T16* __restrict__ zpage,
    float* __restrict__ dw,      // chunk slabs of [KO*3*R_*R_]
    const int N, const int H, const int W, const int KO, const int Ho,
    const int Wo, const int pad, const long m_per_chunk) {
  constexpr int SROW = StemGeo<R_>::SROW;
  constexpr int SKG = StemGeo<R_>::SKG;
  constexpr int VALID = StemGeo<R_>::VALID;
  constexpr int NJ = SKG / 64;  // kg 64-chunks = MFMA j-tiles per wave
  __shared__ T16 lds[(64 + SKG) * SLDM];
  const int tid = threadIdx.x;
  const long Mtot = (long)N * Ho * Wo;
  const int k0 = blockIdx.x * 64;
  const long m_begin = (long)blockIdx.y * m_per_chunk;
  const long m_end = min(Mtot, m_begin + m_per_chunk);
  const int W3 = 3 * W;

  const bool do_dy = tid < 128;
  const bool do_x = tid >= 128;
  const int t = do_dy ? tid : 0;
  const int sm = (t & 15) * 4;
  const int sk = (t >> 4) * 8;
  const int tx = tid & 127;
  const int smx = (tx & 15) * 4;
  const int gx0 = (tx >> 4) * NJ;  // this thread's NJ kg-groups

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int i0 = (wave & 1) * 32;       // KO sub-tile
  const int j0 = (wave >> 1) * 32;      // kg sub-tile base (stride 64)

  f32x16_s acc[NJ] = {};

  int dn = 0, dp = 0, dq = 0;
  {
    const long m_first = m_begin + smx;
    dn = (int)(m_first / ((long)Ho * Wo));
    const int pq = (int)(m_first % ((long)Ho * Wo));
    dp = pq / Wo;
    dq = pq % Wo;
  }
  auto advance = [&](int by) {
    dq += by;
    while (dq >= Wo) {
      dq -= Wo;
      if (++dp == Ho) {
        dp = 0;
        ++dn;
      }
    }
  };

  short8 vdy[4], vx[NJ][4];
  auto load_m = [&](long m0) {
    if (do_dy) {
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const long m = m0 + sm + mi;
        vdy[mi] = (m < m_end) ? *reinterpret_cast<const short8*>(
                                    dy + m * KO + k0 + sk)
                              : short8{};
      }
    }
    if (do_x) {
#pragma unroll
      for (int ii = 0; ii < NJ; ++ii) {
        G8<T16> g[4];
        int n_ = dn, p_ = dp, q_ = dq;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          const long m = m0 + smx + mi;
          g8_plan<T16, R_, STRIDE_>(g[mi], x, zpage, n_, p_, q_,
                                    (gx0 + ii) * 8, H, W3, pad, m < m_end);
          if (mi < 3 && ++q_ == Wo) {
            q_ = 0;
            if (++p_ == Ho) {
              p_ = 0;
              ++n_;
            }
          }
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          vx[ii][mi] = *reinterpret_cast<const short8*>(g[mi].src);
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          vx[ii][mi] = g8_fix<T16, R_>(g[mi], vx[ii][mi], x, W3);
      }
      advance(64);
    }
  };
  auto stage_m = [&]() {
    if (do_dy) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        short4v pk = {vdy[0][e], vdy[1][e], vdy[2][e], vdy[3][e]};
        *reinterpret_cast<short4v*>(
            reinterpret_cast<short*>(lds + (sk + e) * SLDM + sm)) = pk;
      }
    }
    if (do_x) {
      T16* ldsT = lds + 64 * SLDM;
#pragma unroll
      for (int ii = 0; ii < NJ; ++ii) {
        const int krow = (gx0 + ii) * 8;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          short4v pk = {vx[ii][0][e], vx[ii][1][e], vx[ii][2][e],
                        vx[ii][3][e]};
          *reinterpret_cast<short4v*>(
              reinterpret_cast<short*>(ldsT + (krow + e) * SLDM + smx)) = pk;
        }
      }
    }
  };

  load_m(m_begin);
  for (long m0 = m_begin; m0 < m_end; m0 += 64) {
    __syncthreads();
    stage_m();
    __syncthreads();
    if (m0 + 64 < m_end) load_m(m0 + 64);
    const T16* ldsDyT = lds;
    const T16* ldsXT = lds + 64 * SLDM;
#pragma unroll
    for (int kk = 0; kk < 64; kk += 16) {
      const short8 af = *reinterpret_cast<const short8*>(
          ldsDyT + (i0 + li) * SLDM + kk + kh * 8);
#pragma unroll
      for (int jj = 0; jj < NJ; ++jj) {
        const short8 bf = *reinterpret_cast<const short8*>(
            ldsXT + (j0 + jj * 64 + li) * SLDM + kk + kh * 8);
        acc[jj] = SMfma<T16>::run(af, bf, acc[jj]);
      }
    }
  }

  // scatter the 32(ko) x NJx32(kg) fp32 tiles into the [KO,3,R,R] slab;
  // row-pad positions (jj >= VALID) and r >= R_ are dropped
  float* slab = dw + (long)blockIdx.y * ((long)KO * 3 * R_ * R_);
#pragma unroll
  for (int jt = 0; jt < NJ; ++jt) {
    const int kg = j0 + jt * 64 + li;
    const int r = kg / SROW;
    const int jj = kg - r * SROW;
    if (jj < VALID && r < R_) {
      const int s = jj / 3, c = jj - s * 3;
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int i = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
        slab[(((long)(k0 + i0 + i) * 3 + c) * R_ + r) * R_ + s] =
            acc[jt][reg];
      }
    }
  }
}

}  // namespace

// host launchers --------------------------------------------------------

at::Tensor conv_zero_page(const at::Tensor& like);  // conv_mfma.hip

// w: [KO, KGP] tap-major padded ([kg = (r*R+s)*3+c]) -> [KO, SKG]
// row-padded (kg = r*SROW + s*3 + c)
template <int R_>
static at::Tensor stem_wpad(const at::Tensor& w) {
  constexpr int SROW = StemGeo<R_>::SROW;
  constexpr int SKG = StemGeo<R_>::SKG;
  const long KO = w.size(0);
  auto valid = w.narrow(1, 0, 3 * R_ * R_).reshape({KO, R_, 3 * R_});
  auto padded = at::constant_pad_nd(valid, {0, SROW - 3 * R_}, 0);
  return at::constant_pad_nd(padded.reshape({KO, R_ * SROW}),
                             {0, SKG - R_ * SROW}, 0)
      .contiguous();
}

void stats_slab_reduce(at::Tensor slab, at::Tensor stats, int gx, int bnt2,
                       int C);  // conv_mfma.hip

template <int R_, int STRIDE_>
static bool fwd_stem_gemm_impl(at::Tensor x, at::Tensor w, at::Tensor bias,
                               at::Tensor y, long pad, long act,
                               at::Tensor stats) {
  constexpr int SKG = StemGeo<R_>::SKG;
  const int N = x.size(0), H = x.size(1), W = x.size(2);
  const int KO = w.size(0);
  const int Ho = y.size(1), Wo = y.size(2);
  TORCH_CHECK(KO % 64 == 0, "stem GEMM expects KO % 64 == 0");
  auto w24 = stem_wpad<R_>(w);
  const long M = (long)N * Ho * Wo;
  dim3 grid((unsigned)cdiv_l(M, 128), KO / 64);
  const int has_bias = bias.numel() > 0;
  // strip variant (v3) when blocks can't straddle images: the A operand
  // comes from a once-staged LDS strip instead of per-m global gathers
  static const bool strip_on = [] {
    const char* e = getenv("MI355X_STEM_STRIP");
    return !e || e[0] != '0';
  }();
  const long Mimg = (long)Ho * Wo;
  if (strip_on && Mimg % 128 == 0) {
    // rows spanned: 128 outputs cover <= ceil(127/Wo)+1 p-rows; input
    // rows STRIDE*p0-pad .. STRIDE*p_last+R-1-pad
    const int pspan = (int)((127 / Wo) + 1);
    const int nrows = STRIDE_ * (pspan - 1) + R_;
    // row length: left pads (3*pad) + payload (3W) + right slack for the
    // widest 8-element tap read (max strip idx <= 3W + 6*pad + 6)
    const int sstride = (3 * W + 6 * (int)pad + 8 + 3) & ~3;
    const size_t smem =
        ((size_t)(nrows + 1) * sstride + 64 * (SKG + 8)) * x.element_size();
    if (smem <= 150 * 1024) {
      const bool want_stats = stats.defined() && stats.numel() > 0;
      at::Tensor slab;
      float* slab_p = nullptr;
      if (want_stats) {
        slab = at::empty({(long)grid.y * grid.x * 128},
                         x.options().dtype(at::kFloat));
        slab_p = slab.data_ptr<float>();
      }
      DISPATCH_16(x, T16, {
        hipLaunchKernelGGL((conv_fwd_stem_strip<T16, R_, STRIDE_>), grid,
                           dim3(256), smem, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)w24.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           slab_p, (T16*)y.data_ptr(), N, H, W, KO, Ho, Wo,
                           (int)pad, (int)act, has_bias, nrows, sstride);
      });
      if (slab_p)
        stats_slab_reduce(slab, stats, grid.x, 128, KO);
      return want_stats;
    }
  }
  at::Tensor zp = conv_zero_page(x);
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL((conv_fwd_stem_gemm<T16, R_, STRIDE_>), grid,
                       dim3(256), 0, cur_stream(), (const T16*)x.data_ptr(),
                       (const T16*)w24.data_ptr(),
                       has_bias ? bias.data_ptr<float>() : nullptr,
                       (const T16*)zp.data_ptr(), (T16*)y.data_ptr(), N, H,
                       W, KO, Ho, Wo, (int)pad, (int)act, has_bias);
  });
  return false;
}

// returns true when `stats` ([2,KO], zeroed) was filled from the conv
// epilogue (strip path only; the v2 fallback leaves it to the caller)
bool conv_fwd_stem_gemm_launch(at::Tensor x, at::Tensor w, at::Tensor bias,
                               at::Tensor y, long pad, long act,
                               at::Tensor stats, long R, long stride) {
  if (R == 7 && stride == 2)
    return fwd_stem_gemm_impl<7, 2>(x, w, bias, y, pad, act, stats);
  TORCH_CHECK(R == 3 && stride == 1, "stem GEMM: unsupported (R, stride)");
  return fwd_stem_gemm_impl<3, 1>(x, w, bias, y, pad, act, stats);
}

void wgrad_reduce_launch(at::Tensor part, at::Tensor dw, long E, long nz);

template <int R_, int STRIDE_>
static void wgrad_stem_gemm_impl(at::Tensor x, at::Tensor dy, at::Tensor dw,
                                 long pad) {
  const int N = x.size(0), H = x.size(1), W = x.size(2);
  const int Ho = dy.size(1), Wo = dy.size(2), KO = dy.size(3);
  TORCH_CHECK(KO % 64 == 0, "stem wgrad GEMM expects KO % 64 == 0");
  const long M = (long)N * Ho * Wo;
  const long E = (long)KO * 3 * R_ * R_;
  long nchunks = std::min<long>(512, cdiv_l(M, 4096));
  const long m_per_chunk = cdiv_l(M, std::max<long>(nchunks, 1));
  nchunks = cdiv_l(M, m_per_chunk);
  auto part = at::empty({nchunks * E}, x.options().dtype(at::kFloat));
  at::Tensor zp = conv_zero_page(x);
  dim3 grid(KO / 64, (unsigned)nchunks);
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL((conv_wgrad_stem_gemm<T16, R_, STRIDE_>), grid,
                       dim3(256), 0, cur_stream(), (const T16*)x.data_ptr(),
                       (const T16*)dy.data_ptr(), (const T16*)zp.data_ptr(),
                       part.data_ptr<float>(), N, H, W, KO, Ho, Wo, (int)pad,
                       m_per_chunk);
  });
  wgrad_reduce_launch(part, dw, E, nchunks);
}

// dw [KO,3,R,R] fp32
void conv_wgrad_stem_gemm_launch(at::Tensor x, at::Tensor dy, at::Tensor dw,
                                 long pad, long R, long stride) {
  if (R == 7 && stride == 2)
    return wgrad_stem_gemm_impl<7, 2>(x, dy, dw, pad);
  TORCH_CHECK(R == 3 && stride == 1, "stem wgrad GEMM: unsupported (R, stride)");
  wgrad_stem_gemm_impl<3, 1>(x, dy, dw, pad);
}
