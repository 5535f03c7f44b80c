// MFMA implicit-GEMM conv for CDNA4 (gfx950) — the hot path for every
// ResNet conv (C%64==0, K%64==0). One gather-GEMM kernel serves forward
// (TRANS=false) and dgrad (TRANS=true; host pre-flips the weight to
// [R,S,C,K] so dgrad is the same GEMM with transposed-window gather).
//
// GEMM view: OUT[M=N*Ho*Wo, KO] = A[M, R*S*CI] * B,
// iterated as (r,s,c-chunk) k-slices of BK=64. A k-slice of A is a
// contiguous 64-channel read of the input window row (NHWC), staged
// through registers into LDS with zero-fill for padding/out-of-window
// rows (predication the glds path can't do). Structure follows the
// canonical CDNA GEMM anatomy (cdna_hip_programming.md §5): 128x64 block
// tile, 4 waves x (32M x 64N) each as 2 x mfma_f32_32x32x16, LDS tiles
// [row][BK+8] padded against bank conflicts (G4), T14-style staging
// (write after barrier, next global load issued under the MFMA phase).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) _Float16 half8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

namespace {

template <typename T16>
struct Mfma32 {};
template <>
struct Mfma32<__hip_bfloat16> {
  static DEV_INLINE f32x16 run(short8 a, short8 b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
};
template <>
struct Mfma32<__half> {
  static DEV_INLINE f32x16 run(short8 a, short8 b, f32x16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16((half8)a, (half8)b, c, 0, 0,
                                                  0);
  }
};

// lazy-BN A-side transform: z = relu(x*sc + sh) applied to an 8-channel
// group in registers at load/stage time (the normalized activation is
// never materialized; the VALU rides under the MFMA phase). Scale/shift
// come in as four float4 REGISTERS — v1 indexed asc[c0+u] per element
// (64 scalar L1 loads per thread per k-step) and measured -10% end to
// end from load-issue bloat alone.
struct Sc8 {
  float4v s0, s1, h0, h1;
};

DEV_INLINE Sc8 sc8_load(const float* __restrict__ asc,
                        const float* __restrict__ ash, int c) {
  Sc8 r;
  r.s0 = *reinterpret_cast<const float4v*>(asc + c);
  r.s1 = *reinterpret_cast<const float4v*>(asc + c + 4);
  r.h0 = *reinterpret_cast<const float4v*>(ash + c);
  r.h1 = *reinterpret_cast<const float4v*>(ash + c + 4);
  return r;
}

template <typename T16>
DEV_INLINE short8 scale8(short8 v, const Sc8& sc) {
  short8 o;
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    const float f = s16_to_f32<T16>(v[u]) * sc.s0[u] + sc.h0[u];
    o[u] = f32_to_s16<T16>(fmaxf(f, 0.f));
  }
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    const float f = s16_to_f32<T16>(v[4 + u]) * sc.s1[u] + sc.h1[u];
    o[4 + u] = f32_to_s16<T16>(fmaxf(f, 0.f));
  }
  return o;
}

constexpr int BM = 128, BN = 64, BK = 64;
constexpr int LDK = BK + 8;  // +16B pad: spreads fragment reads over banks
// NT = 32-wide N(output-channel) tiles per block: 2 (BN=64) or 4 (BN=128);
// wider tiles double the MFMA work per barrier pair for the K>=128 layers

// TRANS=false (fwd): src row = ho*stride - pad + r, in [0,Hi)
// TRANS=true (dgrad): src row = (ho + pad - r), valid iff %stride==0, /stride in [0,Hi)
// GENC (fwd only): CI % 64 != 0 — the weight is pre-padded to
// [KO, KGP=ceil(R*S*CI/64)*64] with kg=(r*S+s)*CI+c and zeros beyond, and
// the A side gathers per-ELEMENT across tap boundaries (stem convs C=3/6).
// ADDIN (compile-time — a runtime addin branch in this shared epilogue
// measurably slowed the non-addin convs): out += addin elementwise, the
// residual-junction gradient fused into the dgrad epilogue.
template <typename T16, bool TRANS, bool GENC, int NT, bool ADDIN = false,
          bool SCALED = false>
// min-blocks hint: same effect as on the wgrad kernel (see comment there)
__global__ __launch_bounds__(256, 2) void conv_gather_gemm(
    const T16* __restrict__ in,    // [N, Hi, Wi, CI]
    const T16* __restrict__ wgt,   // fwd: [KO, R*S*CI]; dgrad: [R*S, CI... ] via strides
    const float* __restrict__ bias,  // [KO] or null
    const T16* __restrict__ zpage,   // >=256 zero elements (OOB gather target)
    const T16* __restrict__ addin,   // ADDIN only: same layout as out
    const float* __restrict__ asc,   // SCALED only: [CI] scale
    const float* __restrict__ ash,   // SCALED only: [CI] shift
    T16* __restrict__ out,         // [N*Ho*Wo, KO]
    float* __restrict__ stats_slab,  // null, or [gy][gx][2][BNT] partial
    const int N, const int Hi, const int Wi, const int CI, const int KO,  //  (sum,sumsq) of this block's output tile (conv->BN fusion)
    const int Ho, const int Wo, const int R, const int S, const int stride,
    const int pad, const long b_row_stride, const long b_rs_stride,
    const int act, const int has_bias) {
  // single-buffered: 2x LDS for a double buffer measurably LOSES here —
  // it halves blocks/CU and the cross-block overlap it sacrifices was
  // already hiding the staging latency (guide common-mistake #5)
  __shared__ T16 lds[BM * LDK + NT * 32 * LDK];

  const int tid = threadIdx.x;
  const long Mtot = (long)N * Ho * Wo;
  const long bm0 = (long)blockIdx.x * BM;
  constexpr int BNT = NT * 32;
  const int k0 = blockIdx.y * BNT;

  // ---- A staging coords (2 threads per m-row, 32 channels each) ----
  const int sa_m = tid >> 1;
  const int sa_c = (tid & 1) * (BK / 2);
  const long m_a = bm0 + sa_m;
  const bool m_ok = m_a < Mtot;
  int n_ = 0, p_ = 0, q_ = 0;
  if (m_ok) {
    n_ = (int)(m_a / ((long)Ho * Wo));
    const int pq = (int)(m_a % ((long)Ho * Wo));
    p_ = pq / Wo;
    q_ = pq % Wo;
  }
  // fwd: top-left of the receptive field; dgrad: p_,q_ used directly
  const int ih0 = TRANS ? p_ : p_ * stride - pad;
  const int iw0 = TRANS ? q_ : q_ * stride - pad;

  // ---- B staging coords (256/BNT threads per n-row) ----
  constexpr int TPR = 256 / BNT;          // threads per B row
  constexpr int EPT = BK / TPR;           // elements per thread
  const int sb_n = tid / TPR;
  const int sb_c = (tid % TPR) * EPT;
  const T16* wrow = wgt + (long)(k0 + sb_n) * b_row_stride + sb_c;

  // ---- wave/lane fragment coords ----
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;  // k-half: lane holds k = kk + kh*8 + e
  const int wm = wave * 32;  // wave's m-offset inside the block tile

  f32x16 acc[NT] = {};

  const int cchunks = GENC ? 1 : CI / BK;
  const int ksteps = GENC ? (int)(b_row_stride / BK)  // KGP/64
                          : R * S * cchunks;

  short8 sa[4];            // 32 channels = 4 x 16B
  short8 sb[EPT / 8];      // B elements per thread

  auto load_step_genc = [&](int j) {
    // per-element gather: kg walks (tap, c) with carries (fully unrolled,
    // so the vector-register indices stay compile-time — guide rule 20)
    const int kg0 = j * BK + sa_c;
    int tap = kg0 / CI;
    int c = kg0 - tap * CI;
    int r_ = tap / S;
    int s_ = tap - r_ * S;
#pragma unroll
    for (int i = 0; i < 32; ++i) {
      const int ih = ih0 + r_;
      const int iw = iw0 + s_;
      const bool ok = m_ok && tap < R * S && (unsigned)ih < (unsigned)Hi &&
                      (unsigned)iw < (unsigned)Wi;
      // address select, not a branch: keeps all 32 loads unconditional so
      // they batch under one wait instead of 32 serialized round trips
      const T16* src =
          ok ? in + (((long)n_ * Hi + ih) * Wi + iw) * CI + c : zpage + i;
      short val;
      const T16 v = *src;
      __builtin_memcpy(&val, &v, 2);
      sa[i / 8][i % 8] = val;
      if (++c == CI) {
        c = 0;
        ++tap;
        if (++s_ == S) {
          s_ = 0;
          ++r_;
        }
      }
    }
    const T16* wp = wgt + (long)(k0 + sb_n) * b_row_stride + j * BK + sb_c;
#pragma unroll
    for (int i = 0; i < EPT / 8; ++i)
      sb[i] = *reinterpret_cast<const short8*>(wp + 8 * i);
  };

  // incremental (c-chunk, s, r) decode — load_step is called exactly once
  // per j in order, so three runtime divisions per k-step reduce to carry
  // counters (same lever as the wgrad kernels' m-decode). The per-tap
  // position (validity + image offset, incl. the TRANS stride div/mod
  // pair) is CACHED and recomputed only when the tap advances — for a
  // 1x1 dgrad the whole k-loop walks c-chunks of one tap.
  int cc_i = 0, s_i = 0, r_i = 0;
  bool va_tap = m_ok;
  long ioff_tap = 0;
  auto load_step = [&](int j) {
    if constexpr (GENC) {
      load_step_genc(j);
      return;
    }
    (void)j;
    const int c0 = cc_i * BK;
    const int s_ = s_i;
    const int r_ = r_i;
    if (cc_i == 0) {  // first chunk of this tap: refresh the position
      bool va = m_ok;
      long ioff = 0;
      if constexpr (TRANS) {
        const int ph = ih0 + pad - r_;
        const int qw = iw0 + pad - s_;
        const int pp = ph / stride, qq = qw / stride;
        va = va && ph >= 0 && qw >= 0 && (ph % stride) == 0 &&
             (qw % stride) == 0 && pp < Hi && qq < Wi;
        if (va) ioff = (((long)n_ * Hi + pp) * Wi + qq) * CI;
      } else {
        const int ih = ih0 + r_;
        const int iw = iw0 + s_;
        va = va && (unsigned)ih < (unsigned)Hi && (unsigned)iw < (unsigned)Wi;
        if (va) ioff = (((long)n_ * Hi + ih) * Wi + iw) * CI;
      }
      va_tap = va;
      ioff_tap = ioff;
    }
    if (++cc_i == cchunks) {
      cc_i = 0;
      if (++s_i == S) {
        s_i = 0;
        ++r_i;
      }
    }
    const bool va = va_tap;
    const long ioff = ioff_tap;
    const T16* xp = va ? in + ioff + c0 + sa_c : zpage;
#pragma unroll
    for (int i = 0; i < 4; ++i)
      sa[i] = *reinterpret_cast<const short8*>(xp + (va ? 8 * i : 0));
    if constexpr (SCALED) {
      if (va) {
#pragma unroll
        for (int i = 0; i < 4; ++i)
          sa[i] = scale8<T16>(sa[i],
                              sc8_load(asc, ash, c0 + sa_c + 8 * i));
      }
    }
    const T16* wp = wrow + (long)(r_ * S + s_) * b_rs_stride + c0;
#pragma unroll
    for (int i = 0; i < EPT / 8; ++i)
      sb[i] = *reinterpret_cast<const short8*>(wp + 8 * i);
  };

  auto stage = [&]() {
    T16* ldsA = lds;
    T16* ldsB = lds + BM * LDK;
    short* pa = reinterpret_cast<short*>(ldsA + sa_m * LDK + sa_c);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      *reinterpret_cast<short8*>(pa + 8 * i) = sa[i];
    short* pb = reinterpret_cast<short*>(ldsB + sb_n * LDK + sb_c);
#pragma unroll
    for (int i = 0; i < EPT / 8; ++i)
      *reinterpret_cast<short8*>(pb + 8 * i) = sb[i];
  };

  load_step(0);
  for (int j = 0; j < ksteps; ++j) {
    __syncthreads();  // previous MFMA phase done reading LDS
    stage();
    __syncthreads();
    if (j + 1 < ksteps) load_step(j + 1);  // overlaps the MFMA phase
    const T16* ldsA = lds;
    const T16* ldsB = lds + BM * LDK;
#pragma unroll
    for (int kk = 0; kk < BK; kk += 16) {
      const short8 af = *reinterpret_cast<const short8*>(
          ldsA + (wm + li) * LDK + kk + kh * 8);
#pragma unroll
      for (int tnt = 0; tnt < NT; ++tnt) {
        const short8 bf = *reinterpret_cast<const short8*>(
            ldsB + (tnt * 32 + li) * LDK + kk + kh * 8);
        acc[tnt] = Mfma32<T16>::run(af, bf, acc[tnt]);
      }
    }
  }

  // ---- epilogue: bias + act + store (+ optional BN-stats partials) ----
  float bv[NT];
  float ssum[NT] = {}, ssq[NT] = {};
#pragma unroll
  for (int tnt = 0; tnt < NT; ++tnt)
    bv[tnt] = has_bias ? bias[k0 + tnt * 32 + li] : 0.f;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    const long m_out = bm0 + wm + row;
    if (m_out < Mtot) {
#pragma unroll
      for (int tnt = 0; tnt < NT; ++tnt) {
        float v = acc[tnt][reg] + bv[tnt];
        if constexpr (ADDIN)
          v += F16<T16>::to_f32(addin[m_out * KO + k0 + tnt * 32 + li]);
        if (act == 1) v = fmaxf(v, 0.f);
        if (stats_slab) {
          ssum[tnt] += v;
          ssq[tnt] += v * v;
        }
        out[m_out * KO + k0 + tnt * 32 + li] = F16<T16>::from_f32(v);
      }
    }
  }
  if (stats_slab) {
    // conv->BN fusion: fold the block's per-lane channel partials (8
    // lanes per channel: 4 waves x 2 kh halves) — kh halves via one
    // cross-lane xor-shuffle, waves via per-wave LDS rows summed by the
    // storing threads. (LDS float atomicAdd here measured ~3.7 us/block
    // of serialized conflicts — 50% on the whole patch kernel.)
    constexpr int BNTC = NT * 32;
#pragma unroll
    for (int tnt = 0; tnt < NT; ++tnt) {
      ssum[tnt] += __shfl_xor(ssum[tnt], 32, 64);
      ssq[tnt] += __shfl_xor(ssq[tnt], 32, 64);
    }
    float* lsum = reinterpret_cast<float*>(lds);  // [4 waves][2*BNTC]
    __syncthreads();
    if (kh == 0) {
#pragma unroll
      for (int tnt = 0; tnt < NT; ++tnt) {
        lsum[wave * 2 * BNTC + tnt * 32 + li] = ssum[tnt];
        lsum[wave * 2 * BNTC + BNTC + tnt * 32 + li] = ssq[tnt];
      }
    }
    __syncthreads();
    float* slab =
        stats_slab + ((long)blockIdx.y * gridDim.x + blockIdx.x) * 2 * BNTC;
    for (int t = tid; t < 2 * BNTC; t += 256)
      slab[t] = lsum[t] + lsum[2 * BNTC + t] + lsum[4 * BNTC + t] +
                lsum[6 * BNTC + t];
  }
}

// L1 of the two-level slab fold: plain partial sums over a gx-slice
// (large-M convs have 10k-65k slab rows; a single 64-block reduce over
// them measured 139 us latency-bound — this level keeps the chip full)
__global__ void conv_stats_partial(const float* __restrict__ slab,
                                   float* __restrict__ out, int gx, int bnt2,
                                   int bpb) {
  const int by = blockIdx.y;
  const int t = threadIdx.x;
  if (t >= bnt2) return;
  const int b0 = blockIdx.x * bpb;
  const int b1 = min(gx, b0 + bpb);
  float acc = 0.f;
  for (int bx = b0; bx < b1; ++bx)
    acc += slab[((long)by * gx + bx) * bnt2 + t];
  out[((long)by * gridDim.x + blockIdx.x) * bnt2 + t] = acc;
}

// stats[2][C] = sum over gx of slab[gy][gx][2][BNT]; one block per
// (by, gx-slice), coalesced loads, few global atomics
__global__ void conv_stats_reduce(const float* __restrict__ slab,
                                  float* __restrict__ stats, int gx, int bnt2,
                                  int C, int bx_per_block) {
  const int by = blockIdx.y;
  const int t = threadIdx.x;
  if (t >= bnt2) return;
  float acc = 0.f;
  const int b0 = blockIdx.x * bx_per_block;
  const int b1 = min(gx, b0 + bx_per_block);
  for (int bx = b0; bx < b1; ++bx)
    acc += slab[((long)by * gx + bx) * bnt2 + t];
  const int bnt = bnt2 / 2;
  const int h = t / bnt;
  const int c = by * bnt + (t % bnt);
  if (gridDim.x == 1)
    stats[h * C + c] = acc;
  else
    atomicAdd(stats + h * C + c, acc);
}

// stride-2 dgrad, parity-specialized: dx positions partition by
// (ih%2, iw%2) and each class only has contributions from taps of
// matching parity ((ih+pad-r) % 2 == 0), so the k-loop runs
// |Ra|*|Sa|*cchunks steps instead of R*S*cchunks. The plain TRANS gather
// wastes ~75% of its staging+MFMA on parity-invalid (all-zero) slices
// for 3x3/s2, and ALL of it for 3 of the 4 classes of a 1x1/s2
// downsample (those classes here just store zeros). ADDIN as in
// conv_gather_gemm (residual-tap fusion).
template <typename T16, int NT, bool ADDIN = false>
__global__ __launch_bounds__(256, 2) void conv_dgrad_s2_gemm(
    const T16* __restrict__ dy,     // [N, P, Q, KO]
    const T16* __restrict__ wflip,  // [R, S, CI, KO]
    const T16* __restrict__ zpage,
    const T16* __restrict__ addin,  // ADDIN only: dx-shaped
    T16* __restrict__ dx,           // [N, H, W, CI]
    const int N, const int P, const int Q, const int KO, const int CI,
    const int H, const int W, const int R, const int S, const int pad) {
  __shared__ T16 lds[BM * LDK + NT * 32 * LDK];
  const int tid = threadIdx.x;
  const int pa = blockIdx.z >> 1;  // ih % 2
  const int pb = blockIdx.z & 1;   // iw % 2
  const int Ha = (H - pa + 1) >> 1;
  const int Wa = (W - pb + 1) >> 1;
  const long Mc = (long)N * Ha * Wa;
  const long bm0 = (long)blockIdx.x * BM;
  constexpr int BNT = NT * 32;
  const int k0 = blockIdx.y * BNT;  // dx channel tile

  const int r0 = (pa + pad) & 1;  // r parity for this class
  const int s0 = (pb + pad) & 1;
  const int nr = r0 < R ? ((R - r0 + 1) >> 1) : 0;
  const int ns = s0 < S ? ((S - s0 + 1) >> 1) : 0;
  const int cchunks = KO / BK;
  const int ksteps = nr * ns * cchunks;

  const int sa_m = tid >> 1;
  const int sa_c = (tid & 1) * (BK / 2);
  const long m_a = bm0 + sa_m;
  const bool m_ok = m_a < Mc;
  int n_ = 0, ih_ = 0, iw_ = 0;
  if (m_ok) {
    n_ = (int)(m_a / ((long)Ha * Wa));
    const int rem = (int)(m_a % ((long)Ha * Wa));
    ih_ = 2 * (rem / Wa) + pa;
    iw_ = 2 * (rem % Wa) + pb;
  }

  constexpr int TPR = 256 / BNT;
  constexpr int EPT = BK / TPR;
  const int sb_n = tid / TPR;
  const int sb_c = (tid % TPR) * EPT;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int wm = wave * 32;
  f32x16 acc[NT] = {};
  short8 sa[4], sb[EPT / 8];

  // incremental (c-chunk, s, r) decode: runtime divisions per k-step
  // measured as 38% issue-stall (PMC) in this short-k kernel
  int cc_i = 0, s_i = 0, r_i = 0;
  bool va_tap = false;
  long ioff_tap = 0, woff_tap = 0;
  auto load_step = [&](int j) {
    (void)j;
    const int c0 = cc_i * BK;
    if (cc_i == 0) {  // first chunk of this tap: refresh the position
      const int s_ = s0 + 2 * s_i;
      const int r_ = r0 + 2 * r_i;
      const int dh = ih_ + pad - r_;  // even by construction
      const int dw_ = iw_ + pad - s_;
      const int pp = dh >> 1, qq = dw_ >> 1;
      va_tap = m_ok && dh >= 0 && dw_ >= 0 && pp < P && qq < Q;
      ioff_tap = (((long)n_ * P + pp) * Q + qq) * KO;
      woff_tap = ((long)r_ * S + s_) * CI * (long)KO;
    }
    if (++cc_i == cchunks) {
      cc_i = 0;
      if (++s_i == ns) {
        s_i = 0;
        ++r_i;
      }
    }
    const bool va = va_tap;
    const T16* xp = va ? dy + ioff_tap + c0 + sa_c : zpage;
#pragma unroll
    for (int i = 0; i < 4; ++i)
      sa[i] = *reinterpret_cast<const short8*>(xp + (va ? 8 * i : 0));
    const T16* wp = wflip + woff_tap + (long)(k0 + sb_n) * KO + c0 + sb_c;
#pragma unroll
    for (int i = 0; i < EPT / 8; ++i)
      sb[i] = *reinterpret_cast<const short8*>(wp + 8 * i);
  };
  auto stage = [&]() {
    T16* ldsA = lds;
    T16* ldsB = lds + BM * LDK;
    short* pa_ = reinterpret_cast<short*>(ldsA + sa_m * LDK + sa_c);
#pragma unroll
    for (int i = 0; i < 4; ++i)
      *reinterpret_cast<short8*>(pa_ + 8 * i) = sa[i];
    short* pb_ = reinterpret_cast<short*>(ldsB + sb_n * LDK + sb_c);
#pragma unroll
    for (int i = 0; i < EPT / 8; ++i)
      *reinterpret_cast<short8*>(pb_ + 8 * i) = sb[i];
  };

  if (ksteps > 0) {
    load_step(0);
    for (int j = 0; j < ksteps; ++j) {
      __syncthreads();
      stage();
      __syncthreads();
      if (j + 1 < ksteps) load_step(j + 1);
      const T16* ldsA = lds;
      const T16* ldsB = lds + BM * LDK;
#pragma unroll
      for (int kk = 0; kk < BK; kk += 16) {
        const short8 af = *reinterpret_cast<const short8*>(
            ldsA + (wm + li) * LDK + kk + kh * 8);
#pragma unroll
        for (int tnt = 0; tnt < NT; ++tnt) {
          const short8 bf = *reinterpret_cast<const short8*>(
              ldsB + (tnt * 32 + li) * LDK + kk + kh * 8);
          acc[tnt] = Mfma32<T16>::run(af, bf, acc[tnt]);
        }
      }
    }
  }

  // epilogue: scatter into the strided class positions of dx. One
  // image-decode per lane; per register only a rem-walk + one div by Wa
  // (the per-reg /(Ha*Wa) pair was the other half of the issue stalls)
  const long m_base = bm0 + wm;
  const int HaWa = Ha * Wa;
  int nn0 = 0, rem0 = 0;
  if (m_base < Mc) {
    nn0 = (int)(m_base / HaWa);
    rem0 = (int)(m_base % HaWa);
  }
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    const long m_out = m_base + row;
    if (m_out < Mc) {
      int nn = nn0, rem = rem0 + row;
      while (rem >= HaWa) {
        rem -= HaWa;
        ++nn;
      }
      const int ih = 2 * (rem / Wa) + pa;
      const int iw = 2 * (rem - (rem / Wa) * Wa) + pb;
      const long off = (((long)nn * H + ih) * W + iw) * CI;
#pragma unroll
      for (int tnt = 0; tnt < NT; ++tnt) {
        float v = acc[tnt][reg];
        if constexpr (ADDIN)
          v += F16<T16>::to_f32(addin[off + k0 + tnt * 32 + li]);
        dx[off + k0 + tnt * 32 + li] = F16<T16>::from_f32(v);
      }
    }
  }
}

}  // namespace

// fold slab [by][gx][bnt2] -> stats[2,C]; two-level when gx is large
// (external linkage: the stem kernels reuse it)
void stats_slab_reduce(at::Tensor slab, at::Tensor stats, int gx, int bnt2,
                       int C) {
  const int by = (int)(slab.numel() / ((long)gx * bnt2));
  if (gx > 4096) {
    const int bpb1 = (int)cdiv_l(gx, 1024);
    const int gx2 = (int)cdiv_l(gx, bpb1);
    auto l2 = at::empty({(long)by * gx2 * bnt2}, slab.options());
    hipLaunchKernelGGL(conv_stats_partial, dim3(gx2, by), dim3(256), 0,
                       cur_stream(), slab.data_ptr<float>(),
                       l2.data_ptr<float>(), gx, bnt2, bpb1);
    slab = l2;
    gx = gx2;
  }
  const int bpb = (int)cdiv_l(gx, 64);
  dim3 rgrid((unsigned)cdiv_l(gx, bpb), by);
  hipLaunchKernelGGL(conv_stats_reduce, rgrid, dim3(256), 0, cur_stream(),
                     slab.data_ptr<float>(), stats.data_ptr<float>(), gx,
                     bnt2, C, bpb);
}

// ---------------------------------------------------------------------------
// ---------------------------------------------------------------------------
// Patch gather-GEMM for 3x3/stride-1/pad-1 (fwd and dgrad): the geometric
// input patch covering a 64-output tile (its q-rows +- 1 halo row, with
// zeroed pad columns) is staged into LDS ONCE per c-chunk; all NINE taps
// read it by address offset — the per-k-step global re-gather of the
// plain kernel (9x input traffic, 18 barriers/c-chunk) becomes one
// cooperative stage + 2 barriers. The B (weight) fragments live in
// REGISTERS, prefetched one tap ahead straight from L2 — no B tile in
// LDS, no B barriers. Vertical pad / image-crossing / M-tail are handled
// by a per-(lane, tap-row) address select onto a zeroed LDS stub (never a
// branch around the read).
//   block: 128m x 64k; 4 waves each own 32 m-rows x the full 64 k as TWO
//   32x32 acc tiles — one tile per wave makes every MFMA a serial
//   16-cycle dependency chain (measured 260-290 TF); two interleave to
//   the 8-cycle cadence.
//   patch: [NR rows][Wi+2 cols][72] (stride 72: 16B-aligned b128 reads,
//   36-dword lane stride -> worst 2-way bank conflicts).
// ---------------------------------------------------------------------------
constexpr int PCS = 72;  // patch col stride (elements)

template <typename T16, bool TRANS, bool ADDIN = false, bool SCALED = false>
__global__ __launch_bounds__(256, 2) void conv_patch_gemm(
    const T16* __restrict__ in,   // [N, Hi, Wi, CI] (dgrad: dy, CI=KO)
    const T16* __restrict__ wgt,  // fwd: [KO, 9*CI]; dgrad: wflip strides
    const float* __restrict__ bias,
    const T16* __restrict__ addin,  // ADDIN only: out += addin
    const float* __restrict__ asc,  // SCALED only: [CI] lazy-BN scale
    const float* __restrict__ ash,  // SCALED only: [CI] lazy-BN shift
    T16* __restrict__ out,
    float* __restrict__ stats_slab,  // null, or per-block (sum,sumsq) rows
    const int N, const int Hi, const int Wi, const int CI, const int KO,
    const long b_row_stride, const long b_rs_stride, const int act,
    const int has_bias, const int NR) {
  extern __shared__ __attribute__((aligned(16))) char psmem[];
  T16* patch = reinterpret_cast<T16*>(psmem);
  T16* zstub = patch + (long)NR * (Wi + 2) * PCS;  // 72 zero elements
  T16* ldsB = zstub + PCS;  // [64][LDK] weight tile for the current tap

  const int tid = threadIdx.x;
  const long Mtot = (long)N * Hi * Wi;  // Ho==Hi, Wo==Wi (s1p1)
  const int Wo = Wi, Ho = Hi;
  const long bm0 = (long)blockIdx.x * 128;
  const int k0 = blockIdx.y * 64;
  const long gr0 = bm0 / Wo - 1;  // first staged global row (n*Ho + p)

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int wm = wave * 32;  // wave's m-rows (full 64 k per wave)

  // per-lane A-row coordinates (fixed across the whole kernel)
  const long m_lane = bm0 + wm + li;
  const long grl = m_lane / Wo;
  const int p_lane = (int)(grl % Ho);
  const int prow = (int)(grl - gr0);
  const int pcol = (int)(m_lane - grl * Wo) + 1;
  const bool m_ok = m_lane < Mtot;
  const long lane_base = ((long)prow * (Wi + 2) + pcol) * PCS;

  if (tid < 64) zstub[tid] = T16{};  // first barrier publishes it

  const int cchunks = CI / BK;
  f32x16 acc[2] = {};

  // B staged through LDS, cooperatively and COALESCED (per-lane scattered
  // weight-row loads measured 86.6% WAIT_ANY — a wave's worth of strided
  // 64B reads with 1-tap prefetch never hides L2 latency; the block-wide
  // burst + barrier pipeline does). 4 threads per B row, 16 elements each.
  const int sb_n = tid >> 2;
  const int sb_c = (tid & 3) * 16;
  const T16* wrowB = wgt + (long)(k0 + sb_n) * b_row_stride + sb_c;
  short8 breg[2];
  auto load_b = [&](int tap, int cc) {
    const T16* wp = wrowB + (long)tap * b_rs_stride + cc * BK;
    breg[0] = *reinterpret_cast<const short8*>(wp);
    breg[1] = *reinterpret_cast<const short8*>(wp + 8);
  };
  auto stage_b = [&]() {
    short* pb = reinterpret_cast<short*>(ldsB + sb_n * LDK + sb_c);
    *reinterpret_cast<short8*>(pb) = breg[0];
    *reinterpret_cast<short8*>(pb + 8) = breg[1];
  };

  load_b(0, 0);

  for (int cc = 0; cc < cchunks; ++cc) {
    __syncthreads();  // previous c-chunk's patch reads done
    // ---- cooperative patch stage: rows gr0..gr0+NR-1, cols -1..Wi.
    // 4-deep load batches: the load->store pairs of the plain loop expose
    // one L2/HBM round trip per group (the fill is this kernel's dominant
    // serial phase — PMC: 67% parked) ----
    const int ngroups = NR * (Wi + 2) * (BK / 8);
    for (int i0f = tid; i0f < ngroups; i0f += 4 * 256) {
      const T16* srcs[4];
      short8 vals[4];
      int dsts[4], gs[4];
      bool ok[4];
#pragma unroll
      for (int b = 0; b < 4; ++b) {
        const int i = i0f + b * 256;
        const int g = i % (BK / 8);
        const int ce = i / (BK / 8);
        const int col = ce % (Wi + 2);
        const int row = ce / (Wi + 2);
        const long gr = gr0 + row;
        const int iw = col - 1;
        gs[b] = g;
        dsts[b] = i < ngroups ? (row * (Wi + 2) + col) * PCS + g * 8 : -1;
        ok[b] = i < ngroups && gr >= 0 && gr < (long)N * Ho &&
                (unsigned)iw < (unsigned)Wi;
        const long nn = ok[b] ? gr / Ho : 0;
        const long pp = ok[b] ? gr % Ho : 0;
        srcs[b] = ok[b]
                      ? in + ((nn * Hi + pp) * Wi + iw) * CI + cc * BK + g * 8
                      : zstub;  // always-loadable: batches the 4 round trips
      }
#pragma unroll
      for (int b = 0; b < 4; ++b)
        vals[b] = *reinterpret_cast<const short8*>(srcs[b]);
#pragma unroll
      for (int b = 0; b < 4; ++b) {
        if (dsts[b] < 0) break;
        short8 v = vals[b];
        if (!ok[b]) v = short8{};
        if constexpr (SCALED) {
          if (ok[b])
            v = scale8<T16>(v, sc8_load(asc, ash, cc * BK + gs[b] * 8));
        }
        *reinterpret_cast<short8*>(patch + dsts[b]) = v;
      }
    }
    __syncthreads();

#pragma unroll
    for (int r = 0; r < 3; ++r) {
      // tap row validity is per-lane but constant over s and kk
      const int pv = TRANS ? p_lane + 1 - r : p_lane + r - 1;
      const bool rv = m_ok && (unsigned)pv < (unsigned)Ho;
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        const int tap = r * 3 + s;
        const long off = TRANS
                             ? ((long)(1 - r) * (Wi + 2) + (1 - s)) * PCS
                             : ((long)(r - 1) * (Wi + 2) + (s - 1)) * PCS;
        const T16* abase = rv ? patch + lane_base + off : zstub;
        __syncthreads();  // previous tap's MFMAs done reading ldsB
        stage_b();
        __syncthreads();
        // next tap's (or next c-chunk's) B burst rides under the MFMAs
        if (tap < 8)
          load_b(tap + 1, cc);
        else if (cc + 1 < cchunks)
          load_b(0, cc + 1);
#pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          const short8 af = *reinterpret_cast<const short8*>(
              abase + kk * 16 + kh * 8);
          const short8 b0 = *reinterpret_cast<const short8*>(
              ldsB + li * LDK + kk * 16 + kh * 8);
          const short8 b1 = *reinterpret_cast<const short8*>(
              ldsB + (32 + li) * LDK + kk * 16 + kh * 8);
          acc[0] = Mfma32<T16>::run(af, b0, acc[0]);
          acc[1] = Mfma32<T16>::run(af, b1, acc[1]);
        }
      }
    }
  }

  // ---- epilogue: bias + act + store (+ optional BN-stats partials) ----
  float bv[2];
  bv[0] = has_bias ? bias[k0 + li] : 0.f;
  bv[1] = has_bias ? bias[k0 + 32 + li] : 0.f;
  float ssum[2] = {}, ssq[2] = {};
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    const long m_out = bm0 + wm + row;
    if (m_out < Mtot) {
#pragma unroll
      for (int t2 = 0; t2 < 2; ++t2) {
        float v = acc[t2][reg] + bv[t2];
        if constexpr (ADDIN)
          v += F16<T16>::to_f32(addin[m_out * KO + k0 + t2 * 32 + li]);
        if (act == 1) v = fmaxf(v, 0.f);
        if (stats_slab) {
          ssum[t2] += v;
          ssq[t2] += v * v;
        }
        out[m_out * KO + k0 + t2 * 32 + li] = F16<T16>::from_f32(v);
      }
    }
  }
  if (stats_slab) {
    // conv->BN fusion (same fold as conv_gather_gemm): kh halves by
    // xor-shuffle, waves by per-wave LDS rows (no LDS atomics)
#pragma unroll
    for (int t2 = 0; t2 < 2; ++t2) {
      ssum[t2] += __shfl_xor(ssum[t2], 32, 64);
      ssq[t2] += __shfl_xor(ssq[t2], 32, 64);
    }
    float* lsum = reinterpret_cast<float*>(psmem);  // [4 waves][128]
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

// wgrad: dw[KO, kg] = sum_m dy[m, KO] * A_im2col[m, kg]. MFMA reduces over
// its register-minor k dim, which here is m — both operands are m-major in
// memory, so both tiles are staged TRANSPOSED into LDS ([k][m] / [kg][m]),
// packing 4 m-values per ds_write_b64 during staging. Each block owns a
// 64(KO) x 64(kg at one (r,s,c-chunk)) output tile and an m-chunk; each
// chunk stores a disjoint fp32 partial slab (split-M fills the chip for
// the small late-layer filter counts) and wgrad_reduce_chunks sums them —
// fp32 atomicAdd on the small dw region measured ~21 ns/op amortized.
// ---------------------------------------------------------------------------

namespace {

constexpr int WGM = 64;   // m per step
constexpr int LDM = 72;   // m-minor row length (+16B: alignment + banks)

// KT = KO-tile width (64 or 128): wider k-tiles amortize the x staging
// and double the MFMA work per m-step for the K>=128 layers.
template <typename T16, int KT, bool SCALED = false, bool KMIN = false>
// the second launch-bounds arg (min 2 blocks/CU) is load-bearing: without
// it the compiler allocates 188-256 VGPRs (occupancy 1-2) for no speedup;
// with it 104-124 VGPRs, zero spills, occupancy 4 - the extra waves are
// what covers the gather's HBM latency (PMC: 66% WAIT_ANY at occ 2).
// SCALED variants pin the unscaled occupancy (the Sc8 regs cost a tier).
// KMIN: k-minor staging + rotate-swizzled columns (conv_wgrad_mfma_s3).
__global__ __launch_bounds__(256, SCALED ? 4 : 2) void conv_wgrad_mfma_kernel(
    const T16* __restrict__ x,    // [N, Hi, Wi, CI]
    const T16* __restrict__ dy,   // [M, KO]
    const float* __restrict__ asc,  // SCALED only: [CI] lazy-BN scale
    const float* __restrict__ ash,
    float* __restrict__ dw,       // [KO, R*S*CI]
    const int N, const int Hi, const int Wi, const int CI, const int KO,
    const int Ho, const int Wo, const int R, const int S, const int stride,
    const int pad, const long m_per_chunk, const int nchunks) {
  __shared__ T16 lds[(KT + 64) * LDM];

  const int tid = threadIdx.x;
  const long Mtot = (long)N * Ho * Wo;
  const int k0 = blockIdx.x * KT;
  const int cchunks = CI / BK;
  const int c0 = (blockIdx.y % cchunks) * BK;
  const int s_ = (blockIdx.y / cchunks) % S;
  const int r_ = blockIdx.y / (cchunks * S);
  const long m_begin = (long)blockIdx.z * m_per_chunk;
  const long m_end = min(Mtot, m_begin + m_per_chunk);

  // staging: dyT uses 2*KT threads (4 m-rows x 8 k each); xT uses 128
  // threads — the upper half when KT==64 (disjoint roles), the lower half
  // when KT==128 (sequential double duty)
  const bool do_dy = tid < 2 * KT;
  const bool do_x = KT == 64 ? tid >= 128 : tid < 128;
  const int t = do_dy ? tid : 0;
  constexpr int KG = KT / 8;             // k-groups per operand row set
  const int sm = KMIN ? (t / KG) * 4 : (t & 15) * 4;   // m offset (4 rows)
  const int sk = KMIN ? (t % KG) * 8 : (t >> 4) * 8;   // k offset (dyT rows)
  const int tx = tid & 127;
  const int smx = KMIN ? (tx >> 3) * 4 : (tx & 15) * 4;
  const int skx = KMIN ? (tx & 7) * 8 : (tx >> 4) * 8;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  constexpr int NJ = KT / 64;       // kg sub-tiles per wave
  const int i0 = KT == 64 ? (wave & 1) * 32 : wave * 32;  // KO sub-tile
  const int j0base = KT == 64 ? (wave >> 1) * 32 : 0;

  f32x16 acc[NJ] = {};
  Sc8 wsc{};  // SCALED: this thread's x channels are fixed (c0+skx)
  if constexpr (SCALED) {
    if (do_x) wsc = sc8_load(asc, ash, c0 + skx);
  }

  // incremental (n,p,q) decode for the x-gather: one div/mod at entry,
  // add-with-carry as m advances by WGM per step (divisions in the inner
  // loop were ~40% of this kernel's time)
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

  // gather + pack 4 m-rows of 8 elems into registers for m-step m0
  short8 vdy[4], vx[4];
  auto load_m = [&](long m0) {
    if (do_dy) {
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const long m = m0 + sm + mi;
        vdy[mi] = m < m_end ? *reinterpret_cast<const short8*>(
                                  dy + m * KO + k0 + sk)
                            : short8{};
      }
    }
    if (do_x) {
      int n_ = dn, p_ = dp, q_ = dq;
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const long m = m0 + smx + mi;
        const int ih = p_ * stride - pad + r_;
        const int iw = q_ * stride - pad + s_;
        const bool ok = m < m_end && (unsigned)ih < (unsigned)Hi &&
                        (unsigned)iw < (unsigned)Wi;
        vx[mi] = ok ? *reinterpret_cast<const short8*>(
                          x + (((long)n_ * Hi + ih) * Wi + iw) * CI + c0 +
                          skx)
                    : short8{};
        if constexpr (SCALED) {
          if (ok) vx[mi] = scale8<T16>(vx[mi], wsc);
        }
        if (mi < 3 && ++q_ == Wo) {
          q_ = 0;
          if (++p_ == Ho) {
            p_ = 0;
            ++n_;
          }
        }
      }
      advance(WGM);  // position for the NEXT m-step's gather
    }
  };
  // transpose-write the 4x8 register patches
  auto stage_m = [&]() {
    if (do_dy) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        short4v pk = {vdy[0][e], vdy[1][e], vdy[2][e], vdy[3][e]};
        const int col = KMIN ? ((sm + ((sk + e) & 56)) & 63) : sm;
        *reinterpret_cast<short4v*>(
            reinterpret_cast<short*>(lds + (sk + e) * LDM + col)) = pk;
      }
    }
    if (do_x) {
      T16* ldsT = lds + KT * LDM;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        short4v pk = {vx[0][e], vx[1][e], vx[2][e], vx[3][e]};
        const int col = KMIN ? ((smx + ((skx + e) & 56)) & 63) : smx;
        *reinterpret_cast<short4v*>(
            reinterpret_cast<short*>(ldsT + (skx + e) * LDM + col)) = pk;
      }
    }
  };

  load_m(m_begin);
  for (long m0 = m_begin; m0 < m_end; m0 += WGM) {
    __syncthreads();
    stage_m();
    __syncthreads();
    if (m0 + WGM < m_end) load_m(m0 + WGM);
    const T16* ldsDyT = lds;
    const T16* ldsXT = lds + KT * LDM;
    const int arow = i0 + li;
    const int arot = KMIN ? (arow & 56) : 0;
#pragma unroll
    for (int kk = 0; kk < WGM; kk += 16) {
      const short8 af = *reinterpret_cast<const short8*>(
          ldsDyT + arow * LDM + ((kk + kh * 8 + arot) & 63));
#pragma unroll
      for (int jj = 0; jj < NJ; ++jj) {
        const int brow = j0base + jj * 32 + li;
        const short8 bf = *reinterpret_cast<const short8*>(
            ldsXT + brow * LDM +
            ((kk + kh * 8 + (KMIN ? (brow & 56) : 0)) & 63));
        acc[jj] = Mfma32<T16>::run(af, bf, acc[jj]);
      }
    }
  }

  // ---- scatter the 32x32 fp32 tile, directly in the parameter layout
  // [KO, CI, R, S]. Split-M chunks write DISJOINT per-chunk slabs (plain
  // stores; a reduce kernel sums them) — fp32 atomicAdd contention on the
  // small dw region measured ~21 ns/op amortized (layer1 chunk ablation),
  // dominating the kernel at high chunk counts ----
  const int rs = r_ * S + s_;
  const long RS = (long)R * S;
  float* slab = dw + (long)blockIdx.z * ((long)KO * CI * RS);
#pragma unroll
  for (int jj = 0; jj < NJ; ++jj) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int i = (reg & 3) + 8 * (reg >> 2) + 4 * kh;  // KO row
      const int c_abs = c0 + j0base + jj * 32 + li;
      slab[((long)(k0 + i0 + i) * CI + c_abs) * RS + rs] = acc[jj][reg];
    }
  }
}

// 128x128-output-tile wgrad for KO%128 && CI%128 (the ResNet-50 1x1
// bottleneck convs carry ~45% of its FLOPs and can't use the s-grouped
// kernel): each wave owns FOUR 32x32 acc tiles (2 KO x 2 CI), so one
// dy+x stage feeds 16 MFMAs per wave per m-step instead of 8, and each
// fragment read is reused twice. Same transposed-staging/split-M/slab
// machinery as conv_wgrad_mfma_kernel.
// SCALED: min 3 waves/SIMD — the Sc8 registers pushed the allocation 4
// VGPRs over the 168 boundary (occ 3 -> 2, measured 1.77x slower)
template <typename T16, bool SCALED = false, bool KMIN = false>
__global__ __launch_bounds__(256, SCALED ? 3 : 2) void conv_wgrad_mfma_t128(
    const T16* __restrict__ x, const T16* __restrict__ dy,
    const float* __restrict__ asc,  // SCALED only
    const float* __restrict__ ash,
    float* __restrict__ dw,  // chunk slabs of [KO, R*S*CI]
    const int N, const int Hi, const int Wi, const int CI, const int KO,
    const int Ho, const int Wo, const int R, const int S, const int stride,
    const int pad, const long m_per_chunk, const int nchunks) {
  __shared__ T16 lds[(128 + 128) * LDM];

  const int tid = threadIdx.x;
  const long Mtot = (long)N * Ho * Wo;
  const int k0 = blockIdx.x * 128;
  const int cblocks = CI / 128;
  const int c0 = (blockIdx.y % cblocks) * 128;
  const int s_ = (blockIdx.y / cblocks) % S;
  const int r_ = blockIdx.y / (cblocks * S);
  const long m_begin = (long)blockIdx.z * m_per_chunk;
  const long m_end = min(Mtot, m_begin + m_per_chunk);

  // staging: all 256 threads stage dy (4m x 8k over 128 k-rows) and x
  // (4m x 8c over 128 c-rows) with the same (m-group, row-group) map.
  // KMIN: k-minor map + rotate-swizzled columns (see conv_wgrad_mfma_s3)
  const int sm = KMIN ? (tid >> 4) * 4 : (tid & 15) * 4;
  const int sk = KMIN ? (tid & 15) * 8 : (tid >> 4) * 8;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int i0 = (wave & 1) * 32;   // KO sub-offset
  const int j0 = (wave >> 1) * 32;  // CI sub-offset

  f32x16 acc[2][2] = {};
  Sc8 wsc{};
  if constexpr (SCALED) wsc = sc8_load(asc, ash, c0 + sk);

  int dn = 0, dp = 0, dq = 0;
  {
    const long m_first = m_begin + sm;
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

  short8 vdy[4], vx0[4];
  auto load_m = [&](long m0) {
    int n_ = dn, p_ = dp, q_ = dq;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const long m = m0 + sm + mi;
      vdy[mi] = m < m_end ? *reinterpret_cast<const short8*>(
                                dy + m * KO + k0 + sk)
                          : short8{};
      const int ih = p_ * stride - pad + r_;
      const int iw = q_ * stride - pad + s_;
      const bool ok = m < m_end && (unsigned)ih < (unsigned)Hi &&
                      (unsigned)iw < (unsigned)Wi;
      const T16* xp = x + (((long)n_ * Hi + ih) * Wi + iw) * CI + c0 + sk;
      vx0[mi] = ok ? *reinterpret_cast<const short8*>(xp) : short8{};
      if constexpr (SCALED) {
        if (ok) vx0[mi] = scale8<T16>(vx0[mi], wsc);
      }
      if (mi < 3 && ++q_ == Wo) {
        q_ = 0;
        if (++p_ == Ho) {
          p_ = 0;
          ++n_;
        }
      }
    }
    advance(WGM);
  };
  auto stage_m = [&]() {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int col = KMIN ? ((sm + ((sk + e) & 56)) & 63) : sm;
      short4v pk = {vdy[0][e], vdy[1][e], vdy[2][e], vdy[3][e]};
      *reinterpret_cast<short4v*>(
          reinterpret_cast<short*>(lds + (sk + e) * LDM + col)) = pk;
      short4v px0 = {vx0[0][e], vx0[1][e], vx0[2][e], vx0[3][e]};
      *reinterpret_cast<short4v*>(reinterpret_cast<short*>(
          lds + (128 + sk + e) * LDM + col)) = px0;
    }
  };

  load_m(m_begin);
  for (long m0 = m_begin; m0 < m_end; m0 += WGM) {
    __syncthreads();
    stage_m();
    __syncthreads();
    if (m0 + WGM < m_end) load_m(m0 + WGM);
    const T16* ldsDyT = lds;
    const T16* ldsXT = lds + 128 * LDM;
#pragma unroll
    for (int kk = 0; kk < WGM; kk += 16) {
      short8 af[2], bf[2];
#pragma unroll
      for (int a = 0; a < 2; ++a) {
        const int row = i0 + a * 64 + li;
        af[a] = *reinterpret_cast<const short8*>(
            ldsDyT + row * LDM +
            ((kk + kh * 8 + (KMIN ? (row & 56) : 0)) & 63));
      }
#pragma unroll
      for (int b = 0; b < 2; ++b) {
        const int row = j0 + b * 64 + li;
        bf[b] = *reinterpret_cast<const short8*>(
            ldsXT + row * LDM +
            ((kk + kh * 8 + (KMIN ? (row & 56) : 0)) & 63));
      }
#pragma unroll
      for (int a = 0; a < 2; ++a)
#pragma unroll
        for (int b = 0; b < 2; ++b)
          acc[a][b] = Mfma32<T16>::run(af[a], bf[b], acc[a][b]);
    }
  }

  const int rs = r_ * S + s_;
  const long RS = (long)R * S;
  float* slab = dw + (long)blockIdx.z * ((long)KO * CI * RS);
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b)
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int i = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
        const int c_abs = c0 + j0 + b * 64 + li;
        slab[((long)(k0 + i0 + a * 64 + i) * CI + c_abs) * RS + rs] =
            acc[a][b][reg];
      }
}

// s-grouped wgrad for the dominant 3x3/stride-1/pad-1 convs: ONE block
// computes all 3 s-taps of one filter row r from ONE dy+x stage. The
// three x operand images are m-SHIFTED copies of the same center gather
// (x_im2col[m, s] = x_center[m + s - 1] within a q-row): each staging
// thread gathers 6 m-positions (4 + one halo on each side) and packs the
// three aligned 4-m transposed writes from its own registers, zeroing
// slots that cross a q-row boundary (q==0 for s=0, q==Wo-1 for s=2).
// Triples the MFMA work per staged dy/x byte — the plain kernel was
// staging-bound at ~190 TF while the fwd gather-GEMM reaches 470-670.
// KMIN (KT=64 only): k-minor staging-thread map + rotate-swizzled LDS
// columns — one wave's gather covers 8 full m-lines instead of 16
// quarter-lines (PMC: 53% parked on the gather latency), and the rotation
// keeps the transposed b64 writes bank-conflict-free where a plain
// k-minor map would collide 8-way (8*LDM elements = 0 mod 32 banks).
template <typename T16, int KT, bool SCALED = false, bool KMIN = false>
__global__ __launch_bounds__(256, SCALED ? 3 : 2) void conv_wgrad_mfma_s3(
    const T16* __restrict__ x,    // [N, Hi, Wi, CI]
    const T16* __restrict__ dy,   // [M, KO]
    const float* __restrict__ asc,  // SCALED only: [CI] lazy-BN scale
    const float* __restrict__ ash,
    float* __restrict__ dw,       // chunk slabs of [KO, 3*3*CI]
    const int N, const int Hi, const int Wi, const int CI, const int KO,
    const int Ho, const int Wo, const long m_per_chunk, const int nchunks) {
  __shared__ T16 lds[(KT + 3 * 64) * LDM];

  const int tid = threadIdx.x;
  const long Mtot = (long)N * Ho * Wo;
  const int k0 = blockIdx.x * KT;
  const int cchunks = CI / BK;
  const int c0 = (blockIdx.y % cchunks) * BK;
  const int r_ = blockIdx.y / cchunks;
  const long m_begin = (long)blockIdx.z * m_per_chunk;
  const long m_end = min(Mtot, m_begin + m_per_chunk);

  const bool do_dy = tid < 2 * KT;
  const bool do_x = KT == 64 ? tid >= 128 : tid < 128;
  const int t = do_dy ? tid : 0;
  const int sm = KMIN ? ((t >> 3) & 15) * 4 : (t & 15) * 4;
  const int sk = KMIN ? (t & 7) * 8 : (t >> 4) * 8;
  const int tx = tid & 127;
  const int smx = KMIN ? (tx >> 3) * 4 : (tx & 15) * 4;
  const int skx = KMIN ? (tx & 7) * 8 : (tx >> 4) * 8;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  constexpr int NJ = KT / 64;
  const int i0 = KT == 64 ? (wave & 1) * 32 : wave * 32;
  const int j0base = KT == 64 ? (wave >> 1) * 32 : 0;

  f32x16 acc[3][NJ] = {};
  Sc8 wsc{};
  if constexpr (SCALED) {
    if (do_x) wsc = sc8_load(asc, ash, c0 + skx);
  }

  // incremental decode for the CENTER gather position m = m0 + smx (the
  // i=0 halo position is derived from it: clamping a -1 start desyncs the
  // per-step advance for the smx==0 thread — measured as a wrong dw tile)
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

  short8 vdy[4], vx[6];
  int qg[6];  // q of each gathered position (for the row-boundary masks)
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
      int n_ = dn, p_ = dp, q_ = dq;  // position m0 + smx (center start)
      {
        // i = 0 halo: column q-1 of the SAME row; when q == 0 the only
        // consumer slot (copy_0 at a q==0 target) is dead-masked, so a
        // zero stands in
        const int ih = p_ + r_ - 1;  // stride 1, pad 1
        qg[0] = q_ - 1;
        const bool ok = m0 + smx - 1 >= 0 && q_ > 0 &&
                        (unsigned)ih < (unsigned)Hi;
        vx[0] = ok ? *reinterpret_cast<const short8*>(
                         x + (((long)n_ * Hi + ih) * Wi + (q_ - 1)) * CI +
                         c0 + skx)
                   : short8{};
        if constexpr (SCALED) {
          if (ok) vx[0] = scale8<T16>(vx[0], wsc);
        }
      }
#pragma unroll
      for (int i = 1; i < 6; ++i) {
        const long m = m0 + smx - 1 + i;
        const int ih = p_ + r_ - 1;
        qg[i] = q_;
        const bool ok = m < Mtot && (unsigned)ih < (unsigned)Hi;
        vx[i] = ok ? *reinterpret_cast<const short8*>(
                         x + (((long)n_ * Hi + ih) * Wi + q_) * CI + c0 +
                         skx)
                   : short8{};
        if constexpr (SCALED) {
          if (ok) vx[i] = scale8<T16>(vx[i], wsc);
        }
        if (i < 5 && ++q_ == Wo) {
          q_ = 0;
          if (++p_ == Ho) {
            p_ = 0;
            ++n_;
          }
        }
      }
      advance(WGM);
    }
  };
  auto stage_m = [&]() {
    if (do_dy) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        short4v pk = {vdy[0][e], vdy[1][e], vdy[2][e], vdy[3][e]};
        const int col = KMIN ? ((sm + ((sk + e) & 56)) & 63) : sm;
        *reinterpret_cast<short4v*>(
            reinterpret_cast<short*>(lds + (sk + e) * LDM + col)) = pk;
      }
    }
    if (do_x) {
      // copy s slot j <- center gather index j + s; boundary slots zeroed
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        T16* ldsS = lds + (KT + s * 64) * LDM;
        short keep[4];
#pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
          const int qt = qg[jj + 1];  // q of target output m
          const bool dead = (s == 0 && qt == 0) || (s == 2 && qt == Wo - 1);
          keep[jj] = dead ? (short)0 : (short)-1;
        }
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          short4v pk = {(short)(vx[s + 0][e] & keep[0]),
                        (short)(vx[s + 1][e] & keep[1]),
                        (short)(vx[s + 2][e] & keep[2]),
                        (short)(vx[s + 3][e] & keep[3])};
          const int col = KMIN ? ((smx + ((skx + e) & 56)) & 63) : smx;
          *reinterpret_cast<short4v*>(
              reinterpret_cast<short*>(ldsS + (skx + e) * LDM + col)) = pk;
        }
      }
    }
  };

  load_m(m_begin);
  for (long m0 = m_begin; m0 < m_end; m0 += WGM) {
    __syncthreads();
    stage_m();
    __syncthreads();
    if (m0 + WGM < m_end) load_m(m0 + WGM);
    const T16* ldsDyT = lds;
    const int arow = i0 + li;
    const int arot = KMIN ? (arow & 56) : 0;
#pragma unroll
    for (int kk = 0; kk < WGM; kk += 16) {
      const short8 af = *reinterpret_cast<const short8*>(
          ldsDyT + arow * LDM + ((kk + kh * 8 + arot) & 63));
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        const T16* ldsXT = lds + (KT + s * 64) * LDM;
#pragma unroll
        for (int jj = 0; jj < NJ; ++jj) {
          const int brow = j0base + jj * 32 + li;
          const short8 bf = *reinterpret_cast<const short8*>(
              ldsXT + brow * LDM +
              ((kk + kh * 8 + (KMIN ? (brow & 56) : 0)) & 63));
          acc[s][jj] = Mfma32<T16>::run(af, bf, acc[s][jj]);
        }
      }
    }
  }

  float* slab = dw + (long)blockIdx.z * ((long)KO * CI * 9);
#pragma unroll
  for (int s = 0; s < 3; ++s) {
    const int rs = r_ * 3 + s;
#pragma unroll
    for (int jj = 0; jj < NJ; ++jj) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int i = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
        const int c_abs = c0 + j0base + jj * 32 + li;
        slab[((long)(k0 + i0 + i) * CI + c_abs) * 9 + rs] = acc[s][jj][reg];
      }
    }
  }
}

// out[zc][e] = sum over this block-row's z-range of part[z][e] — the
// z-parallel first level of the two-level reduce (small E can't fill the
// chip with element-parallelism alone)
__global__ void wgrad_reduce_zchunk(const float* __restrict__ part,
                                    float* __restrict__ out, long E, int nz,
                                    int z_per) {
  const int z0 = blockIdx.y * z_per;
  const int z1 = min(nz, z0 + z_per);
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i < E; i += stride) {
    if (i + 4 <= E) {
      float4v acc = *reinterpret_cast<const float4v*>(part + (long)z0 * E + i);
      for (int z = z0 + 1; z < z1; ++z) {
        float4v v = *reinterpret_cast<const float4v*>(part + (long)z * E + i);
#pragma unroll
        for (int u = 0; u < 4; ++u) acc[u] += v[u];
      }
      *reinterpret_cast<float4v*>(out + (long)blockIdx.y * E + i) = acc;
    } else {
      for (long e = i; e < E; ++e) {
        float a = part[(long)z0 * E + e];
        for (int z = z0 + 1; z < z1; ++z) a += part[(long)z * E + e];
        out[(long)blockIdx.y * E + e] = a;
      }
    }
  }
}

// dw[e] = sum_z part[z][e] (fp32, vectorized)
__global__ void wgrad_reduce_chunks(const float* __restrict__ part,
                                    float* __restrict__ dw, long E, int nz) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i < E; i += stride) {
    if (i + 4 <= E) {
      float4v acc = *reinterpret_cast<const float4v*>(part + i);
      for (int z = 1; z < nz; ++z) {
        float4v v = *reinterpret_cast<const float4v*>(part + (long)z * E + i);
#pragma unroll
        for (int u = 0; u < 4; ++u) acc[u] += v[u];
      }
      *reinterpret_cast<float4v*>(dw + i) = acc;
    } else {
      for (long e = i; e < E; ++e) {
        float a = part[e];
        for (int z = 1; z < nz; ++z) a += part[(long)z * E + e];
        dw[e] = a;
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// glds-pipelined fwd variant: the A (gathered input) tile is staged by
// `global_load_lds` (direct HBM->LDS DMA, no VGPR round-trip, no ds_write
// pass — guide §5 ladder step 3) into a double-buffered LINEAR [BM][BK]
// image with an XOR-swizzled SOURCE address (rule 21: linear dest +
// inverse-swizzled source + swizzled read), zero-page redirect for
// padding rows. ONE __syncthreads per k-step: its implicit vmcnt(0)
// drains the in-flight DMA for the NEXT step's buffer while this step's
// MFMA runs above it.
// ---------------------------------------------------------------------------
namespace {

template <typename T16, int NT>
__global__ __launch_bounds__(256) void conv_fwd_glds(
    const T16* __restrict__ in, const T16* __restrict__ wgt,
    const float* __restrict__ bias, const T16* __restrict__ zpage,
    T16* __restrict__ out, const int N, const int Hi, const int Wi,
    const int CI, const int KO, const int Ho, const int Wo, const int R,
    const int S, const int stride, const int pad, const int act,
    const int has_bias) {
  constexpr int BNT = NT * 32;
  __shared__ T16 lds[2 * BM * BK + 2 * BNT * LDK];

  const int tid = threadIdx.x;
  const long Mtot = (long)N * Ho * Wo;
  const long bm0 = (long)blockIdx.x * BM;
  const int k0 = blockIdx.y * BNT;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int wm = wave * 32;

  // ---- A rows this lane feeds (4 glds per wave, 8 rows each) ----
  const int lr = lane >> 3;          // row-within-8 of each glds
  const int ce = (lane & 7) * 8;     // element col of the 16B piece
  const int swz = lr * 8;            // XOR swizzle, elements ((r&7)<<4 B)
  long nbase[4];
  int ih0v[4], iw0v[4];
  bool mokv[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int r = 32 * wave + 8 * i + lr;
    const long m = bm0 + r;
    mokv[i] = m < Mtot;
    int n_ = 0, p_ = 0, q_ = 0;
    if (mokv[i]) {
      n_ = (int)(m / ((long)Ho * Wo));
      const int pq = (int)(m % ((long)Ho * Wo));
      p_ = pq / Wo;
      q_ = pq % Wo;
    }
    nbase[i] = (long)n_ * Hi * Wi * CI;
    ih0v[i] = p_ * stride - pad;
    iw0v[i] = q_ * stride - pad;
  }

  // ---- B staging (register -> LDS, padded LDK rows) ----
  constexpr int TPR = 256 / BNT;
  constexpr int EPT = BK / TPR;
  const int sb_n = tid / TPR;
  const int sb_c = (tid % TPR) * EPT;
  const T16* wrow = wgt + (long)(k0 + sb_n) * R * S * CI + sb_c;
  short8 sb[EPT / 8];

  const int cchunks = CI / BK;
  const int ksteps = R * S * cchunks;

  auto issue_gldsA = [&](int j, int buf) {
    const int c0 = (j % cchunks) * BK;
    const int s_ = (j / cchunks) % S;
    const int r_ = j / (cchunks * S);
    T16* dstA = lds + buf * (BM * BK);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int ih = ih0v[i] + r_;
      const int iw = iw0v[i] + s_;
      const bool ok = mokv[i] && (unsigned)ih < (unsigned)Hi &&
                      (unsigned)iw < (unsigned)Wi;
      const T16* src =
          ok ? in + nbase[i] + ((long)ih * Wi + iw) * CI + c0 + (ce ^ swz)
             : zpage + (ce ^ swz);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(dstA +
                                                    (32 * wave + 8 * i) * BK),
          16, 0, 0);
    }
  };
  auto load_B = [&](int j) {
    const int c0 = (j % cchunks) * BK;
    const int s_ = (j / cchunks) % S;
    const int r_ = j / (cchunks * S);
    const T16* wp = wrow + (long)(r_ * S + s_) * CI + c0;
#pragma unroll
    for (int i = 0; i < EPT / 8; ++i)
      sb[i] = *reinterpret_cast<const short8*>(wp + 8 * i);
  };
  auto stage_B = [&](int buf) {
    short* pb = reinterpret_cast<short*>(lds + 2 * BM * BK + buf * BNT * LDK +
                                         sb_n * LDK + sb_c);
#pragma unroll
    for (int i = 0; i < EPT / 8; ++i)
      *reinterpret_cast<short8*>(pb + 8 * i) = sb[i];
  };

  f32x16 acc[NT] = {};

  issue_gldsA(0, 0);
  load_B(0);
  stage_B(0);
  __syncthreads();  // drains glds(0) (implicit vmcnt 0) + B writes
  for (int j = 0; j < ksteps; ++j) {
    const int nxt = (j + 1) & 1;
    if (j + 1 < ksteps) {
      issue_gldsA(j + 1, nxt);  // DMA runs under this step's MFMA
      load_B(j + 1);
    }
    const T16* ldsA = lds + (j & 1) * (BM * BK);
    const T16* ldsB = lds + 2 * BM * BK + (j & 1) * BNT * LDK;
#pragma unroll
    for (int kk = 0; kk < BK; kk += 16) {
      const int m = wm + li;
      const short8 af = *reinterpret_cast<const short8*>(
          ldsA + m * BK + ((kk + kh * 8) ^ ((m & 7) * 8)));
#pragma unroll
      for (int tnt = 0; tnt < NT; ++tnt) {
        const short8 bf = *reinterpret_cast<const short8*>(
            ldsB + (tnt * 32 + li) * LDK + kk + kh * 8);
        acc[tnt] = Mfma32<T16>::run(af, bf, acc[tnt]);
      }
    }
    if (j + 1 < ksteps) stage_B(nxt);  // B reg latency hid under the MFMAs
    __syncthreads();
  }

  float bv[NT];
#pragma unroll
  for (int tnt = 0; tnt < NT; ++tnt)
    bv[tnt] = has_bias ? bias[k0 + tnt * 32 + li] : 0.f;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    const long m_out = bm0 + wm + row;
    if (m_out < Mtot) {
#pragma unroll
      for (int tnt = 0; tnt < NT; ++tnt) {
        float v = acc[tnt][reg] + bv[tnt];
        if (act == 1) v = fmaxf(v, 0.f);
        out[m_out * KO + k0 + tnt * 32 + li] = F16<T16>::from_f32(v);
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// halo wgrad for the dominant 3x3/stride-1/pad-1 convs: ONE block computes
// all NINE taps of a 64(KO) x 64(CI) dw tile over its m-chunk, so dy and x
// are staged ONCE per 64-pixel m-step instead of once per tap (the tap-
// per-block kernel re-reads each ~9x and measured HBM/L3-traffic-bound).
// x is staged as a zero-bordered 2-D spatial halo image [(64/W)+2 rows]
// [W+2 cols][64 ch]; tap (r,s) of pixel (p,q) reads image[p-prow0+r][q+s]
// so padding needs no per-element masking. The b-fragment m-gather is 8
// scalar u16 LDS reads with power-of-two (W) row decode.
// ---------------------------------------------------------------------------

namespace {

constexpr int HALO_IMG_MAX = 224;  // pixels: covers W=8..64 (+2 halo rows)

template <typename T16>
__global__ __launch_bounds__(256) void conv_wgrad_halo3(
    const T16* __restrict__ x,   // [N,H,W,CI]
    const T16* __restrict__ dy,  // [M,KO]
    float* __restrict__ part,    // [nchunks][KO*CI*9] (KCRS slabs)
    const int N, const int H, const int W, const int CI, const int KO,
    const int wshift, const long m_per_chunk) {
  __shared__ T16 lds[64 * LDM + HALO_IMG_MAX * 64];
  T16* dyT = lds;                 // [64 k][LDM m]
  T16* ximg = lds + 64 * LDM;     // [pix][64 c]

  const int tid = threadIdx.x;
  const long Mtot = (long)N * H * W;
  const int k0 = blockIdx.x * 64;
  const int c0 = blockIdx.y * 64;
  const long m_begin = (long)blockIdx.z * m_per_chunk;
  const long m_end = min(Mtot, m_begin + m_per_chunk);

  const int RW = 64 >> wshift;    // pixel rows per m-step
  const int IMG_C = W + 2;
  const int npix = (RW + 2) * IMG_C;

  // dyT staging (threads 0..127): 4 m-rows x 8 k each, transposed write
  const int t7 = tid & 127;
  const int sm = (t7 & 15) * 4;
  const int sk = (t7 >> 4) * 8;
  // ximg staging (all 256): 4 threads per pixel, 16 channels each
  const int px_slot = tid >> 2;
  const int px_c = (tid & 3) * 16;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int i0 = (wave & 1) * 32;   // KO half
  const int j0 = (wave >> 1) * 32;  // CI half

  f32x16 acc[9] = {};

  // uniform per-step position (advanced incrementally below)
  int n_ = (int)(m_begin / ((long)H * W));
  int prow0 = (int)((m_begin % ((long)H * W)) >> wshift);

  for (long m0 = m_begin; m0 < m_end; m0 += 64) {
    __syncthreads();  // previous MFMA phase done with LDS
    // ---- stage dyT ----
    if (tid < 128) {
      short8 v[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        v[mi] = *reinterpret_cast<const short8*>(dy + (m0 + sm + mi) * KO +
                                                 k0 + sk);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        short4v pk = {v[0][e], v[1][e], v[2][e], v[3][e]};
        *reinterpret_cast<short4v*>(
            reinterpret_cast<short*>(dyT + (sk + e) * LDM + sm)) = pk;
      }
    }
    // ---- stage the zero-bordered x halo image ----
    for (int pix = px_slot; pix < npix; pix += 64) {
      const int ir = pix / IMG_C;
      const int ic = pix - ir * IMG_C;
      const int p_src = prow0 - 1 + ir;
      const int q_src = ic - 1;
      const bool ok = (unsigned)p_src < (unsigned)H &&
                      (unsigned)q_src < (unsigned)W;
      short8 a = {}, b = {};
      if (ok) {
        const T16* src =
            x + (((long)n_ * H + p_src) * W + q_src) * CI + c0 + px_c;
        a = *reinterpret_cast<const short8*>(src);
        b = *reinterpret_cast<const short8*>(src + 8);
      }
      short* dst = reinterpret_cast<short*>(ximg + (long)pix * 64 + px_c);
      *reinterpret_cast<short8*>(dst) = a;
      *reinterpret_cast<short8*>(dst + 8) = b;
    }
    __syncthreads();
    // ---- 4 k-steps x 9 taps of MFMA ----
#pragma unroll
    for (int kk = 0; kk < 64; kk += 16) {
      const short8 af = *reinterpret_cast<const short8*>(
          dyT + (i0 + li) * LDM + kk + kh * 8);
#pragma unroll
      for (int tap = 0; tap < 9; ++tap) {
        const int r = tap / 3, ss = tap % 3;
        short8 bf;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int mrel = kk + kh * 8 + e;
          const int prow = mrel >> wshift;
          const int pcol = mrel & (W - 1);
          bf[e] = *reinterpret_cast<const short*>(
              ximg + ((long)(prow + r) * IMG_C + pcol + ss) * 64 + j0 + li);
        }
        acc[tap] = Mfma32<T16>::run(af, bf, acc[tap]);
      }
    }
    prow0 += RW;
    if (prow0 == H) {
      prow0 = 0;
      ++n_;
    }
  }

  // ---- per-chunk slab store, parameter layout [KO,CI,3,3] ----
  float* slab = part + (long)blockIdx.z * ((long)KO * CI * 9);
#pragma unroll
  for (int tap = 0; tap < 9; ++tap) {
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int i = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
      slab[((long)(k0 + i0 + i) * CI + c0 + j0 + li) * 9 + tap] =
          acc[tap][reg];
    }
  }
}

}  // namespace

bool conv_mfma_supported(long CI, long KO) {
  return CI % 64 == 0 && KO % 64 == 0;
}

// 256-element zero page: out-of-bounds gathers SELECT this address instead
// of branching around the load — a per-element `ok ? load : 0` makes hipcc
// branch around each load and wait vmcnt(0) per element (guide §5 trap 4c:
// 32 dependent L2 round trips), which measured as ~1.5 us per OUTPUT in
// the stem kernels.
at::Tensor conv_zero_page(const at::Tensor& like) {
  static at::Tensor zp16, zph;
  at::Tensor& zp = like.scalar_type() == at::kBFloat16 ? zp16 : zph;
  if (!zp.defined() || zp.device() != like.device())
    zp = at::zeros({256}, like.options());
  return zp;
}

void wgrad_reduce_launch(at::Tensor part, at::Tensor dw, long E, long nz) {
  // Small E (stem filters: E ~ 9.4K) caps the element-parallel grid at a
  // handful of blocks — with thousands of slabs that left >95% of the chip
  // idle on a multi-hundred-MB reduction (measured: the stem wgrad was
  // ~5.6 ms END-TO-END of which the main kernel was a fraction — this
  // reduce was the rest). Two-level: z-chunked partial pass into [zc, E],
  // then the final single-pass sum.
  const int eblocks = (int)std::min<long>(cdiv_l(E, 256 * 4), 2048);
  long zc = std::min<long>(nz / 8, 2048 / std::max(eblocks, 1));
  if (zc > 1) {
    auto tmp = at::empty({zc, E}, part.options());
    const long per = cdiv_l(nz, zc);
    zc = cdiv_l(nz, per);
    hipLaunchKernelGGL(wgrad_reduce_zchunk, dim3(eblocks, (unsigned)zc),
                       dim3(256), 0, cur_stream(), part.data_ptr<float>(),
                       tmp.data_ptr<float>(), E, (int)nz, (int)per);
    hipLaunchKernelGGL(wgrad_reduce_chunks, dim3(eblocks), dim3(256), 0,
                       cur_stream(), tmp.data_ptr<float>(),
                       dw.data_ptr<float>(), E, (int)zc);
    return;
  }
  hipLaunchKernelGGL(wgrad_reduce_chunks, dim3(eblocks), dim3(256), 0,
                     cur_stream(), part.data_ptr<float>(),
                     dw.data_ptr<float>(), E, (int)nz);
}

// asc/ash (optional): lazy-BN transform on the x operand (see fwd launcher)
void conv_wgrad_mfma_launch(at::Tensor x, at::Tensor dy, at::Tensor dw,
                            long R, long S, long stride, long pad,
                            at::Tensor asc, at::Tensor ash) {
  const float* asc_p = asc.defined() && asc.numel() ? asc.data_ptr<float>()
                                                    : nullptr;
  const float* ash_p = ash.defined() && ash.numel() ? ash.data_ptr<float>()
                                                    : nullptr;
  const int N = x.size(0), Hi = x.size(1), Wi = x.size(2), CI = x.size(3);
  const int Ho = dy.size(1), Wo = dy.size(2), KO = dy.size(3);
  const long M = (long)N * Ho * Wo;
  // 3x3/s1/p1 with power-of-two W (8..64): all-taps halo kernel.
  // Measured SLOWER than the tap-per-block kernel (413 vs ~280 us avg:
  // v236 VGPRs -> 2 waves/SIMD, and the 8-scalar-u16 b-fragment gather is
  // LDS/VALU-bound) — kept behind MI355X_WGRAD_HALO=1 for further work.
  static const bool halo_on = [] {
    const char* e = getenv("MI355X_WGRAD_HALO");
    return e && e[0] == '1';
  }();
  const bool halo = halo_on && !asc_p && R == 3 && S == 3 && stride == 1 && pad == 1 &&
                    Hi == Ho && Wi == Wo && Wo >= 8 && Wo <= 64 &&
                    (Wo & (Wo - 1)) == 0 && ((long)Ho * Wo) % 64 == 0;
  if (halo) {
    int wshift = 0;
    while ((1 << wshift) < Wo) ++wshift;
    const long blocks_xy = (KO / 64) * (CI / 64);
    long nchunks = std::min<long>(std::max<long>(512 / blocks_xy, 1),
                                  cdiv_l(M, 64));
    long m_per_chunk = cdiv_l(cdiv_l(M, nchunks), 64) * 64;
    nchunks = cdiv_l(M, m_per_chunk);
    const long E = (long)KO * CI * 9;
    at::Tensor part =
        nchunks > 1 ? at::empty({nchunks, E}, dw.options()) : dw;
    dim3 grid(KO / 64, CI / 64, (unsigned)nchunks);
    DISPATCH_16(x, T16, {
      hipLaunchKernelGGL(conv_wgrad_halo3<T16>, grid, dim3(256), 0,
                         cur_stream(), (const T16*)x.data_ptr(),
                         (const T16*)dy.data_ptr(), part.data_ptr<float>(),
                         N, Hi, Wi, CI, KO, wshift, m_per_chunk);
    });
    if (nchunks > 1) wgrad_reduce_launch(part, dw, E, nchunks);
    return;
  }
  // 3x3/s1/p1: s-grouped kernel (3 taps per stage) — MI355X_WGRAD_S3=0
  // falls back to the tap-per-block kernel
  static const bool s3_on = [] {
    const char* e = getenv("MI355X_WGRAD_S3");
    return !e || e[0] != '0';
  }();
  if (s3_on && R == 3 && S == 3 && stride == 1 && pad == 1 && Hi == Ho &&
      Wi == Wo) {
    static const long cap3 = [] {
      const char* e = getenv("MI355X_WGRAD_CHUNKS");
      return e ? atol(e) : 0L;
    }();
    int nchunks = (int)std::max<long>(std::min<long>(cdiv_l(M, 4096), 4096),
                                      1);
    if (cap3 > 0) nchunks = (int)std::min<long>(nchunks, cap3);
    long m_per_chunk = cdiv_l(cdiv_l(M, nchunks), WGM) * WGM;
    nchunks = (int)cdiv_l(M, m_per_chunk);
    static const long ktf = [] {  // A/B knob (r18: KT=64 90.98k vs
      const char* e = getenv("MI355X_WGRAD_S3_KT");  // KT=128 87.9k img/s)
      return e ? atol(e) : 0L;
    }();
    // KT=64 default even when KO%128==0: occupancy 3 vs 2 outweighs the
    // wider tile's staging amortization (measured +3.4% whole-bench)
    const int KT = (ktf == 128 && !asc_p) ? 128 : 64;  // scaled: KT=64 only
    dim3 grid(KO / KT, 3 * (CI / 64), nchunks);
    const long E = (long)KO * 9 * CI;
    at::Tensor part = nchunks > 1
                          ? at::empty({nchunks, E}, dw.options())
                          : dw;
    static const bool kminor = [] {  // default ON (+0.5% r18, interleaved
      const char* e = getenv("MI355X_S3_KMINOR");  // A/B x2); =0 disables
      return !e || e[0] != '0';
    }();
    DISPATCH_16(x, T16, {
      if (KT == 128)
        hipLaunchKernelGGL((conv_wgrad_mfma_s3<T16, 128>), grid, dim3(256),
                           0, cur_stream(), (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, m_per_chunk, nchunks);
      else if (kminor && !asc_p)
        hipLaunchKernelGGL((conv_wgrad_mfma_s3<T16, 64, false, true>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, m_per_chunk, nchunks);
      else if (asc_p)
        hipLaunchKernelGGL((conv_wgrad_mfma_s3<T16, 64, true>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), asc_p, ash_p,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, m_per_chunk, nchunks);
      else
        hipLaunchKernelGGL((conv_wgrad_mfma_s3<T16, 64>), grid, dim3(256),
                           0, cur_stream(), (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, m_per_chunk, nchunks);
    });
    if (nchunks > 1) wgrad_reduce_launch(part, dw, E, nchunks);
    return;
  }
  // KO%128 && CI%128 (notably every r50 1x1 bottleneck conv): 128x128
  // output tile, 4 acc tiles per wave — 2x the MFMA per staged byte of
  // the 64-wide-CI kernel (which measured ~283 TF on these shapes)
  static const bool t128_on = [] {
    const char* e = getenv("MI355X_WGRAD_T128");
    return !e || e[0] != '0';
  }();
  if (t128_on && KO % 128 == 0 && CI % 128 == 0) {
    int nchunks = (int)std::max<long>(std::min<long>(cdiv_l(M, 4096), 4096),
                                      1);
    long m_per_chunk = cdiv_l(cdiv_l(M, nchunks), WGM) * WGM;
    nchunks = (int)cdiv_l(M, m_per_chunk);
    dim3 grid(KO / 128, (unsigned)(R * S * (CI / 128)), nchunks);
    const long E = (long)KO * R * S * CI;
    at::Tensor part = nchunks > 1
                          ? at::empty({nchunks, E}, dw.options())
                          : dw;
    static const bool kmin128 = [] {  // default ON (+0.8% r50,
      const char* e = getenv("MI355X_T128_KMINOR");  // interleaved A/B x2)
      return !e || e[0] != '0';
    }();
    DISPATCH_16(x, T16, {
      if (asc_p)
        hipLaunchKernelGGL((conv_wgrad_mfma_t128<T16, true>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), asc_p, ash_p,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
      else if (kmin128)
        hipLaunchKernelGGL((conv_wgrad_mfma_t128<T16, false, true>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
      else
        hipLaunchKernelGGL((conv_wgrad_mfma_t128<T16>), grid, dim3(256), 0,
                           cur_stream(), (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
    });
    if (nchunks > 1) wgrad_reduce_launch(part, dw, E, nchunks);
    return;
  }
  const long blocks_xy = (KO / 64) * (R * S * (CI / 64));
  static const long cap = [] {  // ablation knob: MI355X_WGRAD_CHUNKS
    const char* e = getenv("MI355X_WGRAD_CHUNKS");
    return e ? atol(e) : 0L;
  }();
  // A/B on layer1 b256 (partial slabs): m/chunk 1216 -> 189us, 4096 ->
  // 137us, 16384 -> 327us: the lever is PER-CHUNK depth (~4096 m), which
  // also holds as M grows with batch (576-total-blocks regressed b1024)
  int nchunks = (int)std::max<long>(
      std::min<long>(cdiv_l(M, 4096), 4096), 1);
  if (cap > 0) nchunks = (int)std::min<long>(nchunks, cap);
  nchunks = std::max(nchunks, 1);
  long m_per_chunk = cdiv_l(cdiv_l(M, nchunks), WGM) * WGM;
  nchunks = (int)cdiv_l(M, m_per_chunk);
  const int KT = KO % 128 == 0 ? 128 : 64;
  dim3 grid(KO / KT, R * S * (CI / 64), nchunks);
  const long E = (long)KO * R * S * CI;
  at::Tensor part = nchunks > 1
                        ? at::empty({nchunks, E}, dw.options())
                        : dw;
  static const bool kming = [] {  // shares the s3 knob family
    const char* e = getenv("MI355X_WGRAD_KMINOR");
    return !e || e[0] != '0';
  }();
  DISPATCH_16(x, T16, {
    if (KT == 128) {
      if (asc_p)
        hipLaunchKernelGGL((conv_wgrad_mfma_kernel<T16, 128, true>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), asc_p, ash_p,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
      else if (kming)
        hipLaunchKernelGGL((conv_wgrad_mfma_kernel<T16, 128, false, true>),
                           grid, dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
      else
        hipLaunchKernelGGL((conv_wgrad_mfma_kernel<T16, 128>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
    } else {
      if (asc_p)
        hipLaunchKernelGGL((conv_wgrad_mfma_kernel<T16, 64, true>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), asc_p, ash_p,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
      else if (kming)
        hipLaunchKernelGGL((conv_wgrad_mfma_kernel<T16, 64, false, true>),
                           grid, dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
      else
        hipLaunchKernelGGL((conv_wgrad_mfma_kernel<T16, 64>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr, nullptr,
                           part.data_ptr<float>(),
                           N, Hi, Wi, CI, KO, Ho, Wo, (int)R, (int)S,
                           (int)stride, (int)pad, m_per_chunk, nchunks);
    }
  });
  if (nchunks > 1) wgrad_reduce_launch(part, dw, E, nchunks);
}

// GENC fwd: wpad = [KO, KGP] zero-padded (cast_permute_krsc_pad)
void conv_fwd_mfma_genc_launch(at::Tensor x, at::Tensor wpad, at::Tensor bias,
                               at::Tensor y, long R, long S, long stride,
                               long pad, long act) {
  const int N = x.size(0), Hi = x.size(1), Wi = x.size(2), CI = x.size(3);
  const int KO = wpad.size(0);
  const long KGP = wpad.size(1);
  const int Ho = y.size(1), Wo = y.size(2);
  const long M = (long)N * Ho * Wo;
  dim3 grid((unsigned)cdiv_l(M, BM), KO / BN);
  const int has_bias = bias.numel() > 0;
  at::Tensor zp = conv_zero_page(x);
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL((conv_gather_gemm<T16, false, true, 2>), grid,
                       dim3(256), 0, cur_stream(), (const T16*)x.data_ptr(),
                       (const T16*)wpad.data_ptr(),
                       has_bias ? bias.data_ptr<float>() : nullptr,
                       (const T16*)zp.data_ptr(), nullptr, nullptr,
                       nullptr,
                       (T16*)y.data_ptr(), nullptr, N, Hi, Wi, CI, KO, Ho,
                       Wo, (int)R, (int)S, (int)stride, (int)pad, KGP, 0,
                       (int)act, has_bias);
  });
}

// fwd: in = x[N,Hi,Wi,CI], wgt = w16 [KO, R,S,CI] row-major.
// stats (optional, zeroed [2,C]): conv->BN fusion — the epilogue emits
// per-block (sum,sumsq) partials to a slab and conv_stats_reduce folds
// them, replacing BN's separate full-tensor bn_stats pass.
// asc/ash (optional [CI] fp32): lazy-BN A-side transform — the x operand
// is normalized+ReLU'd on load (z = relu(x*asc+ash)); the BN apply pass
// and its output tensor disappear.
void conv_fwd_mfma_launch(at::Tensor x, at::Tensor w, at::Tensor bias,
                          at::Tensor y, long stride, long pad, long act,
                          at::Tensor stats, at::Tensor asc, at::Tensor ash) {
  const float* asc_p = asc.defined() && asc.numel() ? asc.data_ptr<float>()
                                                    : nullptr;
  const float* ash_p = ash.defined() && ash.numel() ? ash.data_ptr<float>()
                                                    : nullptr;
  const int N = x.size(0), Hi = x.size(1), Wi = x.size(2), CI = x.size(3);
  const int KO = w.size(0), R = w.size(1), S = w.size(2);
  const int Ho = y.size(1), Wo = y.size(2);
  const long M = (long)N * Ho * Wo;
  const int has_bias = bias.numel() > 0;
  // A/B (b1024 layer shapes): glds 597/685/561 TF vs register staging
  // 632/692/665 — glds LOSES at this kernel's 5-blocks/CU occupancy, as
  // the guide's regime gate predicts (pays only at ~1 block/CU, vgpr>200).
  // Kept behind MI355X_CONV_GLDS=1 as a documented negative result.
  static const int glds_mode = [] {
    const char* e = getenv("MI355X_CONV_GLDS");
    return e ? atoi(e) : 0;
  }();
  if (glds_mode && !asc_p && !(stats.defined() && stats.numel() > 0)) {
    at::Tensor zp = conv_zero_page(x);
    const bool wide = KO % 128 == 0 && cdiv_l(M, BM) * (KO / 128) >= 1024;
    dim3 grid((unsigned)cdiv_l(M, BM), KO / (wide ? 128 : 64));
    DISPATCH_16(x, T16, {
      if (wide)
        hipLaunchKernelGGL((conv_fwd_glds<T16, 4>), grid, dim3(256), 0,
                           cur_stream(), (const T16*)x.data_ptr(),
                           (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           (const T16*)zp.data_ptr(), (T16*)y.data_ptr(), N,
                           Hi, Wi, CI, KO, Ho, Wo, R, S, (int)stride,
                           (int)pad, (int)act, has_bias);
      else
        hipLaunchKernelGGL((conv_fwd_glds<T16, 2>), grid, dim3(256), 0,
                           cur_stream(), (const T16*)x.data_ptr(),
                           (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           (const T16*)zp.data_ptr(), (T16*)y.data_ptr(), N,
                           Hi, Wi, CI, KO, Ho, Wo, R, S, (int)stride,
                           (int)pad, (int)act, has_bias);
    });
    return;
  }
  // 3x3/s1/p1 patch kernel: one geometric LDS patch per c-chunk serves
  // all 9 taps (MI355X_CONV_PATCH=0 falls back to the per-tap gather)
  static const bool patch_on = [] {
    const char* e = getenv("MI355X_CONV_PATCH");
    return !e || e[0] != '0';
  }();
  if (patch_on && R == 3 && S == 3 && stride == 1 && pad == 1 && Ho == Hi &&
      Wo == Wi && Wo <= 64) {
    // rows spanned by a 128-output tile starting mid-row, +2 halo rows:
    // 128/Wo+3 was ONE short for Wo in {56,28,14,7} (tap r=2 then read
    // past the patch -> diverging ResNet-50 training)
    const int NR = (Wo + 126) / Wo + 3;
    const size_t smem = ((size_t)NR * (Wi + 2) * PCS + PCS + 64 * LDK) * 2;
    dim3 pgrid_((unsigned)cdiv_l(M, 128), KO / 64);
    at::Tensor pslab;
    float* pslab_ptr = nullptr;
    if (stats.defined() && stats.numel() > 0) {
      pslab = at::empty({(long)pgrid_.y * pgrid_.x * 128},
                        x.options().dtype(at::kFloat));
      pslab_ptr = pslab.data_ptr<float>();
    }
    DISPATCH_16(x, T16, {
      if (asc_p)
        hipLaunchKernelGGL((conv_patch_gemm<T16, false, false, true>),
                           pgrid_, dim3(256), smem, cur_stream(),
                           (const T16*)x.data_ptr(), (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           nullptr, asc_p, ash_p, (T16*)y.data_ptr(),
                           pslab_ptr, N, Hi, Wi, CI, KO, (long)R * S * CI,
                           (long)CI, (int)act, has_bias, NR);
      else
        hipLaunchKernelGGL((conv_patch_gemm<T16, false>), pgrid_, dim3(256),
                           smem, cur_stream(), (const T16*)x.data_ptr(),
                           (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           nullptr, nullptr, nullptr, (T16*)y.data_ptr(),
                           pslab_ptr, N, Hi, Wi, CI, KO, (long)R * S * CI,
                           (long)CI, (int)act, has_bias, NR);
    });
    if (pslab_ptr)
      stats_slab_reduce(pslab, stats, pgrid_.x, 128, KO);
    return;
  }
  // BN=128 halves barriers per MFMA but also halves the grid — only use
  // it when M is large enough to keep the chip full at BM=128 tiles
  const bool wide = KO % 128 == 0 && cdiv_l(M, BM) * (KO / 128) >= 1024;
  dim3 grid((unsigned)cdiv_l(M, BM), KO / (wide ? 128 : 64));
  at::Tensor zp2 = conv_zero_page(x);
  const int bnt2 = 2 * (wide ? 128 : 64);
  at::Tensor slab;
  float* stats_slab_ptr = nullptr;
  if (stats.defined() && stats.numel() > 0) {
    slab = at::empty({(long)grid.y * grid.x * bnt2},
                     x.options().dtype(at::kFloat));
    stats_slab_ptr = slab.data_ptr<float>();
  }
  DISPATCH_16(x, T16, {
    if (wide) {
      if (asc_p)
        hipLaunchKernelGGL((conv_gather_gemm<T16, false, false, 4, false,
                                             true>),
                           grid, dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(), (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           (const T16*)zp2.data_ptr(), nullptr, asc_p, ash_p,
                           (T16*)y.data_ptr(), stats_slab_ptr, N, Hi, Wi, CI,
                           KO, Ho, Wo, R, S, (int)stride, (int)pad,
                           (long)R * S * CI, (long)CI, (int)act, has_bias);
      else
        hipLaunchKernelGGL((conv_gather_gemm<T16, false, false, 4>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(), (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           (const T16*)zp2.data_ptr(), nullptr, nullptr,
                           nullptr,
                           (T16*)y.data_ptr(), stats_slab_ptr, N, Hi, Wi, CI,
                           KO, Ho, Wo, R, S, (int)stride, (int)pad,
                           (long)R * S * CI, (long)CI, (int)act, has_bias);
    } else {
      if (asc_p)
        hipLaunchKernelGGL((conv_gather_gemm<T16, false, false, 2, false,
                                             true>),
                           grid, dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(), (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           (const T16*)zp2.data_ptr(), nullptr, asc_p, ash_p,
                           (T16*)y.data_ptr(), stats_slab_ptr, N, Hi, Wi, CI,
                           KO, Ho, Wo, R, S, (int)stride, (int)pad,
                           (long)R * S * CI, (long)CI, (int)act, has_bias);
      else
        hipLaunchKernelGGL((conv_gather_gemm<T16, false, false, 2>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(), (const T16*)w.data_ptr(),
                           has_bias ? bias.data_ptr<float>() : nullptr,
                           (const T16*)zp2.data_ptr(), nullptr, nullptr,
                           nullptr,
                           (T16*)y.data_ptr(), stats_slab_ptr, N, Hi, Wi, CI,
                           KO, Ho, Wo, R, S, (int)stride, (int)pad,
                           (long)R * S * CI, (long)CI, (int)act, has_bias);
    }
  });
  if (stats_slab_ptr)
    stats_slab_reduce(slab, stats, grid.x, bnt2, KO);
}

// dgrad: in = dy[N,P,Q,KO], wflip = [R,S,CI,KO] (w[k,R-1-r,S-1-s,c]),
// out = dx[N,H,W,CI]. addin (optional): dx = dgrad(dy) + addin — the
// residual-junction gradient fused into the epilogue (compile-time
// instantiation; the no-addin kernels are untouched).
void conv_dgrad_mfma_launch(at::Tensor dy, at::Tensor wflip, at::Tensor dx,
                            long R, long S, long stride, long pad,
                            at::Tensor addin) {
  const int N = dy.size(0), P = dy.size(1), Q = dy.size(2), KO = dy.size(3);
  const int H = dx.size(1), W = dx.size(2), CI = dx.size(3);
  const long M = (long)N * H * W;
  const bool has_add = addin.defined() && addin.numel() > 0;
  // stride-2: parity-specialized kernel (4 classes in blockIdx.z), ~2.6x
  // fewer k-steps for 3x3/s2 and 4x for the 1x1 downsamples
  static const bool s2_on = [] {
    const char* e = getenv("MI355X_DGRAD_S2");
    return !e || e[0] != '0';
  }();
  if (s2_on && stride == 2 && pad <= 8) {
    const long Mmax = (long)N * ((H + 1) / 2) * ((W + 1) / 2);
    const bool wide =
        CI % 128 == 0 && cdiv_l(Mmax, BM) * (CI / 128) * 4 >= 1024;
    const int BNTs = wide ? 128 : 64;
    dim3 grid((unsigned)cdiv_l(Mmax, BM), CI / BNTs, 4);
    at::Tensor zp = conv_zero_page(dy);
    DISPATCH_16(dy, T16, {
      const T16* add_p = has_add ? (const T16*)addin.data_ptr() : nullptr;
      if (wide) {
        if (has_add)
          hipLaunchKernelGGL((conv_dgrad_s2_gemm<T16, 4, true>), grid,
                             dim3(256), 0, cur_stream(),
                             (const T16*)dy.data_ptr(),
                             (const T16*)wflip.data_ptr(),
                             (const T16*)zp.data_ptr(), add_p,
                             (T16*)dx.data_ptr(), N, P, Q, KO, CI, H, W,
                             (int)R, (int)S, (int)pad);
        else
          hipLaunchKernelGGL((conv_dgrad_s2_gemm<T16, 4>), grid, dim3(256),
                             0, cur_stream(), (const T16*)dy.data_ptr(),
                             (const T16*)wflip.data_ptr(),
                             (const T16*)zp.data_ptr(), nullptr,
                             (T16*)dx.data_ptr(), N, P, Q, KO, CI, H, W,
                             (int)R, (int)S, (int)pad);
      } else {
        if (has_add)
          hipLaunchKernelGGL((conv_dgrad_s2_gemm<T16, 2, true>), grid,
                             dim3(256), 0, cur_stream(),
                             (const T16*)dy.data_ptr(),
                             (const T16*)wflip.data_ptr(),
                             (const T16*)zp.data_ptr(), add_p,
                             (T16*)dx.data_ptr(), N, P, Q, KO, CI, H, W,
                             (int)R, (int)S, (int)pad);
        else
          hipLaunchKernelGGL((conv_dgrad_s2_gemm<T16, 2>), grid, dim3(256),
                             0, cur_stream(), (const T16*)dy.data_ptr(),
                             (const T16*)wflip.data_ptr(),
                             (const T16*)zp.data_ptr(), nullptr,
                             (T16*)dx.data_ptr(), N, P, Q, KO, CI, H, W,
                             (int)R, (int)S, (int)pad);
      }
    });
    return;
  }
  static const bool patch_on = [] {
    const char* e = getenv("MI355X_CONV_PATCH");
    return !e || e[0] != '0';
  }();
  if (patch_on && R == 3 && S == 3 && stride == 1 && pad == 1 && P == H &&
      Q == W && W <= 64) {
    const int NR = (W + 126) / W + 3;  // see fwd launcher comment
    const size_t smem = ((size_t)NR * (W + 2) * PCS + PCS + 64 * LDK) * 2;
    dim3 pgrid_((unsigned)cdiv_l(M, 128), CI / 64);
    DISPATCH_16(dy, T16, {
      if (has_add)
        hipLaunchKernelGGL((conv_patch_gemm<T16, true, true>), pgrid_,
                           dim3(256), smem, cur_stream(),
                           (const T16*)dy.data_ptr(),
                           (const T16*)wflip.data_ptr(), nullptr,
                           (const T16*)addin.data_ptr(), nullptr, nullptr,
                           (T16*)dx.data_ptr(), nullptr, N, P, Q, KO, CI,
                           (long)KO, (long)CI * KO, 0, 0, NR);
      else
        hipLaunchKernelGGL((conv_patch_gemm<T16, true>), pgrid_, dim3(256),
                           smem, cur_stream(), (const T16*)dy.data_ptr(),
                           (const T16*)wflip.data_ptr(), nullptr, nullptr,
                           nullptr, nullptr,
                           (T16*)dx.data_ptr(), nullptr, N, P, Q, KO, CI,
                           (long)KO, (long)CI * KO, 0, 0, NR);
    });
    return;
  }
  const bool wide = CI % 128 == 0 && cdiv_l(M, BM) * (CI / 128) >= 1024;
  dim3 grid((unsigned)cdiv_l(M, BM), CI / (wide ? 128 : 64));
  at::Tensor zp = conv_zero_page(dy);
  DISPATCH_16(dy, T16, {
    const T16* add_p = has_add ? (const T16*)addin.data_ptr() : nullptr;
    if (wide) {
      if (has_add)
        hipLaunchKernelGGL((conv_gather_gemm<T16, true, false, 4, true>),
                           grid, dim3(256), 0, cur_stream(),
                           (const T16*)dy.data_ptr(),
                           (const T16*)wflip.data_ptr(), nullptr,
                           (const T16*)zp.data_ptr(), add_p, nullptr,
                           nullptr,
                           (T16*)dx.data_ptr(), nullptr, N, P, Q, KO, CI, H,
                           W, (int)R, (int)S, (int)stride, (int)pad,
                           (long)KO, (long)CI * KO, 0, 0);
      else
        hipLaunchKernelGGL((conv_gather_gemm<T16, true, false, 4>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)dy.data_ptr(),
                           (const T16*)wflip.data_ptr(), nullptr,
                           (const T16*)zp.data_ptr(), nullptr, nullptr,
                           nullptr,
                           (T16*)dx.data_ptr(), nullptr, N, P, Q, KO, CI, H,
                           W, (int)R, (int)S, (int)stride, (int)pad,
                           (long)KO, (long)CI * KO, 0, 0);
    } else {
      if (has_add)
        hipLaunchKernelGGL((conv_gather_gemm<T16, true, false, 2, true>),
                           grid, dim3(256), 0, cur_stream(),
                           (const T16*)dy.data_ptr(),
                           (const T16*)wflip.data_ptr(), nullptr,
                           (const T16*)zp.data_ptr(), add_p, nullptr,
                           nullptr,
                           (T16*)dx.data_ptr(), nullptr, N, P, Q, KO, CI, H,
                           W, (int)R, (int)S, (int)stride, (int)pad,
                           (long)KO, (long)CI * KO, 0, 0);
      else
        hipLaunchKernelGGL((conv_gather_gemm<T16, true, false, 2>), grid,
                           dim3(256), 0, cur_stream(),
                           (const T16*)dy.data_ptr(),
                           (const T16*)wflip.data_ptr(), nullptr,
                           (const T16*)zp.data_ptr(), nullptr, nullptr,
                           nullptr,
                           (T16*)dx.data_ptr(), nullptr, N, P, Q, KO, CI, H,
                           W, (int)R, (int)S, (int)stride, (int)pad,
                           (long)KO, (long)CI * KO, 0, 0);
    }
  });
}
