// BatchNorm over NHWC (SURVEY.md N8 / BASELINE configs 2-5):
//  - bn_stats: per-channel (sum, sumsq) partials in one pass (f32 accum) —
//    the same partials SyncBN all-reduces across ranks;
//  - bn_apply: fused normalize + affine + optional residual add + ReLU;
//  - bn_bwd_reduce: per-channel (sum dy*xhat, sum dy);
//  - bn_bwd_dx: dx from the reduced terms.
// Channel-minor layout makes every access coalesced along C; reductions are
// per-thread private over an M-chunk, then one atomicAdd per channel per
// chunk (guide Guideline 12).
#include "common.h"

namespace {

template <typename T16>
__global__ void bn_stats_kernel(const T16* __restrict__ x,
                                float* __restrict__ out,  // [2,C]
                                long M, int C, long m_per_chunk) {
  const int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const long m0 = (long)blockIdx.x * m_per_chunk;
  const long m1 = min(M, m0 + m_per_chunk);
  float s = 0.f, ss = 0.f;
  for (long m = m0; m < m1; ++m) {
    const float v = F16<T16>::to_f32(x[m * C + c]);
    s += v;
    ss += v * v;
  }
  if (gridDim.x == 1) {
    out[c] = s;
    out[C + c] = ss;
  } else {
    atomicAdd(out + c, s);
    atomicAdd(out + C + c, ss);
  }
}

template <typename T16>
__global__ void bn_apply_kernel(const T16* __restrict__ x,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                const T16* __restrict__ res,
                                T16* __restrict__ y, long M, int C, int act,
                                int has_res) {
  const long total = M * C;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const int c = (int)(t % C);
    const float sc = gamma[c] * invstd[c];
    float v = (F16<T16>::to_f32(x[t]) - mean[c]) * sc + beta[c];
    if (has_res) v += F16<T16>::to_f32(res[t]);
    if (act == 1) v = fmaxf(v, 0.f);
    y[t] = F16<T16>::from_f32(v);
  }
}

// y != null: the forward ended in ReLU; mask dy by y>0 inline (saves a
// standalone relu_bwd pass over the activation tensor)
template <typename T16>
__global__ void bn_bwd_reduce_kernel(const T16* __restrict__ x,
                                     const T16* __restrict__ dy,
                                     const T16* __restrict__ y,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ out,  // [2,C]
                                     long M, int C, long m_per_chunk) {
  const int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const long m0 = (long)blockIdx.x * m_per_chunk;
  const long m1 = min(M, m0 + m_per_chunk);
  const float mu = mean[c], is = invstd[c];
  float s_dyx = 0.f, s_dy = 0.f;
  for (long m = m0; m < m1; ++m) {
    float d = F16<T16>::to_f32(dy[m * C + c]);
    if (y && F16<T16>::to_f32(y[m * C + c]) <= 0.f) d = 0.f;
    const float xh = (F16<T16>::to_f32(x[m * C + c]) - mu) * is;
    s_dyx += d * xh;
    s_dy += d;
  }
  if (gridDim.x == 1) {
    out[c] = s_dyx;
    out[C + c] = s_dy;
  } else {
    atomicAdd(out + c, s_dyx);
    atomicAdd(out + C + c, s_dy);
  }
}

// y != null: mask dy by y>0 inline; dres != null: also emit the masked dy
// (the residual-branch gradient) so no separate relu_bwd pass is needed.
template <typename T16>
__global__ void bn_bwd_dx_kernel(const T16* __restrict__ x,
                                 const T16* __restrict__ dy,
                                 const T16* __restrict__ y,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ dgamma,
                                 const float* __restrict__ dbeta,
                                 T16* __restrict__ dx,
                                 T16* __restrict__ dres, long M, int C,
                                 float inv_m) {
  const long total = M * C;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const int c = (int)(t % C);
    const float mu = mean[c], is = invstd[c];
    float d = F16<T16>::to_f32(dy[t]);
    if (y && F16<T16>::to_f32(y[t]) <= 0.f) d = 0.f;
    if (dres) dres[t] = F16<T16>::from_f32(d);
    const float xh = (F16<T16>::to_f32(x[t]) - mu) * is;
    const float v =
        gamma[c] * is * (d - dbeta[c] * inv_m - xh * dgamma[c] * inv_m);
    dx[t] = F16<T16>::from_f32(v);
  }
}

// ---- fast reduction path: C % 8 == 0 && 2048 % C == 0 -------------------
// Linear sweep of the [M,C] tensor in 16 B/lane vectors (fully coalesced,
// every thread active); each thread owns a FIXED 8-channel group (c0 =
// (tid*8) % C, invariant because 2048 % C == 0), accumulates 8 partial
// pairs in registers, combines within the block through an LDS f32
// accumulator, then one global atomicAdd per channel per block.
// BWD=false: (sum x, sum x^2). BWD=true: (sum dy*xhat, sum dy), with the
// fused ReLU mask from y when given.
// asc/ash (BWD only, nullable): recompute the ReLU mask as
// (x*asc[c] + ash[c]) > 0 — the lazy-BN path (apply fused into the
// consuming conv) never materializes y or a mask
template <typename T16, bool BWD, bool RC = false>
__global__ __launch_bounds__(256) void bn_reduce_fast(
    const T16* __restrict__ x, const T16* __restrict__ dy,
    const unsigned char* __restrict__ msk, const float* __restrict__ mean,
    const float* __restrict__ invstd,
    const float* __restrict__ asc, const float* __restrict__ ash,
    float* __restrict__ out,  // [2,C]
    long E, int C, long e_per_block) {
  extern __shared__ __attribute__((aligned(16))) float lsum[];  // [2*C]
  const int tid = threadIdx.x;
  const int c0 = (tid * 8) % C;
  float s0[8] = {}, s1[8] = {};
  float mu[8], is[8], ac[8], ah[8];
  if (BWD) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      mu[u] = mean[c0 + u];
      is[u] = invstd[c0 + u];
      if (RC) {
        ac[u] = asc[c0 + u];
        ah[u] = ash[c0 + u];
      }
    }
  }
  const long e0 = (long)blockIdx.x * e_per_block;
  const long e1 = min(E, e0 + e_per_block);
  constexpr long ESTEP = 256 * 8;
  // body for one 16B group (channel group is e-invariant: strides are
  // multiples of 2048 and 2048 % C == 0)
  auto body = [&](long e) {
    if (!BWD) {
      const short8 vx = *reinterpret_cast<const short8*>(x + e);
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const float v = s16_to_f32<T16>(vx[u]);
        s0[u] += v;
        s1[u] += v * v;
      }
    } else {
      const short8 vx = *reinterpret_cast<const short8*>(x + e);
      const short8 vd = *reinterpret_cast<const short8*>(dy + e);
      const unsigned mb = msk ? msk[e >> 3] : 0xffu;
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float d = s16_to_f32<T16>(vd[u]);
        const float xv = s16_to_f32<T16>(vx[u]);
        if (RC) {
          if (xv * ac[u] + ah[u] <= 0.f) d = 0.f;
        } else if (!((mb >> u) & 1)) {
          d = 0.f;
        }
        const float xh = (xv - mu[u]) * is[u];
        s0[u] += d * xh;
        s1[u] += d;
      }
    }
  };
  // 4 sub-streams per iteration: one outstanding 16B load per thread caps
  // at ~2.4 TB/s by Little's law (512 blocks x 256 thr x 16 B in flight);
  // four batched loads quadruple the in-flight bytes
  long e = e0 + (long)tid * 8;
  for (; e + 3 * ESTEP < e1; e += 4 * ESTEP) {
#pragma unroll
    for (int v = 0; v < 4; ++v) body(e + v * ESTEP);
  }
  for (; e < e1; e += ESTEP) body(e);
  // wave-level pre-reduction: lanes whose 8-channel group repeats within
  // the wave (group stride C/8 lanes) fold via shfl_xor before touching
  // LDS — cuts LDS-atomic collisions by 512/C
  const int gstride = C / 8;  // lanes between same-group threads
  for (int off = 32; off >= gstride && off >= 1; off >>= 1) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      s0[u] += __shfl_xor(s0[u], off, kWave);
      s1[u] += __shfl_xor(s1[u], off, kWave);
    }
  }
  const bool leader = (threadIdx.x & 63) < gstride || gstride >= 64;
  for (int c = tid; c < 2 * C; c += 256) lsum[c] = 0.f;
  __syncthreads();
  if (leader) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      atomicAdd(lsum + c0 + u, s0[u]);
      atomicAdd(lsum + C + c0 + u, s1[u]);
    }
  }
  __syncthreads();
  for (int c = tid; c < 2 * C; c += 256)
    if (gridDim.x == 1)
      out[c] = lsum[c];
    else
      atomicAdd(out + c, lsum[c]);
}

// vectorized elementwise BN apply / backward-dx: 16 B/lane linear sweep,
// per-thread FIXED 8-channel group (same condition as bn_reduce_fast)
template <typename T16>
__global__ __launch_bounds__(256) void bn_apply_fast(
    const T16* __restrict__ x, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, const T16* __restrict__ res,
    T16* __restrict__ y, unsigned char* __restrict__ msk, long E, int C,
    int act) {
  const int c0 = ((blockIdx.x * blockDim.x + threadIdx.x) * 8) % C;
  float sc[8], sh[8];
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    sc[u] = gamma[c0 + u] * invstd[c0 + u];
    sh[u] = beta[c0 + u] - mean[c0 + u] * sc[u];
  }
  const long stride = (long)gridDim.x * blockDim.x * 8;
  auto body = [&](long e) {
    const short8 vx = *reinterpret_cast<const short8*>(x + e);
    short8 vr = {};
    if (res) vr = *reinterpret_cast<const short8*>(res + e);
    short8 o;
    unsigned mb = 0;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      float v = s16_to_f32<T16>(vx[u]) * sc[u] + sh[u];
      if (res) v += s16_to_f32<T16>(vr[u]);
      if (act == 1) v = fmaxf(v, 0.f);
      o[u] = f32_to_s16<T16>(v);
      // mask bit mirrors the bf16-rounded "y > 0" test the backward used
      // to make by re-reading y (1/16 the bytes)
      if (s16_to_f32<T16>(o[u]) > 0.f) mb |= 1u << u;
    }
    *reinterpret_cast<short8*>(y + e) = o;
    if (msk) msk[e >> 3] = (unsigned char)mb;
  };
  // two batched sub-streams per iteration (same in-flight-bytes rationale
  // as bn_reduce_fast; the channel group is stride-invariant)
  long e = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  for (; e + stride < E; e += 2 * stride) {
    body(e);
    body(e + stride);
  }
  for (; e < E; e += stride) body(e);
}

template <typename T16, bool RC = false>
__global__ __launch_bounds__(256) void bn_bwd_dx_fast(
    const T16* __restrict__ x, const T16* __restrict__ dy,
    const unsigned char* __restrict__ msk, const float* __restrict__ mean,
    const float* __restrict__ invstd, const float* __restrict__ gamma,
    const float* __restrict__ dgamma, const float* __restrict__ dbeta,
    const float* __restrict__ asc, const float* __restrict__ ash,
    T16* __restrict__ dx, T16* __restrict__ dres, long E, int C,
    float inv_m) {
  const int c0 = ((blockIdx.x * blockDim.x + threadIdx.x) * 8) % C;
  float mu[8], is[8], g_[8], a_[8], b_[8], ac[8], ah[8];
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    mu[u] = mean[c0 + u];
    is[u] = invstd[c0 + u];
    g_[u] = gamma[c0 + u] * is[u];
    a_[u] = dgamma[c0 + u] * inv_m;
    b_[u] = dbeta[c0 + u] * inv_m;
    if (RC) {
      ac[u] = asc[c0 + u];
      ah[u] = ash[c0 + u];
    }
  }
  const long stride = (long)gridDim.x * blockDim.x * 8;
  auto body = [&](long e) {
    const short8 vx = *reinterpret_cast<const short8*>(x + e);
    const short8 vd = *reinterpret_cast<const short8*>(dy + e);
    const unsigned mb = msk ? msk[e >> 3] : 0xffu;
    short8 odx, ods;
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      float d = s16_to_f32<T16>(vd[u]);
      const float xv = s16_to_f32<T16>(vx[u]);
      if (RC) {
        if (xv * ac[u] + ah[u] <= 0.f) d = 0.f;
      } else if (!((mb >> u) & 1)) {
        d = 0.f;
      }
      if (dres) ods[u] = f32_to_s16<T16>(d);
      const float xh = (xv - mu[u]) * is[u];
      odx[u] = f32_to_s16<T16>(g_[u] * (d - b_[u] - xh * a_[u]));
    }
    *reinterpret_cast<short8*>(dx + e) = odx;
    if (dres) *reinterpret_cast<short8*>(dres + e) = ods;
  };
  long e = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  for (; e + stride < E; e += 2 * stride) {
    body(e);
    body(e + stride);
  }
  for (; e < E; e += stride) body(e);
}

// one tiny kernel replacing the Python stats glue (mean/var/invstd/
// running-stat updates were ~8 torch launches per BN layer per step)
__global__ void bn_finalize_kernel(const float* __restrict__ stats,  // [2,C]
                                   float* __restrict__ out,  // [2,C] mean,invstd
                                   float* __restrict__ rmean,
                                   float* __restrict__ rvar, int C,
                                   float inv_m, float unbias, float momentum,
                                   float eps) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float mean = stats[c] * inv_m;
  const float var = stats[C + c] * inv_m - mean * mean;
  out[c] = mean;
  out[C + c] = rsqrtf(var + eps);
  if (rmean) {
    rmean[c] = rmean[c] * (1.f - momentum) + mean * momentum;
    rvar[c] = rvar[c] * (1.f - momentum) + var * unbias * momentum;
  }
}

inline bool bn_fast_ok(long M, int C) {
  return C % 8 == 0 && 2048 % C == 0;
}

inline dim3 bn_fast_grid(long E, long& e_per_block) {
  static const long cap = [] {  // ablation knob
    const char* e = getenv("MI355X_BN_BLOCKS");
    return e ? atol(e) : 512L;  // A/B: 512 blocks 3.4/5.0 TB/s vs 1024 2.9/4.5
  }();
  long blocks = std::min<long>(cdiv_l(E, 2048), cap);
  e_per_block = cdiv_l(cdiv_l(E, blocks), 2048) * 2048;
  blocks = cdiv_l(E, e_per_block);
  return dim3((unsigned)blocks);
}

inline dim3 chan_grid(long M, int C, long& m_per_chunk, int block = 256) {
  // enough (chunk, channel-block) pairs to fill 256 CUs
  const int cblocks = cdiv_i(C, block);
  long chunks = std::min<long>(cdiv_l(M, 1024), std::max(1, 2048 / cblocks));
  chunks = std::max<long>(chunks, 1);
  m_per_chunk = cdiv_l(M, chunks);
  return dim3((unsigned)chunks, cblocks);
}

inline int ew_grid2(long n, int block = 256) {
  return (int)std::min<long>(cdiv_l(n, block), 4096);
}

}  // namespace

at::Tensor bn_finalize(at::Tensor stats, at::Tensor running_mean,
                       at::Tensor running_var, double m_total,
                       double momentum, double eps) {
  const int C = stats.size(1);
  auto out = at::empty_like(stats);
  const bool has_run = running_mean.numel() > 0;
  const double unbias = m_total / std::max(m_total - 1.0, 1.0);
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(cdiv_i(C, 256)), dim3(256), 0,
                     cur_stream(), stats.data_ptr<float>(),
                     out.data_ptr<float>(),
                     has_run ? running_mean.data_ptr<float>() : nullptr,
                     has_run ? running_var.data_ptr<float>() : nullptr, C,
                     (float)(1.0 / m_total), (float)unbias, (float)momentum,
                     (float)eps);
  return out;
}

at::Tensor bn_stats(at::Tensor x) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_16BIT(x);
  const int C = x.size(-1);
  const long M = x.numel() / C;
  if (bn_fast_ok(M, C)) {
    const long E = M * (long)C;
    long e_per_block;
    dim3 grid = bn_fast_grid(E, e_per_block);
    auto out = grid.x == 1 ? at::empty({2, C}, x.options().dtype(at::kFloat))
                           : at::zeros({2, C}, x.options().dtype(at::kFloat));
    DISPATCH_16(x, T16, {
      hipLaunchKernelGGL((bn_reduce_fast<T16, false>), grid, dim3(256),
                         2 * C * sizeof(float), cur_stream(),
                         (const T16*)x.data_ptr(), nullptr, nullptr, nullptr,
                         nullptr, nullptr, nullptr, out.data_ptr<float>(), E,
                         C, e_per_block);
    });
    return out;
  }
  long m_per_chunk;
  dim3 grid = chan_grid(M, C, m_per_chunk);
  auto out = grid.x == 1 ? at::empty({2, C}, x.options().dtype(at::kFloat))
                         : at::zeros({2, C}, x.options().dtype(at::kFloat));
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(bn_stats_kernel<T16>, grid, dim3(256), 0, cur_stream(),
                       (const T16*)x.data_ptr(), out.data_ptr<float>(), M, C,
                       m_per_chunk);
  });
  return out;
}

// returns {y, mask}: mask is a 1-bit-per-element "y > 0" map (u8, one
// byte per 8 channel-elements) when act==relu AND want_mask (training)
// on the fast path — the backward reads it instead of re-reading y (1/16
// the bytes). Empty tensor otherwise (eval callers skip the allocation
// and mask stores entirely; the fallback kernels keep y-based masking).
std::vector<at::Tensor> bn_apply(at::Tensor x, at::Tensor mean,
                                 at::Tensor invstd, at::Tensor gamma,
                                 at::Tensor beta, at::Tensor res, long act,
                                 bool want_mask) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_16BIT(x);
  const int C = x.size(-1);
  const long M = x.numel() / C;
  auto y = at::empty_like(x);
  const int has_res = res.numel() > 0;
  if (bn_fast_ok(M, C)) {
    const long E = M * (long)C;
    at::Tensor mask = (act == 1 && want_mask)
                          ? at::empty({E / 8}, x.options().dtype(at::kByte))
                          : at::Tensor();
    const int grid = (int)std::min<long>(cdiv_l(E, 256 * 8), 2048);
    DISPATCH_16(x, T16, {
      hipLaunchKernelGGL(bn_apply_fast<T16>, dim3(grid), dim3(256), 0,
                         cur_stream(), (const T16*)x.data_ptr(),
                         mean.data_ptr<float>(), invstd.data_ptr<float>(),
                         gamma.data_ptr<float>(), beta.data_ptr<float>(),
                         has_res ? (const T16*)res.data_ptr() : nullptr,
                         (T16*)y.data_ptr(),
                         mask.defined() ? mask.data_ptr<unsigned char>()
                                        : nullptr,
                         E, C, (int)act);
    });
    return {y, mask.defined() ? mask : at::Tensor()};
  }
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(bn_apply_kernel<T16>, dim3(ew_grid2(M * C)), dim3(256),
                       0, cur_stream(), (const T16*)x.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       has_res ? (const T16*)res.data_ptr() : nullptr,
                       (T16*)y.data_ptr(), M, C, (int)act, has_res);
  });
  return {y, at::Tensor()};
}

// asc/ash (optional [C] fp32): lazy-BN path — recompute the ReLU mask
// as (x*asc+ash) > 0 instead of reading a stored mask or y
at::Tensor bn_bwd_reduce(at::Tensor x, at::Tensor dy, at::Tensor y,
                         at::Tensor mean, at::Tensor invstd,
                         at::Tensor mask, at::Tensor asc, at::Tensor ash) {
  TORCH_CHECK(!(asc.defined() && asc.numel() > 0) ||
                  bn_fast_ok(x.numel() / x.size(-1), x.size(-1)),
              "mask recompute (asc/ash) needs the fast BN path");
  CHECK_GPU(x);
  CHECK_CONTIG(dy);
  const int C = x.size(-1);
  const long M = x.numel() / C;
  if (bn_fast_ok(M, C)) {
    const long E = M * (long)C;
    long e_per_block;
    dim3 grid = bn_fast_grid(E, e_per_block);
    auto out = grid.x == 1 ? at::empty({2, C}, x.options().dtype(at::kFloat))
                           : at::zeros({2, C}, x.options().dtype(at::kFloat));
    const bool rc = asc.defined() && asc.numel() > 0;
    DISPATCH_16(x, T16, {
      if (rc)
        hipLaunchKernelGGL((bn_reduce_fast<T16, true, true>), grid,
                           dim3(256), 2 * C * sizeof(float), cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           asc.data_ptr<float>(), ash.data_ptr<float>(),
                           out.data_ptr<float>(), E, C, e_per_block);
      else
        hipLaunchKernelGGL((bn_reduce_fast<T16, true>), grid, dim3(256),
                           2 * C * sizeof(float), cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(),
                           mask.numel() ? mask.data_ptr<unsigned char>()
                                        : nullptr,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           nullptr, nullptr,
                           out.data_ptr<float>(), E, C, e_per_block);
    });
    return out;
  }
  long m_per_chunk;
  dim3 grid = chan_grid(M, C, m_per_chunk);
  auto out = grid.x == 1 ? at::empty({2, C}, x.options().dtype(at::kFloat))
                         : at::zeros({2, C}, x.options().dtype(at::kFloat));
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(bn_bwd_reduce_kernel<T16>, grid, dim3(256), 0,
                       cur_stream(), (const T16*)x.data_ptr(),
                       (const T16*)dy.data_ptr(),
                       y.numel() ? (const T16*)y.data_ptr() : nullptr,
                       mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), out.data_ptr<float>(), M, C,
                       m_per_chunk);
  });
  return out;
}

std::vector<at::Tensor> bn_bwd_dx(at::Tensor x, at::Tensor dy, at::Tensor y,
                                  at::Tensor mean, at::Tensor invstd,
                                  at::Tensor gamma, at::Tensor dgamma,
                                  at::Tensor dbeta, double m_total,
                                  bool want_dres, at::Tensor mask,
                                  at::Tensor asc, at::Tensor ash) {
  TORCH_CHECK(!(asc.defined() && asc.numel() > 0) ||
                  bn_fast_ok(x.numel() / x.size(-1), x.size(-1)),
              "mask recompute (asc/ash) needs the fast BN path");
  CHECK_GPU(x);
  CHECK_CONTIG(dy);
  const int C = x.size(-1);
  const long M = x.numel() / C;
  auto dx = at::empty_like(x);
  auto dres = want_dres ? at::empty_like(x) : at::Tensor();
  if (bn_fast_ok(M, C)) {
    const long E = M * (long)C;
    const int grid = (int)std::min<long>(cdiv_l(E, 256 * 8), 2048);
    const bool rc = asc.defined() && asc.numel() > 0;
    DISPATCH_16(x, T16, {
      if (rc)
        hipLaunchKernelGGL((bn_bwd_dx_fast<T16, true>), dim3(grid),
                           dim3(256), 0, cur_stream(),
                           (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(), nullptr,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), dgamma.data_ptr<float>(),
                           dbeta.data_ptr<float>(), asc.data_ptr<float>(),
                           ash.data_ptr<float>(), (T16*)dx.data_ptr(),
                           want_dres ? (T16*)dres.data_ptr() : nullptr, E, C,
                           (float)(1.0 / m_total));
      else
        hipLaunchKernelGGL((bn_bwd_dx_fast<T16>), dim3(grid), dim3(256), 0,
                           cur_stream(), (const T16*)x.data_ptr(),
                           (const T16*)dy.data_ptr(),
                           mask.numel() ? mask.data_ptr<unsigned char>()
                                        : nullptr,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), dgamma.data_ptr<float>(),
                           dbeta.data_ptr<float>(), nullptr, nullptr,
                           (T16*)dx.data_ptr(),
                           want_dres ? (T16*)dres.data_ptr() : nullptr, E, C,
                           (float)(1.0 / m_total));
    });
    return {dx, want_dres ? dres : at::Tensor()};
  }
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(bn_bwd_dx_kernel<T16>, dim3(ew_grid2(M * C)),
                       dim3(256), 0, cur_stream(), (const T16*)x.data_ptr(),
                       (const T16*)dy.data_ptr(),
                       y.numel() ? (const T16*)y.data_ptr() : nullptr,
                       mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                       (T16*)dx.data_ptr(),
                       want_dres ? (T16*)dres.data_ptr() : nullptr, M, C,
                       (float)(1.0 / m_total));
  });
  return {dx, want_dres ? dres : at::Tensor()};
}
