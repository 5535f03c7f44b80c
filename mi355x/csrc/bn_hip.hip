#include "hip/hip_runtime.h"
// BatchNorm over NHWC (SURVEY.md N8 / BASELINE configs 2-5):
//  - bn_stats: per-channel (sum, sumsq) partials in one pass (f32 accum) —
//    the same partials SyncBN all-reduces across ranks;
//  - bn_apply: fused normalize + affine + optional residual add + ReLU;
//  - bn_bwd_reduce: per-channel (sum dy*xhat, sum dy);
//  - bn_bwd_dx: dx from the reduced terms.
// Channel-minor layout makes every access coalesced along C; reductions are
// per-thread private over an M-chunk, then one atomicAdd per channel per
// chunk (guide Guideline 12).
#include "common_hip.h"

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

template <typename T16>
__global__ void bn_bwd_reduce_kernel(const T16* __restrict__ x,
                                     const T16* __restrict__ dy,
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
    const float d = F16<T16>::to_f32(dy[m * C + c]);
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

template <typename T16>
__global__ void bn_bwd_dx_kernel(const T16* __restrict__ x,
                                 const T16* __restrict__ dy,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ dgamma,
                                 const float* __restrict__ dbeta,
                                 T16* __restrict__ dx, long M, int C,
                                 float inv_m) {
  const long total = M * C;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const int c = (int)(t % C);
    const float mu = mean[c], is = invstd[c];
    const float d = F16<T16>::to_f32(dy[t]);
    const float xh = (F16<T16>::to_f32(x[t]) - mu) * is;
    const float v =
        gamma[c] * is * (d - dbeta[c] * inv_m - xh * dgamma[c] * inv_m);
    dx[t] = F16<T16>::from_f32(v);
  }
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

at::Tensor bn_stats(at::Tensor x) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_16BIT(x);
  const int C = x.size(-1);
  const long M = x.numel() / C;
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

at::Tensor bn_apply(at::Tensor x, at::Tensor mean, at::Tensor invstd,
                    at::Tensor gamma, at::Tensor beta, at::Tensor res,
                    long act) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_16BIT(x);
  const int C = x.size(-1);
  const long M = x.numel() / C;
  auto y = at::empty_like(x);
  const int has_res = res.numel() > 0;
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(bn_apply_kernel<T16>, dim3(ew_grid2(M * C)), dim3(256),
                       0, cur_stream(), (const T16*)x.data_ptr(),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       has_res ? (const T16*)res.data_ptr() : nullptr,
                       (T16*)y.data_ptr(), M, C, (int)act, has_res);
  });
  return y;
}

at::Tensor bn_bwd_reduce(at::Tensor x, at::Tensor dy, at::Tensor mean,
                         at::Tensor invstd) {
  CHECK_GPU(x);
  CHECK_CONTIG(dy);
  const int C = x.size(-1);
  const long M = x.numel() / C;
  long m_per_chunk;
  dim3 grid = chan_grid(M, C, m_per_chunk);
  auto out = grid.x == 1 ? at::empty({2, C}, x.options().dtype(at::kFloat))
                         : at::zeros({2, C}, x.options().dtype(at::kFloat));
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(bn_bwd_reduce_kernel<T16>, grid, dim3(256), 0,
                       cur_stream(), (const T16*)x.data_ptr(),
                       (const T16*)dy.data_ptr(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), out.data_ptr<float>(), M, C,
                       m_per_chunk);
  });
  return out;
}

at::Tensor bn_bwd_dx(at::Tensor x, at::Tensor dy, at::Tensor mean,
                     at::Tensor invstd, at::Tensor gamma, at::Tensor dgamma,
                     at::Tensor dbeta, double m_total) {
  CHECK_GPU(x);
  CHECK_CONTIG(dy);
  const int C = x.size(-1);
  const long M = x.numel() / C;
  auto dx = at::empty_like(x);
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(bn_bwd_dx_kernel<T16>, dim3(ew_grid2(M * C)),
                       dim3(256), 0, cur_stream(), (const T16*)x.data_ptr(),
                       (const T16*)dy.data_ptr(), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                       (T16*)dx.data_ptr(), M, C, (float)(1.0 / m_total));
  });
  return dx;
}
