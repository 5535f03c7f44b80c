#include "hip/hip_runtime.h"
// Weight-cache kernels: fp32 [K,C,R,S] parameter -> 16-bit kernel layouts
// in ONE launch each (replaces torch permute+cast+contiguous chains that
// showed up as dozens of 5 us glue kernels per step in rocprof).
#include "common_hip.h"

namespace {

// out[k][r][s][c] = w[k][c][r][s]; contiguous over c (vector stores)
template <typename T16>
__global__ void cast_krsc_kernel(const float* __restrict__ w,
                                 T16* __restrict__ out, int K, int C, int R,
                                 int S) {
  const long total = (long)K * R * S * C;
  const long RS = (long)R * S;
  for (long t = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; t < total;
       t += (long)gridDim.x * blockDim.x * 8) {
    if (t + 8 <= total && C % 8 == 0) {
      const int c0 = (int)(t % C);
      const long krs = t / C;
      const int s = (int)(krs % S);
      const int r = (int)((krs / S) % R);
      const int k = (int)(krs / RS);
      short8 o;
#pragma unroll
      for (int u = 0; u < 8; ++u)
        o[u] = f32_to_s16<T16>(
            w[((long)k * C + c0 + u) * RS + (long)r * S + s]);
      *reinterpret_cast<short8*>(out + t) = o;
    } else {
      for (long e = t; e < min(t + 8, total); ++e) {
        const int c = (int)(e % C);
        const long krs = e / C;
        const int s = (int)(krs % S);
        const int r = (int)((krs / S) % R);
        const int k = (int)(krs / RS);
        out[e] = F16<T16>::from_f32(
            w[((long)k * C + c) * RS + (long)r * S + s]);
      }
    }
  }
}

// out[r][s][c][k] = w[k][c][r][s]; contiguous over k (vector stores)
template <typename T16>
__global__ void cast_rsck_kernel(const float* __restrict__ w,
                                 T16* __restrict__ out, int K, int C, int R,
                                 int S) {
  const long total = (long)K * R * S * C;
  const long RS = (long)R * S;
  const long CRS = (long)C * RS;
  for (long t = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8; t < total;
       t += (long)gridDim.x * blockDim.x * 8) {
    if (t + 8 <= total && K % 8 == 0) {
      const int k0 = (int)(t % K);
      const long rsc = t / K;
      const int c = (int)(rsc % C);
      const int s = (int)((rsc / C) % S);
      const int r = (int)(rsc / ((long)C * S));
      const long src = (long)c * RS + (long)r * S + s;
      short8 o;
#pragma unroll
      for (int u = 0; u < 8; ++u)
        o[u] = f32_to_s16<T16>(w[(long)(k0 + u) * CRS + src]);
      *reinterpret_cast<short8*>(out + t) = o;
    } else {
      for (long e = t; e < min(t + 8, total); ++e) {
        const int k = (int)(e % K);
        const long rsc = e / K;
        const int c = (int)(rsc % C);
        const int s = (int)((rsc / C) % S);
        const int r = (int)(rsc / ((long)C * S));
        out[e] = F16<T16>::from_f32(
            w[(long)k * CRS + (long)c * RS + (long)r * S + s]);
      }
    }
  }
}

// out[k][kg] with kg=(r*S+s)*C+c over row stride KGP; scalar (weights tiny)
template <typename T16>
__global__ void cast_krsc_pad_kernel(const float* __restrict__ w,
                                     T16* __restrict__ out, int K, int C,
                                     int R, int S, long KGP) {
  const long KG = (long)R * S * C;
  const long total = (long)K * KG;
  const long RS = (long)R * S;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long kg = t % KG;
    const int k = (int)(t / KG);
    const int c = (int)(kg % C);
    const int tap = (int)(kg / C);
    const int r = tap / S, s0 = tap % S;
    out[(long)k * KGP + kg] = F16<T16>::from_f32(
        w[((long)k * C + c) * RS + (long)r * S + s0]);
  }
}

inline int wgrid(long n) {
  return (int)std::min<long>(cdiv_l(n, 256 * 8), 1024);
}

}  // namespace

at::Tensor cast_permute_krsc(at::Tensor w, at::Tensor like) {
  CHECK_GPU(w);
  CHECK_CONTIG(w);
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  auto out = at::empty({K, R, S, C}, like.options());
  const long total = (long)K * C * R * S;
  DISPATCH_16(like, T16, {
    hipLaunchKernelGGL(cast_krsc_kernel<T16>, dim3(wgrid(total)), dim3(256),
                       0, cur_stream(), w.data_ptr<float>(),
                       (T16*)out.data_ptr(), K, C, R, S);
  });
  return out;
}

// out [K, KGP]: kg = (r*S+s)*C + c, zero-padded to KGP (GENC conv path)
at::Tensor cast_permute_krsc_pad(at::Tensor w, at::Tensor like) {
  CHECK_GPU(w);
  CHECK_CONTIG(w);
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  const long KG = (long)R * S * C;
  const long KGP = cdiv_l(KG, 64) * 64;
  auto out = at::zeros({K, KGP}, like.options());
  // reuse the KRSC kernel into a strided view: simplest correct form is a
  // per-element copy kernel below
  const long total = (long)K * KG;
  DISPATCH_16(like, T16, {
    hipLaunchKernelGGL(cast_krsc_pad_kernel<T16>, dim3(wgrid(total)),
                       dim3(256), 0, cur_stream(), w.data_ptr<float>(),
                       (T16*)out.data_ptr(), K, C, R, S, KGP);
  });
  return out;
}

at::Tensor cast_permute_rsck(at::Tensor w, at::Tensor like) {
  CHECK_GPU(w);
  CHECK_CONTIG(w);
  const int K = w.size(0), C = w.size(1), R = w.size(2), S = w.size(3);
  auto out = at::empty({R, S, C, K}, like.options());
  const long total = (long)K * C * R * S;
  DISPATCH_16(like, T16, {
    hipLaunchKernelGGL(cast_rsck_kernel<T16>, dim3(wgrid(total)), dim3(256),
                       0, cur_stream(), w.data_ptr<float>(),
                       (T16*)out.data_ptr(), K, C, R, S);
  });
  return out;
}
