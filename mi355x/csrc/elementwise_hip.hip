#include "hip/hip_runtime.h"
// Elementwise kernels: ReLU backward mask, fused flat SGD, f32->16 cast.
// All memory-bound: vectorized 16 B/lane accesses, grid-stride loops capped
// at ~2048 workgroups (guide Guideline 11/13).
#include "common_hip.h"

namespace {

template <typename T16>
__global__ void relu_bwd_kernel(const T16* __restrict__ dy,
                                const T16* __restrict__ y,
                                T16* __restrict__ dx, long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n) {
      short8 vdy = *reinterpret_cast<const short8*>(dy + i);
      short8 vy = *reinterpret_cast<const short8*>(y + i);
      short8 o;
#pragma unroll
      for (int k = 0; k < 8; ++k)
        o[k] = s16_to_f32<T16>(vy[k]) > 0.f ? (short)vdy[k] : f32_to_s16<T16>(0.f);
      *reinterpret_cast<short8*>(dx + i) = o;
    } else {
      for (long k = i; k < n; ++k)
        dx[k] = F16<T16>::to_f32(y[k]) > 0.f ? dy[k] : F16<T16>::from_f32(0.f);
    }
  }
}

__global__ void sgd_kernel(float* __restrict__ p, const float* __restrict__ g,
                           float* __restrict__ m, long n, float lr, float mu,
                           float wd, float gscale) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i < n; i += stride) {
    if (i + 4 <= n) {
      float4v vg = *reinterpret_cast<const float4v*>(g + i);
      float4v vp = *reinterpret_cast<const float4v*>(p + i);
      float4v vm = *reinterpret_cast<const float4v*>(m + i);
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float gr = vg[k] * gscale + wd * vp[k];
        vm[k] = mu * vm[k] + gr;
        vp[k] -= lr * vm[k];
      }
      *reinterpret_cast<float4v*>(m + i) = vm;
      *reinterpret_cast<float4v*>(p + i) = vp;
    } else {
      for (long k = i; k < n; ++k) {
        float gr = g[k] * gscale + wd * p[k];
        m[k] = mu * m[k] + gr;
        p[k] -= lr * m[k];
      }
    }
  }
}

template <typename T16>
__global__ void cast_f32_to_16_kernel(const float* __restrict__ src,
                                      T16* __restrict__ dst, long n) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i < n; i += stride) {
    if (i + 4 <= n) {
      float4v v = *reinterpret_cast<const float4v*>(src + i);
      short4v o;
#pragma unroll
      for (int k = 0; k < 4; ++k) o[k] = f32_to_s16<T16>(v[k]);
      *reinterpret_cast<short4v*>(dst + i) = o;
    } else {
      for (long k = i; k < n; ++k) dst[k] = F16<T16>::from_f32(src[k]);
    }
  }
}

inline int ew_grid(long n, int per_thread, int block = 256) {
  long blocks = cdiv_l(n, (long)block * per_thread);
  return (int)std::min<long>(blocks, 2048);
}

}  // namespace

at::Tensor relu_bwd(at::Tensor dy, at::Tensor y) {
  CHECK_GPU(dy);
  CHECK_CONTIG(dy);
  CHECK_16BIT(dy);
  auto dx = at::empty_like(dy);
  long n = dy.numel();
  DISPATCH_16(dy, T16, {
    hipLaunchKernelGGL(relu_bwd_kernel<T16>, dim3(ew_grid(n, 8)), dim3(256), 0,
                       cur_stream(), (const T16*)dy.data_ptr(),
                       (const T16*)y.data_ptr(), (T16*)dx.data_ptr(), n);
  });
  return dx;
}

void sgd_step(at::Tensor p, at::Tensor g, at::Tensor m, double lr, double mu,
              double wd, double gscale) {
  CHECK_GPU(p);
  long n = p.numel();
  hipLaunchKernelGGL(sgd_kernel, dim3(ew_grid(n, 4)), dim3(256), 0,
                     cur_stream(), p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), n, (float)lr, (float)mu, (float)wd,
                     (float)gscale);
}

at::Tensor cast_to_16(at::Tensor src, at::Tensor like) {
  CHECK_GPU(src);
  auto dst = at::empty(src.sizes(), like.options());
  long n = src.numel();
  DISPATCH_16(like, T16, {
    hipLaunchKernelGGL(cast_f32_to_16_kernel<T16>, dim3(ew_grid(n, 4)),
                       dim3(256), 0, cur_stream(),
                       src.data_ptr<float>(), (T16*)dst.data_ptr(), n);
  });
  return dst;
}
