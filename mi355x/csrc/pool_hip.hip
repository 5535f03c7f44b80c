#include "hip/hip_runtime.h"
// NHWC pooling kernels: maxpool fwd (+argmax indices),
// maxpool bwd (index scatter; fp32 accumulation for overlapping windows),
// and global average pool.
#include "common_hip.h"

namespace {

template <typename T16>
__global__ void maxpool_fwd_kernel(const T16* __restrict__ x,
                                   T16* __restrict__ y,
                                   int* __restrict__ idx, int N, int H, int W,
                                   int C, int P, int Q, int kh, int kw,
                                   int stride, int pad) {
  const long total = (long)N * P * Q * C;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const int c = (int)(t % C);
    long npq = t / C;
    const int q = (int)(npq % Q);
    long np = npq / Q;
    const int p = (int)(np % P);
    const int n = (int)(np / P);
    float best = -3.4e38f;
    int best_hw = 0;
    for (int r = 0; r < kh; ++r) {
      const int ih = p * stride - pad + r;
      if (ih < 0 || ih >= H) continue;
      for (int s = 0; s < kw; ++s) {
        const int iw = q * stride - pad + s;
        if (iw < 0 || iw >= W) continue;
        const float v =
            F16<T16>::to_f32(x[(((long)n * H + ih) * W + iw) * C + c]);
        if (v > best) {
          best = v;
          best_hw = ih * W + iw;
        }
      }
    }
    y[t] = F16<T16>::from_f32(best);
    idx[t] = best_hw;
  }
}

template <typename T16>
__global__ void maxpool_bwd_kernel(const T16* __restrict__ dy,
                                   const int* __restrict__ idx,
                                   float* __restrict__ dxf, int N, int H,
                                   int W, int C, int P, int Q, int overlap) {
  const long total = (long)N * P * Q * C;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const int c = (int)(t % C);
    const int n = (int)(t / ((long)P * Q * C));
    const long dst = ((long)n * H * W + idx[t]) * C + c;
    const float v = F16<T16>::to_f32(dy[t]);
    if (overlap)
      atomicAdd(dxf + dst, v);
    else
      dxf[dst] = v;
  }
}

template <typename T16>
__global__ void gap_kernel(const T16* __restrict__ x, T16* __restrict__ y,
                           int N, long HW, int C) {
  const long total = (long)N * C;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const int c = (int)(t % C);
    const int n = (int)(t / C);
    const T16* xp = x + (long)n * HW * C + c;
    float s = 0.f;
    for (long m = 0; m < HW; ++m) s += F16<T16>::to_f32(xp[m * C]);
    y[t] = F16<T16>::from_f32(s / (float)HW);
  }
}

inline int pgrid(long n) { return (int)std::min<long>(cdiv_l(n, 256), 4096); }

}  // namespace

at::Tensor cast_to_16(at::Tensor src, at::Tensor like);  // elementwise.hip

std::vector<at::Tensor> maxpool_fwd(at::Tensor x, long kernel, long stride,
                                    long pad) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_16BIT(x);
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int P = (H + 2 * pad - kernel) / stride + 1;
  const int Q = (W + 2 * pad - kernel) / stride + 1;
  auto y = at::empty({N, P, Q, C}, x.options());
  auto idx = at::empty({N, P, Q, C}, x.options().dtype(at::kInt));
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(maxpool_fwd_kernel<T16>,
                       dim3(pgrid((long)N * P * Q * C)), dim3(256), 0,
                       cur_stream(), (const T16*)x.data_ptr(),
                       (T16*)y.data_ptr(), idx.data_ptr<int>(), N, H, W, C, P,
                       Q, (int)kernel, (int)kernel, (int)stride, (int)pad);
  });
  return {y, idx};
}

at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor idx, long H, long W,
                       long kernel, long stride, long pad) {
  CHECK_GPU(dy);
  CHECK_CONTIG(dy);
  const int N = dy.size(0), P = dy.size(1), Q = dy.size(2), C = dy.size(3);
  const int overlap = stride < kernel;
  auto dxf = at::zeros({N, H, W, (long)C}, dy.options().dtype(at::kFloat));
  DISPATCH_16(dy, T16, {
    hipLaunchKernelGGL(maxpool_bwd_kernel<T16>,
                       dim3(pgrid((long)N * P * Q * C)), dim3(256), 0,
                       cur_stream(), (const T16*)dy.data_ptr(),
                       idx.data_ptr<int>(), dxf.data_ptr<float>(), N, (int)H,
                       (int)W, C, P, Q, overlap);
  });
  return cast_to_16(dxf, dy);
}

at::Tensor global_avg_pool(at::Tensor x) {
  CHECK_GPU(x);
  CHECK_CONTIG(x);
  CHECK_16BIT(x);
  const int N = x.size(0), C = x.size(3);
  const long HW = (long)x.size(1) * x.size(2);
  auto y = at::empty({N, C}, x.options());
  DISPATCH_16(x, T16, {
    hipLaunchKernelGGL(gap_kernel<T16>, dim3(pgrid((long)N * C)), dim3(256),
                       0, cur_stream(), (const T16*)x.data_ptr(),
                       (T16*)y.data_ptr(), N, HW, C);
  });
  return y;
}
