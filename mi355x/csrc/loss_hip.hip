#include "hip/hip_runtime.h"
// Cross-entropy: one fused fwd kernel (row max + logsumexp +
// NLL, one wave per row, shuffle reduction — guide Appendix B 'Reduction'),
// one bwd kernel (softmax - onehot, scaled).
#include "common_hip.h"

namespace {

template <typename T16>
__global__ void ce_fwd_kernel(const T16* __restrict__ logits,
                              const long* __restrict__ target,
                              float* __restrict__ loss,
                              float* __restrict__ lse, int B, int C) {
  const int b = blockIdx.x;  // one wave per row
  if (b >= B) return;
  const int lane = threadIdx.x;
  const T16* row = logits + (long)b * C;
  float mx = -3.4e38f;
  for (int c = lane; c < C; c += kWave) mx = fmaxf(mx, F16<T16>::to_f32(row[c]));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    mx = fmaxf(mx, __shfl_down(mx, off, kWave));
  mx = __shfl(mx, 0, kWave);
  float s = 0.f;
  for (int c = lane; c < C; c += kWave) s += __expf(F16<T16>::to_f32(row[c]) - mx);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, kWave);
  if (lane == 0) {
    const float l = mx + __logf(s);
    lse[b] = l;
    const long t = target[b];
    // out-of-range target: poison the loss (NaN) instead of faulting
    loss[b] = (t >= 0 && t < C) ? l - F16<T16>::to_f32(row[t])
                                : __builtin_nanf("");
  }
}

template <typename T16>
__global__ void ce_bwd_kernel(const T16* __restrict__ logits,
                              const long* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,  // 0-dim, device
                              T16* __restrict__ dlogits, int B, int C,
                              float inv_b) {
  const float scale = (dloss ? *dloss : 1.f) * inv_b;
  const long total = (long)B * C;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const int c = (int)(t % C);
    const int b = (int)(t / C);
    float p = __expf(F16<T16>::to_f32(logits[t]) - lse[b]);
    if (c == (int)target[b]) p -= 1.f;
    dlogits[t] = F16<T16>::from_f32(p * scale);
  }
}

}  // namespace

std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits,
                                          at::Tensor target) {
  CHECK_GPU(logits);
  CHECK_CONTIG(logits);
  CHECK_16BIT(logits);
  const int B = logits.size(0), C = logits.size(1);
  auto loss = at::empty({B}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({B}, logits.options().dtype(at::kFloat));
  DISPATCH_16(logits, T16, {
    hipLaunchKernelGGL(ce_fwd_kernel<T16>, dim3(B), dim3(kWave), 0,
                       cur_stream(), (const T16*)logits.data_ptr(),
                       target.data_ptr<long>(), loss.data_ptr<float>(),
                       lse.data_ptr<float>(), B, C);
  });
  return {loss.mean(), lse};
}

// dloss: 0-dim fp32 CUDA tensor (the upstream grad) read ON DEVICE so the
// backward is hipGraph-capturable (no host .item() sync); empty => 1.0.
at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor target,
                             at::Tensor lse, at::Tensor dloss) {
  CHECK_GPU(logits);
  const int B = logits.size(0), C = logits.size(1);
  auto dlogits = at::empty_like(logits);
  const long total = (long)B * C;
  const int grid = (int)std::min<long>(cdiv_l(total, 256), 2048);
  const float* dl = dloss.numel() ? dloss.data_ptr<float>() : nullptr;
  DISPATCH_16(logits, T16, {
    hipLaunchKernelGGL(ce_bwd_kernel<T16>, dim3(grid), dim3(256), 0,
                       cur_stream(), (const T16*)logits.data_ptr(),
                       target.data_ptr<long>(), lse.data_ptr<float>(), dl,
                       (T16*)dlogits.data_ptr(), B, C, 1.f / B);
  });
  return dlogits;
}
