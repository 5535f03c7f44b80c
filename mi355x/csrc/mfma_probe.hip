// Exact-layout probe for v_mfma_f32_32x32x16_bf16 (debug/verification):
// stages the A[32,16] / B[16,32] fragments per the layout the conv/gemm
// kernels assume (A: lane l&31 = row, k = (l>>5)*8+e; B: lane l&31 = col,
// same k split; D: col = l&31, row = (reg&3)+8*(reg>>2)+4*(l>>5)) and
// returns D so tests can assert D == A@B elementwise against torch
// (asymmetric inputs — guide §3 transpose-detection rule).
#include "common.h"

typedef __attribute__((ext_vector_type(16))) float f32x16;

namespace {

__global__ void mfma_probe_kernel(const float* __restrict__ A,  // [32,16]
                                  const float* __restrict__ B,  // [16,32]
                                  float* __restrict__ D) {      // [32,32]
  const int lane = threadIdx.x;
  const int li = lane & 31;
  const int kh = lane >> 5;
  short8 a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    __hip_bfloat16 av = __float2bfloat16(A[li * 16 + kh * 8 + e]);
    __hip_bfloat16 bv = __float2bfloat16(B[(kh * 8 + e) * 32 + li]);
    short as, bs;
    __builtin_memcpy(&as, &av, 2);
    __builtin_memcpy(&bs, &bv, 2);
    a[e] = as;
    b[e] = bs;
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    D[row * 32 + li] = acc[reg];
  }
}

}  // namespace

at::Tensor mfma_probe32(at::Tensor A, at::Tensor B) {
  CHECK_GPU(A);
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}) &&
              B.sizes() == at::IntArrayRef({16, 32}));
  auto D = at::zeros({32, 32}, A.options());
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     A.data_ptr<float>(), B.data_ptr<float>(),
                     D.data_ptr<float>());
  return D;
}
