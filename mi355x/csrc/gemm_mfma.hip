// MFMA tile GEMM for the linear layers (reference call
// sites /root/reference/cifar_example.py:23-25,31-33). One kernel shape
// serves all three linear GEMMs:
//   fwd   : y[M,N]  = x[M,K] @ w[N,K]^T          (A=x,   B=w)
//   dgrad : dx[M,K] = dy[M,N] @ w[N,K]           (A=dy,  B=wT[K,N])
//   wgrad : dw[N,K] = dy[M,N]^T @ x[M,K]         (A=dyT, B=xT, fp32 out)
// because each is an out = A @ B^T with both operands contiguous along
// the reduction dim. Same CDNA4 anatomy as the conv gather-GEMM
// (conv_mfma.hip): 128x64 block tile, 4 waves x (32M x 64N) as
// 2 x v_mfma_f32_32x32x16, single-buffered LDS tiles [row][BK+8],
// next global load issued under the MFMA phase. M/N/K tails are
// zero-filled at 8-element granularity (callers guarantee K % 8 == 0;
// gemm.hip keeps direct kernels for the sub-MFMA shapes).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) _Float16 half8_g;
typedef __attribute__((ext_vector_type(16))) float f32x16_g;

namespace {

constexpr int GBM = 128, GBN = 64, GBK = 64;
constexpr int GLDK = GBK + 8;  // +16B pad spreads fragment reads over banks

template <typename T16>
struct GMfma {};
template <>
struct GMfma<__hip_bfloat16> {
  static DEV_INLINE f32x16_g run(short8 a, short8 b, f32x16_g c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
};
template <>
struct GMfma<__half> {
  static DEV_INLINE f32x16_g run(short8 a, short8 b, f32x16_g c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16((half8_g)a, (half8_g)b, c,
                                                  0, 0, 0);
  }
};

template <typename T16, bool OUT32>
__global__ __launch_bounds__(256, 2) void gemm_mfma_kernel(
    const T16* __restrict__ A,      // [M, K]
    const T16* __restrict__ B,      // [N, K]
    const float* __restrict__ bias,  // [N] or null
    void* __restrict__ out,         // [M, N] T16 (or float when OUT32)
    const int M, const int N, const int K, const int act,
    const int has_bias) {
  __shared__ T16 lds[GBM * GLDK + GBN * GLDK];
  const int tid = threadIdx.x;
  const long bm0 = (long)blockIdx.x * GBM;
  const int n0 = blockIdx.y * GBN;

  // A staging: 2 threads per m-row, 32 k-elements each
  const int sa_m = tid >> 1;
  const int sa_k = (tid & 1) * 32;
  const long m_a = bm0 + sa_m;
  const bool m_ok = m_a < M;
  // B staging: 4 threads per n-row, 16 k-elements each
  const int sb_n = tid >> 2;
  const int sb_k = (tid & 3) * 16;
  const bool n_ok = n0 + sb_n < N;

  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 31;
  const int kh = lane >> 5;
  const int wm = wave * 32;

  f32x16_g acc[2] = {};
  const int ksteps = (K + GBK - 1) / GBK;
  const short8 zero8 = {};

  short8 sa[4], sb[2];
  auto load_step = [&](int j) {
    const int k0 = j * GBK;
    const T16* ap = A + m_a * (long)K + k0 + sa_k;
#pragma unroll
    for (int i = 0; i < 4; ++i)
      sa[i] = (m_ok && k0 + sa_k + 8 * i + 8 <= K)
                  ? *reinterpret_cast<const short8*>(ap + 8 * i)
                  : zero8;
    const T16* bp = B + (long)(n0 + sb_n) * K + k0 + sb_k;
#pragma unroll
    for (int i = 0; i < 2; ++i)
      sb[i] = (n_ok && k0 + sb_k + 8 * i + 8 <= K)
                  ? *reinterpret_cast<const short8*>(bp + 8 * i)
                  : zero8;
  };
  auto stage = [&]() {
    short* pa = reinterpret_cast<short*>(lds + sa_m * GLDK + sa_k);
#pragma unroll
    for (int i = 0; i < 4; ++i) *reinterpret_cast<short8*>(pa + 8 * i) = sa[i];
    short* pb =
        reinterpret_cast<short*>(lds + GBM * GLDK + sb_n * GLDK + sb_k);
#pragma unroll
    for (int i = 0; i < 2; ++i) *reinterpret_cast<short8*>(pb + 8 * i) = sb[i];
  };

  load_step(0);
  for (int j = 0; j < ksteps; ++j) {
    __syncthreads();  // previous MFMA phase done reading LDS
    stage();
    __syncthreads();
    if (j + 1 < ksteps) load_step(j + 1);  // rides under the MFMA phase
    const T16* ldsA = lds;
    const T16* ldsB = lds + GBM * GLDK;
#pragma unroll
    for (int kk = 0; kk < GBK; kk += 16) {
      const short8 af = *reinterpret_cast<const short8*>(
          ldsA + (wm + li) * GLDK + kk + kh * 8);
      const short8 b0 = *reinterpret_cast<const short8*>(
          ldsB + li * GLDK + kk + kh * 8);
      const short8 b1 = *reinterpret_cast<const short8*>(
          ldsB + (32 + li) * GLDK + kk + kh * 8);
      acc[0] = GMfma<T16>::run(af, b0, acc[0]);
      acc[1] = GMfma<T16>::run(af, b1, acc[1]);
    }
  }

  // epilogue: bias + act + store (N-tail guarded per column)
  float bv[2];
  bv[0] = (has_bias && n0 + li < N) ? bias[n0 + li] : 0.f;
  bv[1] = (has_bias && n0 + 32 + li < N) ? bias[n0 + 32 + li] : 0.f;
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    const int row = (reg & 3) + 8 * (reg >> 2) + 4 * kh;
    const long m_out = bm0 + wm + row;
    if (m_out < M) {
#pragma unroll
      for (int t2 = 0; t2 < 2; ++t2) {
        const int n = n0 + t2 * 32 + li;
        if (n < N) {
          float v = acc[t2][reg] + bv[t2];
          if (act == 1) v = fmaxf(v, 0.f);
          if (OUT32)
            reinterpret_cast<float*>(out)[m_out * (long)N + n] = v;
          else
            reinterpret_cast<T16*>(out)[m_out * (long)N + n] =
                F16<T16>::from_f32(v);
        }
      }
    }
  }
}

}  // namespace

// out[M,N] = A[M,K] @ B[N,K]^T (+bias, +relu). K % 8 == 0 required.
// out32 stores fp32 (wgrad path); otherwise A's 16-bit dtype.
at::Tensor gemm_nt_mfma(at::Tensor A, at::Tensor B, at::Tensor bias, long act,
                        bool out32) {
  CHECK_GPU(A);
  CHECK_CONTIG(A);
  CHECK_CONTIG(B);
  CHECK_16BIT(A);
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K, "gemm_nt_mfma: inner-dim mismatch");
  TORCH_CHECK(K % 8 == 0, "gemm_nt_mfma requires K % 8 == 0");
  auto out = at::empty({M, N},
                       out32 ? A.options().dtype(at::kFloat) : A.options());
  const int has_bias = bias.numel() > 0;
  dim3 grid((unsigned)cdiv_l(M, GBM), (unsigned)cdiv_l(N, GBN));
  DISPATCH_16(A, T16, {
    if (out32)
      hipLaunchKernelGGL((gemm_mfma_kernel<T16, true>), grid, dim3(256), 0,
                         cur_stream(), (const T16*)A.data_ptr(),
                         (const T16*)B.data_ptr(),
                         has_bias ? bias.data_ptr<float>() : nullptr,
                         out.data_ptr(), M, N, K, (int)act, has_bias);
    else
      hipLaunchKernelGGL((gemm_mfma_kernel<T16, false>), grid, dim3(256), 0,
                         cur_stream(), (const T16*)A.data_ptr(),
                         (const T16*)B.data_ptr(),
                         has_bias ? bias.data_ptr<float>() : nullptr,
                         out.data_ptr(), M, N, K, (int)act, has_bias);
  });
  return out;
}
