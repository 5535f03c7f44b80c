// Shared helpers for the mi355x gfx950 kernel library.
// Native HIP for CDNA4 only — no CUDA dual path, no hipify.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#define DEV_INLINE __device__ __forceinline__

// 16-bit element traits: T16 is __hip_bfloat16 or __half.
template <typename T16>
struct F16 {};

template <>
struct F16<__hip_bfloat16> {
  static DEV_INLINE float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }
  static DEV_INLINE __hip_bfloat16 from_f32(float f) { return __float2bfloat16(f); }
};

template <>
struct F16<__half> {
  static DEV_INLINE float to_f32(__half v) { return __half2float(v); }
  static DEV_INLINE __half from_f32(float f) { return __float2half(f); }
};

// 8 x 16-bit lane vector (16 B — the CDNA4 coalescing sweet spot, guide G13).
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;

constexpr int kWave = 64;  // CDNA4 wavefront width

__host__ __device__ inline int cdiv_i(int a, int b) { return (a + b - 1) / b; }
static inline long cdiv_l(long a, long b) { return (a + b - 1) / b; }

// bit-exact short <-> 16-bit float element moves (vector lanes carry shorts)
template <typename T16>
DEV_INLINE float s16_to_f32(short v) {
  T16 t;
  __builtin_memcpy(&t, &v, 2);
  return F16<T16>::to_f32(t);
}

template <typename T16>
DEV_INLINE short f32_to_s16(float f) {
  T16 t = F16<T16>::from_f32(f);
  short v;
  __builtin_memcpy(&v, &t, 2);
  return v;
}

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define CHECK_16BIT(x)                                       \
  TORCH_CHECK((x).scalar_type() == at::kBFloat16 ||          \
                  (x).scalar_type() == at::kHalf,            \
              #x " must be bf16 or fp16")

// Dispatch on the 16-bit activation dtype.
#define DISPATCH_16(TENSOR, T16, ...)                  \
  do {                                                 \
    if ((TENSOR).scalar_type() == at::kBFloat16) {     \
      using T16 = __hip_bfloat16;                      \
      __VA_ARGS__;                                     \
    } else {                                           \
      using T16 = __half;                              \
      __VA_ARGS__;                                     \
    }                                                  \
  } while (0)

static inline hipStream_t cur_stream() {
  return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}
