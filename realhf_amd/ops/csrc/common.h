// Common helpers for realhf_amd CDNA4 (gfx950) kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;

DEVINL float bf2f(short s) {
  union { float f; unsigned u; } v;
  v.u = ((unsigned)(unsigned short)s) << 16;
  return v.f;
}

DEVINL short f2bf(float f) {
  union { float f; unsigned u; } v;
  v.f = f;
  // round-to-nearest-even
  unsigned r = v.u + 0x7FFF + ((v.u >> 16) & 1);
  return (short)(r >> 16);
}

// wave-wide reductions (64 lanes)
DEVINL float wave_sum(float x) {
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, 64);
  return __shfl(x, 0, 64);
}
DEVINL float wave_max(float x) {
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_down(x, off, 64));
  return __shfl(x, 0, 64);
}

#define CHECK_CUDA_OK() do { \
  hipError_t e = hipGetLastError(); \
  TORCH_CHECK(e == hipSuccess, "HIP kernel launch failed: ", hipGetErrorString(e)); \
} while (0)

inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

#define DISPATCH_BF16_FP16_FP32(TYPE, NAME, ...) \
  [&] { \
    if (TYPE == at::ScalarType::BFloat16) { using scalar_t = __hip_bfloat16; return __VA_ARGS__(); } \
    else if (TYPE == at::ScalarType::Half) { using scalar_t = __half; return __VA_ARGS__(); } \
    else if (TYPE == at::ScalarType::Float) { using scalar_t = float; return __VA_ARGS__(); } \
    else { TORCH_CHECK(false, #NAME, " unsupported dtype"); } \
  }()

template <typename T> DEVINL float to_f32(T x);
template <> DEVINL float to_f32<__hip_bfloat16>(__hip_bfloat16 x) { return __bfloat162float(x); }
template <> DEVINL float to_f32<__half>(__half x) { return __half2float(x); }
template <> DEVINL float to_f32<float>(float x) { return x; }

template <typename T> DEVINL T from_f32(float x);
template <> DEVINL __hip_bfloat16 from_f32<__hip_bfloat16>(float x) { return __float2bfloat16(x); }
template <> DEVINL __half from_f32<__half>(float x) { return __float2half(x); }
template <> DEVINL float from_f32<float>(float x) { return x; }
