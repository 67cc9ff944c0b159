// Common device helpers for the moco_amd gfx950 kernels.
// Pure HIP — no torch headers in device TUs.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64  // CDNA4 wavefront (hard-coded per the gfx950 guide)

using bf16 = __hip_bfloat16;

// MFMA fragment types (gfx950: mfma_f32_16x16x32_bf16 takes v8bf16 operands,
// accumulates f32x4)
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

__device__ __forceinline__ float wave_reduce_sum(float v) {
  // full 64-lane butterfly reduction
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

__device__ __forceinline__ float bf2f(bf16 x) { return __bfloat162float(x); }
__device__ __forceinline__ bf16 f2bf(float x) { return __float2bfloat16(x); }

#define HIP_CHECK_LAST()                                                   \
  do {                                                                     \
    hipError_t e_ = hipGetLastError();                                     \
    if (e_ != hipSuccess) {                                                \
      return e_;                                                           \
    }                                                                      \
  } while (0)
