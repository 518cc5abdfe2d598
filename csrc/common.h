// Common device helpers for kfac_amd CDNA4 (gfx950) kernels.
//
// All kernels are written natively for MI355X: wave64, MFMA via
// __builtin_amdgcn_mfma_*, LDS staging tuned for the 64-dword bank row.
// No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define KFAC_HIP_CHECK(expr)                                              \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess) {                                               \
      return _e;                                                          \
    }                                                                     \
  } while (0)

namespace kfac {

constexpr int kWave = 64;  // CDNA wavefront width

__host__ __device__ inline int ceil_div(int a, int b) {
  return (a + b - 1) / b;
}

// fp32x4 accumulator for mfma_f32_16x16x4_f32 (4 AGPRs per lane).
typedef __attribute__((ext_vector_type(4))) float f32x4;

template <typename T>
__device__ __forceinline__ float to_f32(T v);

template <>
__device__ __forceinline__ float to_f32<float>(float v) {
  return v;
}
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <>
__device__ __forceinline__ float to_f32<__half>(__half v) {
  return __half2float(v);
}

}  // namespace kfac
