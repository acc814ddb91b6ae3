// Common helpers for nerrf-amd CDNA4 (gfx950) kernels.
// Target: MI355X only — wave64, 256 CUs / 8 XCDs, LDS 160 KiB/CU.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define NERRF_WAVE 64

namespace nerrf {

__device__ __forceinline__ float to_f32(float x) { return x; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
__device__ __forceinline__ T from_f32(float x);
template <>
__device__ __forceinline__ float from_f32<float>(float x) { return x; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// Wave-wide f32 sum over all 64 lanes.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, NERRF_WAVE);
  return __shfl(v, 0, NERRF_WAVE);
}

}  // namespace nerrf

#define NERRF_CHECK_HIP(expr)                                            \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess) {                                              \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",   \
                  __FILE__, ":", __LINE__);                              \
    }                                                                    \
  } while (0)
