// Common helpers for the distegnn_amd gfx950 HIP kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64  // CDNA4 wavefront width (gfx950)

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));             \
    }                                                                       \
  } while (0)

// dtype conversion helpers: accumulate in fp32 regardless of storage type.
template <typename T>
__device__ __forceinline__ float to_f32(T v) { return (float)v; }
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

template <typename T>
__device__ __forceinline__ T from_f32(float v) { return (T)v; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

static inline int num_blocks(long total, int per_block) {
  long b = (total + per_block - 1) / per_block;
  // Grid-stride loops cap the grid: 256 CUs x 8 blocks (G11).
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}
