// Common helpers for lws_amd CDNA4 (gfx950 / MI355X) kernels.
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wavefront = 64 lanes; wave-width constants are hard-coded 64
//  - bf16 loads are vectorized as uint4 (16 B = 8 bf16 per lane) because
//    hipcc does not auto-vectorize scalar bf16 loads (Guideline 13)
//  - memory-bound kernels target the ~6.3 TB/s achievable HBM3E ceiling
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

using bf16 = __hip_bfloat16;

// 8 bf16 packed in one 16-byte vector load.
union bf16x8 {
  uint4 u;
  ushort h[8];
};

__device__ __forceinline__ float bf16_to_f32(ushort x) {
  union { float f; uint32_t u; } cvt;
  cvt.u = ((uint32_t)x) << 16;
  return cvt.f;
}

__device__ __forceinline__ ushort f32_to_bf16(float f) {
  // round-to-nearest-even, matching PyTorch's float->bfloat16 conversion
  union { float f; uint32_t u; } cvt;
  cvt.f = f;
  uint32_t x = cvt.u;
  uint32_t rounding_bias = 0x7FFF + ((x >> 16) & 1);
  if ((x & 0x7FFFFFFF) > 0x7F800000) return (ushort)((x >> 16) | 0x0040);  // NaN
  return (ushort)((x + rounding_bias) >> 16);
}

// Full-wave (64-lane) sum reduction.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// Block-wide sum reduction for block sizes up to 1024 (<=16 waves).
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int nwaves = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  v = (threadIdx.x < nwaves) ? lds_scratch[threadIdx.x] : 0.0f;
  if (wave == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
    if (lane == 0) lds_scratch[0] = v;
  }
  __syncthreads();
  return lds_scratch[0];
}

#define LWS_CHECK_HIP(expr)                                           \
  do {                                                                \
    hipError_t _e = (expr);                                           \
    if (_e != hipSuccess) {                                           \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));       \
    }                                                                 \
  } while (0)
