// Common helpers for tensorlink_amd CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes; LDS 160 KiB/CU; HBM3E ~8 TB/s. All kernels are
// written for gfx950 only — no CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define WAVE_SIZE 64

#define DEVINLINE __device__ __forceinline__

using bf16 = __hip_bfloat16;
using bf16x2 = __hip_bfloat162;

// 16-byte vector of 8 bf16 elements — the unit of global-memory access.
struct alignas(16) bf16x8 {
  bf16 v[8];
};

union f32x4_u {
  float4 f4;
  float f[4];
};

DEVINLINE float bf2f(bf16 x) { return __bfloat162float(x); }
DEVINLINE bf16 f2bf(float x) { return __float2bfloat16(x); }

// ---------------------------------------------------------------------------
// Wave-level reductions (64-lane). Use xor shuffles; __shfl_xor operates on
// the full 64-lane wave on gfx950.
// ---------------------------------------------------------------------------
DEVINLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

DEVINLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE_SIZE));
  return x;
}

// Reduce within contiguous groups of G lanes (G power of two <= 64).
template <int G>
DEVINLINE float group_reduce_sum(float x) {
#pragma unroll
  for (int off = G / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE_SIZE);
  return x;
}

// Block-level sum across NW waves using LDS scratch (caller provides >= NW
// floats). Result valid in all threads.
template <int NW>
DEVINLINE float block_reduce_sum(float x, float* lds) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds[wave] = x;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < NW; ++w) total += lds[w];
  return total;
}

__host__ __device__ inline int cdiv(int a, int b) { return (a + b - 1) / b; }
__host__ __device__ inline int64_t cdiv64(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
             __LINE__);                                                     \
    }                                                                       \
  } while (0)
