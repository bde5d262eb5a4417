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

// Flash-decode split count for ONE row of length L. Depends only on L —
// never on batch size or launch shape — so a sequence's attention
// numerics are identical whatever batch it is decoded in (the serving
// engine's exact-greedy guarantee; see decode_attn_mfma.hip). Monotone
// in L, so a host-side length upper bound gives a valid launch width.
__host__ __device__ inline int tl_split_for_len(int L) {
  int ns = 1;
  while (ns < 16 && L > 1024 * ns) ns <<= 1;
  return ns;
}
__host__ __device__ inline int64_t cdiv64(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 (gfx950 LDS transpose read) — no clang builtin; inline
// asm. Measured semantics (scripts/tr16_probe.hip): within each 16-lane
// group, the 16 lanes' 8-byte reads form a 64-element vector V (in lane
// order, 4 elems each); lane c receives V[c], V[c+16], V[c+32], V[c+48].
//
// For an MFMA bf16 B-fragment (lane l needs B[k=(l>>4)*8+j][n=l&15] from a
// row-major [32][16] panel), lane a=l&15 in group q=l>>4 supplies
//   addr_elems = panel + (q*8 + (a>>2))*16 + 4*(a&3)
// and the pair of reads (second at +128 B = +4 k-rows) yields the 8
// fragment elements in order. The waitcnt lives INSIDE the asm so the
// consuming MFMA (ordered by its register dependency) is safe.
// ---------------------------------------------------------------------------
typedef __bf16 tr_bf16x8 __attribute__((ext_vector_type(8)));

DEVINLINE tr_bf16x8 ds_read_tr16_frag(unsigned addr_bytes) {
  unsigned long long lo, hi;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %2 offset:128\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=v"(lo), "=v"(hi)
      : "v"(addr_bytes));
  union {
    struct { unsigned long long a, b; } u;
    tr_bf16x8 v;
  } cvt;
  cvt.u.a = lo;
  cvt.u.b = hi;
  return cvt.v;
}

// Per-lane address (in BYTES) for the fragment above: panel_base_bytes must
// be the LDS byte offset of a row-major [32][16] bf16 panel.
DEVINLINE unsigned tr16_frag_addr(unsigned panel_base_bytes, int lane) {
  const int a = lane & 15, q = lane >> 4;
  return panel_base_bytes +
         (unsigned)(((q * 8 + (a >> 2)) * 16 + 4 * (a & 3)) * 2);
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
             __LINE__);                                                     \
    }                                                                       \
  } while (0)
