// Fused RoPE + KV-cache append for gfx950.
//
// One kernel: rotates q in place, rotates k and writes it into the KV cache
// at its position, copies v into the cache — removing the separate rope
// kernel + two scatter copies per layer per decode step. (The reference
// instead re-serializes the whole DynamicCache over TCP per step —
// tensorlink/ml/utils.py:210-221.)
//
// q [T, Hq, D] (in/out), k/v [T, Hkv, D] (k rotated in place too, so the
// prefill attention kernel can consume it directly), caches
// [B, Hkv, Smax, D] with T = B*S tokens, token t belongs to batch t/S.
// positions [T] int32 gives both the RoPE angle and the cache slot.

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

__global__ __launch_bounds__(BLOCK) void rope_append_kernel(
    bf16* __restrict__ q, bf16* __restrict__ k, const bf16* __restrict__ v,
    bf16* __restrict__ k_cache, bf16* __restrict__ v_cache,
    const int* __restrict__ positions, const float* __restrict__ inv_freq,
    int S, int Hq, int Hkv, int D, int Smax) {
  const int64_t token = blockIdx.x;
  const int b = token / S;
  const int pos = positions[token];
  const int D2 = D / 2;
  const float fpos = (float)pos;

  // rotate q (Hq heads) and k (Hkv heads); write k,v to cache
  const int total = (Hq + Hkv) * D2;
  for (int idx = threadIdx.x; idx < total; idx += BLOCK) {
    const int h = idx / D2;
    const int d = idx - h * D2;
    float c, s;
    __sincosf(fpos * inv_freq[d], &s, &c);
    if (h < Hq) {
      bf16* base = q + (token * Hq + h) * D;
      const float x1 = bf2f(base[d]);
      const float x2 = bf2f(base[d + D2]);
      base[d] = f2bf(x1 * c - x2 * s);
      base[d + D2] = f2bf(x2 * c + x1 * s);
    } else {
      const int hk = h - Hq;
      bf16* kbase = k + (token * Hkv + hk) * D;
      const float x1 = bf2f(kbase[d]);
      const float x2 = bf2f(kbase[d + D2]);
      const bf16 r1 = f2bf(x1 * c - x2 * s);
      const bf16 r2 = f2bf(x2 * c + x1 * s);
      kbase[d] = r1;
      kbase[d + D2] = r2;
      bf16* kc = k_cache + (((int64_t)b * Hkv + hk) * Smax + pos) * D;
      kc[d] = r1;
      kc[d + D2] = r2;
    }
  }
  // copy v into cache (no rotation), vectorized
  const int vtotal = Hkv * D / 8;
  for (int idx = threadIdx.x; idx < vtotal; idx += BLOCK) {
    const int h = idx / (D / 8);
    const int d8 = idx - h * (D / 8);
    const bf16x8 val = reinterpret_cast<const bf16x8*>(
        v + (token * Hkv + h) * D)[d8];
    reinterpret_cast<bf16x8*>(
        v_cache + (((int64_t)b * Hkv + h) * Smax + pos) * D)[d8] = val;
  }
}

}  // namespace

extern "C" {

void tl_rope_append(void* q, void* k, const void* v, void* k_cache,
                    void* v_cache, const void* positions,
                    const void* inv_freq, int64_t T, int S, int Hq, int Hkv,
                    int D, int Smax, hipStream_t stream) {
  dim3 grid((uint32_t)T), block(BLOCK);
  hipLaunchKernelGGL(rope_append_kernel, grid, block, 0, stream, (bf16*)q,
                     (bf16*)k, (const bf16*)v, (bf16*)k_cache,
                     (bf16*)v_cache, (const int*)positions,
                     (const float*)inv_freq, S, Hq, Hkv, D, Smax);
}

}  // extern "C"
