// Fused RoPE + KV-cache append for gfx950.
//
// One kernel consumes the fused-QKV GEMM output directly (strided rows:
// [T, q_size + 2*kv_size]): rotates q into a contiguous [T, Hq, D] buffer,
// rotates k and writes it into the KV cache at its position, and copies v
// into the cache — replacing a rope kernel + two cache scatters + three
// .contiguous() splits per layer per step. (The reference instead
// re-serializes the whole DynamicCache over TCP per step —
// tensorlink/ml/utils.py:210-221.)
//
// qkv [T, row_stride] bf16 with q at offset 0, k at q_size, v at
// q_size+kv_size; caches [B, Hkv, Smax, D] with T = B*S tokens; positions
// [T] int32 gives both the RoPE angle and the cache slot.

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

// PAGED != 0: caches are page pools [n_pages, Hkv, 128, D] addressed via
// block_table [B, bt_stride].
template <int PAGED>
__global__ __launch_bounds__(BLOCK) void rope_append_kernel(
    const bf16* __restrict__ qkv, bf16* __restrict__ q_out,
    bf16* __restrict__ k_cache, bf16* __restrict__ v_cache,
    const int* __restrict__ positions, const float* __restrict__ inv_freq,
    const int* __restrict__ block_table, int row_stride, int S, int Hq,
    int Hkv, int D, int Smax, int bt_stride) {
  const int64_t token = blockIdx.x;
  const int b = token / S;
  const int pos = positions[token];
  int64_t slot;   // cache slot index (in units of D elems, per kv head 0)
  if (PAGED) {
    const int page = block_table[b * bt_stride + (pos >> 7)];
    slot = ((int64_t)page * Hkv) * 128 + (pos & 127);
  } else {
    slot = ((int64_t)b * Hkv) * Smax + pos;
  }
  const int64_t head_stride = PAGED ? 128 : Smax;
  const int D2 = D / 2;
  const float fpos = (float)pos;
  const bf16* row = qkv + token * row_stride;
  const bf16* krow = row + Hq * D;
  const bf16* vrow = krow + Hkv * D;

  // rotate q (Hq heads) -> contiguous q_out; rotate k -> cache
  const int total = (Hq + Hkv) * D2;
  for (int idx = threadIdx.x; idx < total; idx += BLOCK) {
    const int h = idx / D2;
    const int d = idx - h * D2;
    float c, s;
    __sincosf(fpos * inv_freq[d], &s, &c);
    if (h < Hq) {
      const bf16* src = row + h * D;
      const float x1 = bf2f(src[d]);
      const float x2 = bf2f(src[d + D2]);
      bf16* dst = q_out + (token * Hq + h) * D;
      dst[d] = f2bf(x1 * c - x2 * s);
      dst[d + D2] = f2bf(x2 * c + x1 * s);
    } else {
      const int hk = h - Hq;
      const bf16* src = krow + hk * D;
      const float x1 = bf2f(src[d]);
      const float x2 = bf2f(src[d + D2]);
      bf16* kc = k_cache + (slot + hk * head_stride) * D;
      kc[d] = f2bf(x1 * c - x2 * s);
      kc[d + D2] = f2bf(x2 * c + x1 * s);
    }
  }
  // copy v into cache (no rotation), vectorized
  const int vtotal = Hkv * D / 8;
  for (int idx = threadIdx.x; idx < vtotal; idx += BLOCK) {
    const int h = idx / (D / 8);
    const int d8 = idx - h * (D / 8);
    const bf16x8 val =
        *reinterpret_cast<const bf16x8*>(vrow + h * D + d8 * 8);
    reinterpret_cast<bf16x8*>(
        v_cache + (slot + h * head_stride) * D)[d8] = val;
  }
}

}  // namespace

extern "C" {

void tl_rope_append(const void* qkv, void* q_out, void* k_cache,
                    void* v_cache, const void* positions,
                    const void* inv_freq, const void* block_table,
                    int64_t T, int row_stride, int S, int Hq, int Hkv,
                    int D, int Smax, int bt_stride, hipStream_t stream) {
  dim3 grid((uint32_t)T), block(BLOCK);
  if (block_table)
    hipLaunchKernelGGL((rope_append_kernel<1>), grid, block, 0, stream,
                       (const bf16*)qkv, (bf16*)q_out, (bf16*)k_cache,
                       (bf16*)v_cache, (const int*)positions,
                       (const float*)inv_freq, (const int*)block_table,
                       row_stride, S, Hq, Hkv, D, Smax, bt_stride);
  else
    hipLaunchKernelGGL((rope_append_kernel<0>), grid, block, 0, stream,
                       (const bf16*)qkv, (bf16*)q_out, (bf16*)k_cache,
                       (bf16*)v_cache, (const int*)positions,
                       (const float*)inv_freq, nullptr, row_stride, S, Hq,
                       Hkv, D, Smax, bt_stride);
}

}  // extern "C"
