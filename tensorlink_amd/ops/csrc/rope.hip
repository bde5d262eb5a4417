// Rotary position embedding (RoPE) apply, in place, for gfx950.
//
// The reference ships HF rotary cos/sin buffers over the network as
// "required_buffers" (tensorlink/ml/module.py:1320-1422,
// ml/worker.py:1192-1266); here cos/sin are computed on device from
// inv_freq — no tables move, ever.
//
// Convention: HF rotate_half — pair (x[d], x[d + D/2]) for d < D/2:
//   x1' = x1*cos - x2*sin ; x2' = x2*cos + x1*sin
// Backward of RoPE is RoPE with sin negated (rotation transpose), selected
// by `sign`.
//
// q: [T, Hq, D] bf16, k: [T, Hkv, D] bf16 (T = B*S flattened tokens),
// positions: [T] int32, inv_freq: [D/2] float.

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

__global__ __launch_bounds__(BLOCK) void rope_kernel(
    bf16* __restrict__ q, bf16* __restrict__ k,
    const int* __restrict__ positions, const float* __restrict__ inv_freq,
    int64_t T, int Hq, int Hkv, int D, float sign) {
  const int64_t token = blockIdx.x;
  if (token >= T) return;
  const float pos = (float)positions[token];
  const int D2 = D / 2;
  const int Htot = Hq + Hkv;
  // each thread handles one (head, pair) element; stride over all pairs
  for (int idx = threadIdx.x; idx < Htot * D2; idx += BLOCK) {
    const int h = idx / D2;
    const int d = idx - h * D2;
    float c, s;
    __sincosf(pos * inv_freq[d], &s, &c);
    s *= sign;
    bf16* base = (h < Hq) ? q + (token * Hq + h) * D
                          : k + (token * Hkv + (h - Hq)) * D;
    const float x1 = bf2f(base[d]);
    const float x2 = bf2f(base[d + D2]);
    base[d] = f2bf(x1 * c - x2 * s);
    base[d + D2] = f2bf(x2 * c + x1 * s);
  }
}

}  // namespace

extern "C" {

void tl_rope(void* q, void* k, const void* positions, const void* inv_freq,
             int64_t T, int Hq, int Hkv, int D, float sign,
             hipStream_t stream) {
  dim3 grid((uint32_t)T), block(BLOCK);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, stream, (bf16*)q, (bf16*)k,
                     (const int*)positions, (const float*)inv_freq, T, Hq,
                     Hkv, D, sign);
}

}  // extern "C"
