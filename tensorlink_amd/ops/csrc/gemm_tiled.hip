// Tiled serving GEMM for mid-M (65 <= M <= ~512+): out[M,N] = x[M,K] @
// W[N,K]^T (+bias). Companion to skinny_gemm.hip (M <= 64): SAME per-row
// accumulation semantics — K stepped in 64-element chunks from k_begin,
// two mfma_f32_16x16x32_bf16 per chunk in kk order, f32 accumulators,
// identical split-K policy/k_per and s-ordered reduce — so a row's result
// is bitwise IDENTICAL for any M and either kernel. That property is what
// lets chunked prefill, speculative verify, batched decode and plain
// generate produce equal tokens (the serving engine's exact-greedy
// guarantee); do not change the loop order here without changing
// skinny_gemm.hip to match.
//
// Differences from skinny v5, both perf-motivated (guide §5 decode-GEMM
// notes):
//  * W panels are ALSO staged through LDS by global_load_lds, making the
//    k-loop all-glds: hipcc's counted-vmcnt scheduling survives (an
//    ordinary VGPR-destination load next to a glds forces vmcnt(0) at its
//    use — the v5 mixing trap), and the W fragment read becomes a
//    conflict-reduced b128 like the x fragments.
//  * blockIdx.z tiles M in 256-row blocks, so serving batches beyond 256
//    rows (chunked-prefill chunks, big decode batches) stay on this
//    kernel.
// Grid (N/64, n_split, ceil(M/256)); block = 4 waves; wave w owns output
// columns [blk.x*64 + w*16, +16).

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

template <int MT, int HAS_BIAS, int SPLIT>
__global__ __launch_bounds__(BLOCK, 2) void gemm_tiled_kernel(
    const bf16* __restrict__ x,     // [M, K]
    const bf16* __restrict__ w,     // [N, K]
    const bf16* __restrict__ bias,  // [N] or null
    bf16* __restrict__ out,         // [M, N]
    float* __restrict__ partial,    // [SPLITK, M, N] when SPLIT
    int M, int N, int K, int n_split) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int quad = lane >> 4;
  const int oc = blockIdx.x * 64 + wave * 16 + col;  // output column
  const int m_base = blockIdx.z * (MT * 16);

  const int split = SPLIT ? blockIdx.y : 0;
  const int k_per = SPLIT ? ((K / 64 + n_split - 1) / n_split) * 64 : K;
  const int k_begin = split * k_per;
  const int k_end = min(K, k_begin + k_per);

  // ONE shared object (a second one de-pipelines glds: guide §5 trap 4a).
  // Layout: [2 buffers][MT*16 x-rows + 64 w-rows][64 k]; glds writes are
  // lane-linear so the bank swizzle lives on the SOURCE k-slot address
  // (slot p of row r holds logical slot p^(r&7)); fragment reads apply
  // the same XOR.
  __shared__ bf16 lds[2][MT * 16 + 64][64];

  f32x4 acc[MT];
#pragma unroll
  for (int m = 0; m < MT; ++m) acc[m] = (f32x4){0.f, 0.f, 0.f, 0.f};

  constexpr int GPW = MT > 1 ? MT / 2 : 1;  // x glds chunks per wave
  const int l8 = lane >> 3, c8 = lane & 7;

  // x rows chunk*8+l8 (clamped to M) and W rows wchunk*8+l8 of this
  // block's 64-column panel, 1 KB per instruction.
#define GT_GLDS_TILE(bufi, k0)                                              \
  if ((k0) < k_end) {                                                       \
    _Pragma("unroll") for (int g = 0; g < GPW; ++g) {                       \
      const int chunk = wave * GPW + g;                                     \
      if (chunk >= MT * 2) break; /* MT=1: wave-uniform */                  \
      const int row = chunk * 8 + l8;                                       \
      const int xr = min(m_base + row, M - 1);                              \
      const int kc = min((k0) + (c8 ^ (row & 7)) * 8, K - 8);               \
      auto gsrc = (const __attribute__((address_space(1))) void*)(          \
          x + (int64_t)xr * K + kc);                                        \
      auto ldst = (__attribute__((address_space(3))) void*)(                \
          &lds[bufi][chunk * 8][0]);                                        \
      __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);               \
    }                                                                       \
    _Pragma("unroll") for (int g = 0; g < 2; ++g) {                         \
      const int wchunk = wave * 2 + g;                                      \
      const int wr = wchunk * 8 + l8;                                       \
      const int kc = min((k0) + (c8 ^ (wr & 7)) * 8, K - 8);                \
      auto gsrc = (const __attribute__((address_space(1))) void*)(          \
          w + ((int64_t)blockIdx.x * 64 + wr) * K + kc);                    \
      auto ldst = (__attribute__((address_space(3))) void*)(                \
          &lds[bufi][MT * 16 + wchunk * 8][0]);                             \
      __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);               \
    }                                                                       \
  }

  GT_GLDS_TILE(0, k_begin);
  int buf = 0;
  for (int k0 = k_begin; k0 < k_end; k0 += 64) {
    __syncthreads();                          // drains this tile's glds
    if (k0 + 64 < k_end)
      GT_GLDS_TILE(buf ^ 1, k0 + 64);         // in flight under compute
    const int tile_k = min(64, k_end - k0);
    const int wrow = wave * 16 + col;         // this lane's w slab row
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      if (kk * 32 < tile_k) {
        const int wslot = (kk * 4 + quad) ^ (wrow & 7);
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
            &lds[buf][MT * 16 + wrow][wslot * 8]);
#pragma unroll
        for (int m = 0; m < MT; ++m) {
          const int row = m * 16 + col;
          const int slot = (kk * 4 + quad) ^ (row & 7);
          bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
              &lds[buf][row][slot * 8]);
          acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc[m], 0, 0, 0);
        }
      }
    }
    buf ^= 1;
  }
#undef GT_GLDS_TILE

  const float b = HAS_BIAS ? bf2f(bias[oc]) : 0.f;
#pragma unroll
  for (int m = 0; m < MT; ++m) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = m_base + m * 16 + quad * 4 + r;
      if (orow < M) {
        if (SPLIT)
          partial[((int64_t)split * M + orow) * N + oc] = acc[m][r];
        else
          out[(int64_t)orow * N + oc] = f2bf(acc[m][r] + b);
      }
    }
  }
}

template <int HAS_BIAS>
__global__ void gemm_reduce_kernel(const float* __restrict__ partial,
                                   const bf16* __restrict__ bias,
                                   bf16* __restrict__ out, int64_t MN,
                                   int N, int n_split) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= MN) return;
  float acc = 0.f;
  for (int s = 0; s < n_split; ++s) acc += partial[s * MN + i];
  if (HAS_BIAS) acc += bf2f(bias[i % N]);
  out[i] = f2bf(acc);
}

}  // namespace

extern "C" {

// partial: scratch [n_split * M * N] floats (only read when n_split > 1).
void tl_gemm_tiled(const void* x, const void* w, const void* bias, void* out,
                   void* partial, int M, int N, int K, int n_split,
                   hipStream_t stream) {
  const int m_tiles = (M + 15) / 16;
  const int split = n_split > 1 ? 1 : 0;
  const int MT = m_tiles <= 8 ? 8 : 16;
  const int n_mblk = cdiv(m_tiles, MT);
  dim3 grid(N / 64, n_split, n_mblk), block(BLOCK);
#define DISPATCH(MTV)                                                       \
  do {                                                                      \
    if (split) {                                                            \
      hipLaunchKernelGGL((gemm_tiled_kernel<MTV, 0, 1>), grid, block, 0,    \
                         stream, (const bf16*)x, (const bf16*)w, nullptr,   \
                         (bf16*)out, (float*)partial, M, N, K, n_split);    \
    } else if (bias) {                                                      \
      hipLaunchKernelGGL((gemm_tiled_kernel<MTV, 1, 0>), grid, block, 0,    \
                         stream, (const bf16*)x, (const bf16*)w,            \
                         (const bf16*)bias, (bf16*)out, nullptr, M, N, K,   \
                         1);                                                \
    } else {                                                                \
      hipLaunchKernelGGL((gemm_tiled_kernel<MTV, 0, 0>), grid, block, 0,    \
                         stream, (const bf16*)x, (const bf16*)w, nullptr,   \
                         (bf16*)out, nullptr, M, N, K, 1);                  \
    }                                                                       \
  } while (0)
  if (MT == 8) DISPATCH(8);
  else DISPATCH(16);
#undef DISPATCH
  if (split) {
    const int64_t MN = (int64_t)M * N;
    dim3 rgrid((uint32_t)((MN + 255) / 256)), rblock(256);
    if (bias)
      hipLaunchKernelGGL((gemm_reduce_kernel<1>), rgrid, rblock, 0, stream,
                         (const float*)partial, (const bf16*)bias,
                         (bf16*)out, MN, N, n_split);
    else
      hipLaunchKernelGGL((gemm_reduce_kernel<0>), rgrid, rblock, 0, stream,
                         (const float*)partial, nullptr, (bf16*)out, MN, N,
                         n_split);
  }
}

}  // extern "C"
