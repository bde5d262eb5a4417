// Skinny-M GEMM for the decode path: out[M,N] = x[M,K] @ W[N,K]^T (+bias).
//
// torch Linear stores W as [N, K] row-major, so both operands are
// k-contiguous — the same A·B^T fragment pattern as QK^T. At decode M
// (<= 256 rows) the GEMM is pure W-streaming.
//
// v3 geometry (measured on MI355X):
//   v1 gridded over M: every W panel re-streamed per M-tile -> 0.4 TB/s.
//   v2 looped M-tiles with fragment-shaped x loads straight from global:
//   16 scatter-line loads per W load -> TA transaction-bound, 0.23-0.37
//   TB/s (the guide's "fragment-shaped x" trap).
//   v3 stages the x k-slab into LDS ONCE per block with full-line
//   coalesced loads (+16 B row pad -> conflict-free ds_read_b128
//   A-fragments), shared by all 4 waves x all M-tiles; W streams straight
//   from HBM exactly once chip-wide.
// Grid (N/64, SPLITK); block = 4 waves; wave w owns columns w*16; SPLITK
// slices K when N/64 can't fill the chip (fp32 partials + reduce kernel).

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

template <int MT, int HAS_BIAS, int SPLIT>
__global__ __launch_bounds__(BLOCK, 2) void skinny_gemm_kernel(
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
  const int oc = blockIdx.x * 64 + wave * 16 + col;   // output column

  const int m_tiles = (M + 15) / 16;
  const int split = SPLIT ? blockIdx.y : 0;
  const int k_per = SPLIT ? ((K / 64 + n_split - 1) / n_split) * 64 : K;
  const int k_begin = split * k_per;
  const int k_end = min(K, k_begin + k_per);

  // Double-buffered x slab staged by global_load_lds (HBM->LDS DMA: no
  // staging registers — register staging at MT=16 spilled 272 B/lane).
  // glds writes lane-linear (base + lane*16), so the LDS image is
  // unpadded; the bank swizzle moves to the SOURCE address (guide rule
  // 21): physical 16-B slot p of row r holds logical k-slot p^(r&7), and
  // the fragment read applies the same XOR — 8-way read conflicts drop to
  // 2-way with FETCH-neutral sources (the XOR permutes within one row's
  // 128 B).
  __shared__ bf16 x_lds[2][MT * 16][64];

  f32x4 acc[MT];
#pragma unroll
  for (int m = 0; m < MT; ++m) acc[m] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const bf16* wrow = w + (int64_t)oc * K;   // this lane's W row

  constexpr int GPW = MT > 1 ? MT / 2 : 1;  // glds chunks per wave
  const int l8 = lane >> 3, c8 = lane & 7;

#define SG_GLDS_TILE(bufi, k0)                                              \
  if ((k0) < K) {                                                           \
    _Pragma("unroll") for (int g = 0; g < GPW; ++g) {                       \
      const int chunk = wave * GPW + g;      /* 8 rows per 1-KB chunk */    \
      if (chunk >= MT * 2) break; /* MT=1: only 2 chunks (wave-uniform) */  \
      const int row = chunk * 8 + l8;                                       \
      const int xr = min(row, M - 1);                                       \
      const int kc = min((k0) + (c8 ^ (row & 7)) * 8, K - 8);               \
      auto gsrc = (const __attribute__((address_space(1))) void*)(          \
          x + (int64_t)xr * K + kc);                                        \
      auto ldst = (__attribute__((address_space(3))) void*)(                \
          &x_lds[bufi][chunk * 8][0]);                                      \
      __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);               \
    }                                                                       \
  }

  SG_GLDS_TILE(0, k_begin);
  int buf = 0;
  for (int k0 = k_begin; k0 < k_end; k0 += 64) {
    __syncthreads();                          // drains this tile's glds
    if (k0 + 64 < k_end)
      SG_GLDS_TILE(buf ^ 1, k0 + 64);         // in flight under compute
    const int tile_k = min(64, k_end - k0);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      if (kk * 32 < tile_k) {
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
            wrow + k0 + kk * 32 + quad * 8);
#pragma unroll
        for (int m = 0; m < MT; ++m) {
          // unconditional across MT (dispatch picks smallest MT >=
          // m_tiles; guarded acc updates made the compiler home the
          // accumulator array in scratch)
          const int row = m * 16 + col;
          const int slot = (kk * 4 + quad) ^ (row & 7);
          bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
              &x_lds[buf][row][slot * 8]);
          acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc[m], 0, 0, 0);
        }
      }
    }
    buf ^= 1;
  }
#undef SG_GLDS_TILE

  // epilogue: lane holds C[row=quad*4+r][col] per m-tile
  const float b = HAS_BIAS ? bf2f(bias[oc]) : 0.f;
#pragma unroll
  for (int m = 0; m < MT; ++m) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = m * 16 + quad * 4 + r;
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
__global__ void skinny_reduce_kernel(const float* __restrict__ partial,
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
void tl_skinny_gemm(const void* x, const void* w, const void* bias,
                    void* out, void* partial, int M, int N, int K,
                    int n_split, hipStream_t stream) {
  const int m_tiles = (M + 15) / 16;
  const int split = n_split > 1 ? 1 : 0;
  dim3 grid(N / 64, n_split), block(BLOCK);
#define DISPATCH(MT)                                                        \
  do {                                                                      \
    if (split) {                                                            \
      hipLaunchKernelGGL((skinny_gemm_kernel<MT, 0, 1>), grid, block, 0,    \
                         stream, (const bf16*)x, (const bf16*)w, nullptr,   \
                         (bf16*)out, (float*)partial, M, N, K, n_split);    \
    } else if (bias) {                                                      \
      hipLaunchKernelGGL((skinny_gemm_kernel<MT, 1, 0>), grid, block, 0,    \
                         stream, (const bf16*)x, (const bf16*)w,            \
                         (const bf16*)bias, (bf16*)out, nullptr, M, N, K,   \
                         1);                                                \
    } else {                                                                \
      hipLaunchKernelGGL((skinny_gemm_kernel<MT, 0, 0>), grid, block, 0,    \
                         stream, (const bf16*)x, (const bf16*)w, nullptr,   \
                         (bf16*)out, nullptr, M, N, K, 1);                  \
    }                                                                       \
  } while (0)
  if (m_tiles <= 1) DISPATCH(1);
  else if (m_tiles <= 2) DISPATCH(2);
  else if (m_tiles <= 4) DISPATCH(4);
  else if (m_tiles <= 8) DISPATCH(8);
  else DISPATCH(16);
#undef DISPATCH
  if (split) {
    const int64_t MN = (int64_t)M * N;
    dim3 rgrid((uint32_t)((MN + 255) / 256)), rblock(256);
    if (bias)
      hipLaunchKernelGGL((skinny_reduce_kernel<1>), rgrid, rblock, 0, stream,
                         (const float*)partial, (const bf16*)bias,
                         (bf16*)out, MN, N, n_split);
    else
      hipLaunchKernelGGL((skinny_reduce_kernel<0>), rgrid, rblock, 0, stream,
                         (const float*)partial, nullptr, (bf16*)out, MN, N,
                         n_split);
  }
}

}  // extern "C"
