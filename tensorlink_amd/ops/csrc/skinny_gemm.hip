// Skinny-M GEMM for the decode path: out[M,N] = x[M,K] @ W[N,K]^T (+bias).
//
// torch Linear stores W as [N, K] row-major, so both operands are
// k-contiguous — the same A·B^T fragment pattern as QK^T. At decode M
// (<= 256 rows) the GEMM is pure W-streaming; hipBLASLt measured only
// 1.7-3.5 TB/s on these shapes (profiles/).
//
// Geometry (v2 — v1 gridded over M too, which re-streamed every W panel
// once per M-tile through different XCD L2s: 4x HBM traffic, 7x slower
// than hipBLASLt): one block per 64-column W panel (grid.x = N/64,
// grid.y = SPLITK), 4 waves; wave w owns columns w*16 and loops ALL
// M-tiles in registers (acc[16][4] fp32), so every W byte is loaded
// exactly once chip-wide and each wave walks its 16 W rows sequentially
// along k (DRAM-friendly streams). x is tiny (M*K bf16, L2-resident) and
// re-read per wave. SPLITK slices K when N/64 alone can't fill the chip;
// fp32 partials are combined by a tiny second kernel.

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

// MT = max M-tiles (compile-time); runtime m_tiles <= MT.
template <int MT, int HAS_BIAS, int SPLIT>
__global__ __launch_bounds__(BLOCK) void skinny_gemm_kernel(
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
  const int k_per = SPLIT ? ((K / 32 + n_split - 1) / n_split) * 32 : K;
  const int k_begin = split * k_per;
  const int k_end = min(K, k_begin + k_per);

  f32x4 acc[MT];
#pragma unroll
  for (int m = 0; m < MT; ++m) acc[m] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const bf16* wrow = w + (int64_t)oc * K;   // this lane's W row

  for (int kc = k_begin + quad * 8; kc < k_end; kc += 32) {
    bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(wrow + kc);
    const bf16* xcol = x + kc;
    // fully unrolled with compile-time m: a runtime-indexed acc[m] would
    // allocate in scratch (guide §5.4 rule 20 — measured 5x+ slower)
#pragma unroll
    for (int m = 0; m < MT; ++m) {
      if (m < m_tiles) {
        const int row = min(m * 16 + col, M - 1);
        bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
            xcol + (int64_t)row * K);
        acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         acc[m], 0, 0, 0);
      }
    }
  }

  // epilogue: lane holds C[row=quad*4+r][col] per m-tile
  const float b = HAS_BIAS ? bf2f(bias[blockIdx.x * 64 + wave * 16 + col])
                           : 0.f;
#pragma unroll
  for (int m = 0; m < MT; ++m) {
    if (m >= m_tiles) break;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = m * 16 + quad * 4 + r;
      if (orow < M) {
        if (SPLIT)
          partial[((int64_t)split * M + orow) * N + blockIdx.x * 64 +
                  wave * 16 + col] = acc[m][r];
        else
          out[(int64_t)orow * N + blockIdx.x * 64 + wave * 16 + col] =
              f2bf(acc[m][r] + b);
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
