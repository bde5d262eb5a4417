// Grouped expert GEMM for MoE serving (gfx950).
//
// Replaces the per-expert Python loop (round-1 models/dense.py MoEMLP /
// models/quant.py): ONE launch computes out[p] = x[tok_idx[p]] @
// W[e]^T for every (token, expert-slot) pair p, where pairs are sorted
// by expert into segments (seg_off[e]..seg_off[e+1]). Expert weights
// are addressed through a device pointer table — no stacked copy — and
// may be bf16 or fp8 e4m3 (OCP): fp8 rows are dequantized in-kernel
// through a 256-entry LDS LUT and the per-output-channel scale is
// applied in the f32 epilogue (MORE accurate than scaling the fp8
// operand, and unlike torch._scaled_mm it is hipGraph-capture-safe,
// which re-enables captured decode for quantized MoE models).
//
// Structure: streaming-GEMM family member (skinny_gemm.hip lineage):
// grid (N/64, E); block = 4 waves; per 64-row m-chunk the x slab stages
// through LDS by global_load_lds (indirect rows via tok_idx) and W
// streams from HBM; K accumulates in 64-chunks, two
// mfma_f32_16x16x32_bf16 per chunk in kk order (family contract).

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

// fp8 e4m3 (OCP) -> f32 for the LUT (host-independent, computed on
// device at kernel start: 256 entries)
DEVINLINE float e4m3_to_f32(int b) {
  const int s = (b >> 7) & 1;
  const int e = (b >> 3) & 0xF;
  const int m = b & 0x7;
  float v;
  if (e == 0) {
    v = (m / 8.0f) * 0.015625f;            // subnormal: m/8 * 2^-6
  } else if (e == 15 && m == 7) {
    v = __int_as_float(0x7FC00000);        // NaN (e4m3fn: only S.1111.111)
  } else {
    v = (1.0f + m / 8.0f) * exp2f((float)(e - 7));
  }
  return s ? -v : v;
}

template <int FP8>
__global__ __launch_bounds__(BLOCK, 2) void moe_gemm_kernel(
    const bf16* __restrict__ x,          // [T, K] (or [P, K] pair rows)
    const int* __restrict__ tok_idx,     // [P] row of x per pair, or null
    const int* __restrict__ seg_off,     // [E+1] pair segments by expert
    const uint64_t* __restrict__ w_ptrs, // [E] -> weight [N, K]
    const uint64_t* __restrict__ s_ptrs, // [E] -> fp32 scale [N] (FP8)
    bf16* __restrict__ out,              // [P, N]
    int N, int K) {
  const int e = blockIdx.y;
  const int m0 = seg_off[e];
  const int m_tot = seg_off[e + 1] - m0;
  if (m_tot <= 0) return;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int quad = lane >> 4;
  const int oc = blockIdx.x * 64 + wave * 16 + col;   // output column

  // ONE shared object: [2 buffers][64 rows][64 k] x slab + fp8 LUT
  __shared__ bf16 lds[2][64][64];
  __shared__ float f8lut[256];
  if (FP8) {
    for (int i = threadIdx.x; i < 256; i += BLOCK)
      f8lut[i] = e4m3_to_f32(i);
  }
  __syncthreads();

  const uint64_t wp = w_ptrs[e];
  const bf16* wrow_bf = FP8 ? nullptr
      : reinterpret_cast<const bf16*>(wp) + (int64_t)oc * K;
  const uint8_t* wrow_f8 = FP8
      ? reinterpret_cast<const uint8_t*>(wp) + (int64_t)oc * K : nullptr;
  const float wscale = FP8
      ? reinterpret_cast<const float*>(s_ptrs[e])[oc] : 1.f;

  const int l8 = lane >> 3, c8 = lane & 7;

  // outer loop over 64-row m-chunks of this expert's segment (W panels
  // re-read per chunk stay L2-resident — same block, back to back)
  for (int mb = 0; mb < m_tot; mb += 64) {
    const int mrows = min(64, m_tot - mb);
    f32x4 acc[4];
#pragma unroll
    for (int m = 0; m < 4; ++m) acc[m] = (f32x4){0.f, 0.f, 0.f, 0.f};

    // x glds: 8 chunks of 8 rows -> 2 per wave; source row indirect
#define MG_GLDS_TILE(bufi, k0)                                              \
    if ((k0) < K) {                                                         \
      _Pragma("unroll") for (int g = 0; g < 2; ++g) {                       \
        const int chunk = wave * 2 + g;                                     \
        const int row = chunk * 8 + l8;                                     \
        const int p = m0 + mb + min(row, mrows - 1);                        \
        const int xr = tok_idx ? tok_idx[p] : p;                            \
        const int kc = min((k0) + (c8 ^ (row & 7)) * 8, K - 8);             \
        auto gsrc = (const __attribute__((address_space(1))) void*)(        \
            x + (int64_t)xr * K + kc);                                      \
        auto ldst = (__attribute__((address_space(3))) void*)(              \
            &lds[bufi][chunk * 8][0]);                                      \
        __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);             \
      }                                                                     \
    }

    MG_GLDS_TILE(0, 0);
    int buf = 0;
    for (int k0 = 0; k0 < K; k0 += 64) {
      __syncthreads();
      if (k0 + 64 < K) MG_GLDS_TILE(buf ^ 1, k0 + 64);
      const int tile_k = min(64, K - k0);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        if (kk * 32 < tile_k) {
          bf16x8_t bfrag;
          if (FP8) {
            // 8 fp8 bytes -> bf16x8 via LDS LUT (scale in epilogue)
            const uint8_t* src = wrow_f8 + k0 + kk * 32 + quad * 8;
            uint64_t raw = *reinterpret_cast<const uint64_t*>(src);
#pragma unroll
            for (int j = 0; j < 8; ++j)
              bfrag[j] = (__bf16)f8lut[(raw >> (8 * j)) & 0xFF];
          } else {
            bfrag = *reinterpret_cast<const bf16x8_t*>(
                wrow_bf + k0 + kk * 32 + quad * 8);
          }
#pragma unroll
          for (int m = 0; m < 4; ++m) {
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
#undef MG_GLDS_TILE

#pragma unroll
    for (int m = 0; m < 4; ++m) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int lrow = m * 16 + quad * 4 + r;
        if (lrow < mrows) {
          out[(int64_t)(m0 + mb + lrow) * N + oc] =
              f2bf(acc[m][r] * wscale);
        }
      }
    }
    // next m-chunk restarts the glds pipeline; drain + resync
    __syncthreads();
  }
}

}  // namespace

extern "C" {

void tl_moe_gemm(const void* x, const void* tok_idx, const void* seg_off,
                 const void* w_ptrs, const void* s_ptrs, void* out, int E,
                 int N, int K, int fp8, hipStream_t stream) {
  dim3 grid(N / 64, E), block(BLOCK);
  if (fp8)
    hipLaunchKernelGGL((moe_gemm_kernel<1>), grid, block, 0, stream,
                       (const bf16*)x, (const int*)tok_idx,
                       (const int*)seg_off, (const uint64_t*)w_ptrs,
                       (const uint64_t*)s_ptrs, (bf16*)out, N, K);
  else
    hipLaunchKernelGGL((moe_gemm_kernel<0>), grid, block, 0, stream,
                       (const bf16*)x, (const int*)tok_idx,
                       (const int*)seg_off, (const uint64_t*)w_ptrs,
                       nullptr, (bf16*)out, N, K);
}

}  // extern "C"
