// Skinny-M GEMM for the decode path: out[M,N] = x[M,K] @ W[N,K]^T (+bias).
//
// torch Linear stores W as [N, K] row-major, so both operands are
// k-contiguous — the same A·B^T fragment pattern as QK^T: the MFMA
// B-fragment (lane l = 16 B of row l&15 at k-offset (l>>4)*8) reads W rows
// straight from HBM, no LDS staging. At decode M (<= 512 rows) the GEMM is
// pure W-streaming: hipBLASLt's picks measured only 1.7-3.5 TB/s on these
// shapes (profiles/); this kernel's goal is the HBM roofline.
//
// Geometry: block = 4 waves; tile M=64 (wave w owns rows w*16) x N=64
// (4 n-subtiles per wave; all waves share the W panel through L1).
// Grid (ceil(M/64), N/64). K-loop unrolled 2x32. M tail masked; x rows
// beyond M contribute garbage*0 via masked epilogue writes only (A-frag
// loads are clamped).

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

template <int HAS_BIAS>
__global__ __launch_bounds__(BLOCK) void skinny_gemm_kernel(
    const bf16* __restrict__ x,     // [M, K]
    const bf16* __restrict__ w,     // [N, K]
    const bf16* __restrict__ bias,  // [N] or null
    bf16* __restrict__ out,         // [M, N]
    int M, int N, int K) {
  const int m_base = blockIdx.x * 64;
  const int n_base = blockIdx.y * 64;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int quad = lane >> 4;

  const int row = m_base + wave * 16 + col;      // A-fragment row
  const int arow = min(row, M - 1);              // clamped load row

  f32x4 acc[4];
#pragma unroll
  for (int n = 0; n < 4; ++n) acc[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const bf16* xrow = x + (int64_t)arow * K;
  const bf16* wbase = w + (int64_t)n_base * K;

  int k0 = 0;
  for (; k0 + 64 <= K; k0 += 64) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int kc = k0 + kk * 32 + quad * 8;
      bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(xrow + kc);
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
            wbase + (int64_t)(n * 16 + col) * K + kc);
        acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         acc[n], 0, 0, 0);
      }
    }
  }
  for (; k0 < K; k0 += 32) {  // K % 64 == 32 tail
    const int kc = k0 + quad * 8;
    bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(xrow + kc);
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
          wbase + (int64_t)(n * 16 + col) * K + kc);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[n],
                                                       0, 0, 0);
    }
  }

  // epilogue: lane holds C[row=quad*4+r][col] per n-subtile
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    const int oc = n_base + n * 16 + col;
    float b = 0.f;
    if (HAS_BIAS) b = bf2f(bias[oc]);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int orow = m_base + wave * 16 + quad * 4 + r;
      if (orow < M)
        out[(int64_t)orow * N + oc] = f2bf(acc[n][r] + b);
    }
  }
}

}  // namespace

extern "C" {

void tl_skinny_gemm(const void* x, const void* w, const void* bias,
                    void* out, int M, int N, int K, hipStream_t stream) {
  dim3 grid((M + 63) / 64, N / 64), block(BLOCK);
  if (bias)
    hipLaunchKernelGGL((skinny_gemm_kernel<1>), grid, block, 0, stream,
                       (const bf16*)x, (const bf16*)w, (const bf16*)bias,
                       (bf16*)out, M, N, K);
  else
    hipLaunchKernelGGL((skinny_gemm_kernel<0>), grid, block, 0, stream,
                       (const bf16*)x, (const bf16*)w, nullptr, (bf16*)out,
                       M, N, K);
}

}  // extern "C"
