// Causal GQA flash-attention forward (prefill) on gfx950 MFMA.
//
// Replaces the reference's HF SDPA-inside-eager-blocks prefill
// (tensorlink/ml/worker.py:330-335). Tiled for CDNA4: 64-query-row blocks,
// 64-key tiles staged through LDS, QK^T and P·V on
// v_mfma_f32_16x16x32_bf16, online softmax with fp32 running stats.
//
// v1 structure (correctness-first): scores round-trip through LDS for the
// softmax phase; V is transposed into LDS at load time so the P·V B-fragment
// reads are contiguous ds_read_b128. In-register softmax / tr_b16 reads are
// later optimizations (guide T10/T12).
//
// q [B,S,Hq,D], k,v [B,S,Hkv,D] bf16 row-major; out [B,S,Hq,D].

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;   // 4 waves
constexpr int BM = 64;       // query rows per block
constexpr int BN = 64;       // key cols per tile
constexpr int PPAD = 8;      // LDS padding for P tile

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x4_t = __attribute__((ext_vector_type(4))) __bf16;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

// mfma_f32_16x16x32_bf16 fragment maps (A 16x32, B 32x16, C 16x16):
//   A: lane l holds A[l&15][(l>>4)*8 + j], j=0..7   (8 consecutive k)
//   B: lane l holds B[(l>>4)*8 + j][l&15]
//   C: lane l reg r holds C[(l>>4)*4 + r][l&15]
template <int D>
__global__ __launch_bounds__(BLOCK) void prefill_attn_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out, int B, int S, int Hq,
    int Hkv, float scale, int causal) {
  static_assert(D == 64 || D == 128);
  constexpr int NSUB_PV = D / 16;  // MFMA col tiles for PV (output cols)

  __shared__ bf16 q_lds[BM][D];
  __shared__ bf16 k_lds[BN][D];
  __shared__ bf16 vt_lds[D][BN + 8];   // V transposed [d][n]; +8 keeps
                                       // 16-B alignment of b128 reads
  __shared__ float s_lds[BM][BN];
  __shared__ bf16 p_lds[BM][BN + PPAD];
  __shared__ float m_lds[BM], l_lds[BM], alpha_lds[BM];

  const int qt = blockIdx.x;           // query tile index
  const int h = blockIdx.y;            // query head
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);
  const int q0 = qt * BM;              // first query row (global)

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int r0 = wave * 16;            // wave's first row within tile

  // ---- load Q tile [BM][D] ----
  {
    const int64_t qbase = ((int64_t)b * S) * Hq * D + (int64_t)h * D;
    for (int i = threadIdx.x * 8; i < BM * D; i += BLOCK * 8) {
      const int row = i / D, col = i % D;
      bf16x8 val;
      if (q0 + row < S)
        val = *reinterpret_cast<const bf16x8*>(
            q + qbase + (int64_t)(q0 + row) * Hq * D + col);
      else
#pragma unroll
        for (int j = 0; j < 8; ++j) val.v[j] = f2bf(0.f);
      *reinterpret_cast<bf16x8*>(&q_lds[row][col]) = val;
    }
  }
  if (threadIdx.x < BM) {
    m_lds[threadIdx.x] = -1e30f;
    l_lds[threadIdx.x] = 0.f;
  }
  __syncthreads();

  // O accumulators: wave's 16 rows x D cols -> NSUB_PV x 4 regs per lane
  f32x4 oacc[NSUB_PV];
#pragma unroll
  for (int i = 0; i < NSUB_PV; ++i) oacc[i] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(S, q0 + BM) : S;
  const int64_t kbase = ((int64_t)b * S) * Hkv * D + (int64_t)hkv * D;

  for (int kt0 = 0; kt0 < kv_end; kt0 += BN) {
    // ---- load K tile [BN][D]; V transposed into vt_lds[D][BN] ----
    for (int i = threadIdx.x * 8; i < BN * D; i += BLOCK * 8) {
      const int row = i / D, col = i % D;
      bf16x8 kval, vval;
      if (kt0 + row < S) {
        const int64_t off = kbase + (int64_t)(kt0 + row) * Hkv * D + col;
        kval = *reinterpret_cast<const bf16x8*>(k + off);
        vval = *reinterpret_cast<const bf16x8*>(v + off);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) { kval.v[j] = f2bf(0.f); vval.v[j] = f2bf(0.f); }
      }
      *reinterpret_cast<bf16x8*>(&k_lds[row][col]) = kval;
#pragma unroll
      for (int j = 0; j < 8; ++j) vt_lds[col + j][row] = vval.v[j];
    }
    __syncthreads();

    // ---- S = Q K^T for wave's 16 rows x BN cols ----
#pragma unroll
    for (int ns = 0; ns < BN / 16; ++ns) {
      f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < D / 32; ++ks) {
        bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
            &q_lds[r0 + (lane & 15)][ks * 32 + (lane >> 4) * 8]);
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
            &k_lds[ns * 16 + (lane & 15)][ks * 32 + (lane >> 4) * 8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r)
        s_lds[r0 + (lane >> 4) * 4 + r][ns * 16 + (lane & 15)] = acc[r];
    }
    __syncthreads();

    // ---- softmax phase: 4 threads per row, 16 cols each ----
    {
      const int row = threadIdx.x >> 2;          // [0, 64)
      const int c0 = (threadIdx.x & 3) * 16;
      const int qrow = q0 + row;
      float mx = -1e30f;
      float sc[16];
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const int kcol = kt0 + c0 + j;
        float val = s_lds[row][c0 + j] * scale;
        const bool masked = (kcol >= S) || (causal && kcol > qrow);
        sc[j] = masked ? -1e30f : val;
        mx = fmaxf(mx, sc[j]);
      }
      // row reduce across the 4 sibling threads (lanes differ in bits 0..1)
      mx = fmaxf(mx, __shfl_xor(mx, 1, WAVE_SIZE));
      mx = fmaxf(mx, __shfl_xor(mx, 2, WAVE_SIZE));
      const float m_old = m_lds[row];
      const float m_new = fmaxf(m_old, mx);
      float lsum = 0.f;
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const float p = (sc[j] <= -1e29f) ? 0.f : __expf(sc[j] - m_new);
        p_lds[row][c0 + j] = f2bf(p);
        lsum += p;
      }
      lsum += __shfl_xor(lsum, 1, WAVE_SIZE);
      lsum += __shfl_xor(lsum, 2, WAVE_SIZE);
      if ((threadIdx.x & 3) == 0) {
        const float alpha = __expf(m_old - m_new);
        alpha_lds[row] = alpha;
        l_lds[row] = l_lds[row] * alpha + lsum;
        m_lds[row] = m_new;
      }
    }
    __syncthreads();

    // ---- O = O*alpha + P V ----
    {
      float al[4];
#pragma unroll
      for (int r = 0; r < 4; ++r)
        al[r] = alpha_lds[r0 + (lane >> 4) * 4 + r];
#pragma unroll
      for (int ns = 0; ns < NSUB_PV; ++ns) {
#pragma unroll
        for (int r = 0; r < 4; ++r) oacc[ns][r] *= al[r];
#pragma unroll
        for (int ks = 0; ks < BN / 32; ++ks) {
          bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
              &p_lds[r0 + (lane & 15)][ks * 32 + (lane >> 4) * 8]);
          bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
              &vt_lds[ns * 16 + (lane & 15)][ks * 32 + (lane >> 4) * 8]);
          oacc[ns] =
              __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, oacc[ns], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // before next tile overwrites K/V/P
  }

  // ---- epilogue: divide by l, store ----
  {
    float linv[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float lv = l_lds[r0 + (lane >> 4) * 4 + r];
      linv[r] = lv > 0.f ? 1.f / lv : 0.f;
    }
    const int64_t obase = ((int64_t)b * S) * Hq * D + (int64_t)h * D;
#pragma unroll
    for (int ns = 0; ns < NSUB_PV; ++ns) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = r0 + (lane >> 4) * 4 + r;
        if (q0 + row < S)
          out[obase + (int64_t)(q0 + row) * Hq * D + ns * 16 + (lane & 15)] =
              f2bf(oacc[ns][r] * linv[r]);
      }
    }
  }
}

}  // namespace

extern "C" {

void tl_prefill_attn(const void* q, const void* k, const void* v, void* out,
                     int B, int S, int Hq, int Hkv, int D, float scale,
                     int causal, hipStream_t stream) {
  dim3 grid((S + BM - 1) / BM, Hq, B), block(BLOCK);
  if (D == 128)
    hipLaunchKernelGGL((prefill_attn_kernel<128>), grid, block, 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (bf16*)out, B, S, Hq, Hkv, scale, causal);
  else if (D == 64)
    hipLaunchKernelGGL((prefill_attn_kernel<64>), grid, block, 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (bf16*)out, B, S, Hq, Hkv, scale, causal);
}

}  // extern "C"
