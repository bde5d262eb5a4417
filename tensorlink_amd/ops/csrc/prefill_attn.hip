// Causal GQA flash-attention forward (prefill) on gfx950 MFMA.
//
// Replaces the reference's HF SDPA-inside-eager-blocks prefill
// (tensorlink/ml/worker.py:330-335).
//
// v2 structure: v1 round-tripped the fp32 score tile through LDS for a
// separate softmax phase — 16-way bank conflicts on the [64][64] fp32 tile
// and a serial 4-threads-per-row pass held it at 59 TF. v2 keeps scores in
// the MFMA C-fragments and does the online softmax with 16-lane shuffles
// (rows live on fixed (quad, reg) lanes across all K-tiles, so m/l/O state
// stays in registers); only the bf16 P tile touches LDS (for the PV
// A-fragment layout), padded conflict-free.
//
//   block = 256 threads (4 waves); BM = 64 query rows (wave w owns rows
//   w*16); K-tiles of BN = 64 staged once per block (K direct-read
//   fragments, V transposed into LDS for contiguous PV B-fragments).
//
// q [B,S,Hq,D], k,v [B,S,Hkv,D] bf16 row-major; out [B,S,Hq,D].

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;   // 4 waves
constexpr int BM = 64;       // query rows per block
constexpr int BN = 64;       // key cols per tile
constexpr int PPAD = 8;      // P row pad: stride 72 elems = 144 B

using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

// Sq = query length, Skv = key/value length, q_off = global position of
// query row 0 (chunked prefill: Skv = q_off + Sq history+chunk keys; the
// plain full prefill is Sq == Skv, q_off == 0).
template <int D>
__global__ __launch_bounds__(BLOCK) void prefill_attn_kernel(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, bf16* __restrict__ out,
    float* __restrict__ lse, int B, int Sq, int Skv, int q_off, int Hq,
    int Hkv, float scale, int causal) {
  static_assert(D == 64 || D == 128);
  constexpr int KCH = D / 32;      // QK^T k-chunks
  constexpr int NSUB = BN / 16;    // score col tiles = 4
  constexpr int NS_PV = D / 16;    // PV output col tiles

  constexpr int NPAN = (BN / 32) * (D / 16);  // V panels [32 keys][16 dv]
  constexpr int PSTRIDE = 32 * 16 + 8;        // panel stride (elems), padded
  // k_lds rows padded +8 elems: stride 272 B (17 slots) -> fragment reads
  // of 16 different rows land on distinct banks (an unpadded 256-B stride
  // puts EVERY row on bank 0: 16-way conflict).
  __shared__ bf16 k_lds[BN][D + 8];
  // V stored as row-major [32][16] panels, written with single b128 stores
  // and consumed via ds_read_b64_tr_b16 (hardware transpose) — replaces the
  // v1/v2 element-wise LDS scatter transpose that made the kernel
  // LDS-bound (PMC: 43% WAIT_INST_LDS, 26% bank-conflict cycles).
  __shared__ bf16 v_pan[NPAN * PSTRIDE];
  __shared__ bf16 p_lds[4][16][BN + PPAD];   // per wave: 16 rows x 64 keys

  const int qt = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);
  const int q0 = qt * BM;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int quad = lane >> 4;
  const int r0 = wave * 16;                  // wave's first row in tile

  // ---- Q fragments in registers: rows r0+col, all KCH chunks ----
  bf16x8_t qfrag[KCH];
  {
    const int qrow = q0 + r0 + col;
    const int safe = min(qrow, Sq - 1);
    const bf16* qp = q + (((int64_t)b * Sq + safe) * Hq + h) * D;
#pragma unroll
    for (int c = 0; c < KCH; ++c)
      qfrag[c] = *reinterpret_cast<const bf16x8_t*>(qp + c * 32 + quad * 8);
  }

  float m_run[4], l_run[4];
  f32x4 oacc[NS_PV];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
#pragma unroll
  for (int n = 0; n < NS_PV; ++n) oacc[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(Skv, q_off + q0 + BM) : Skv;
  const int64_t kbase = ((int64_t)b * Skv) * Hkv * D + (int64_t)hkv * D;

  for (int kt0 = 0; kt0 < kv_end; kt0 += BN) {
    // ---- stage K tile [BN][D]; V into [32][16] tr panels ----
    for (int i = threadIdx.x * 8; i < BN * D; i += BLOCK * 8) {
      const int row = i / D, c = i % D;
      const int key = kt0 + row;
      const int64_t off = kbase + (int64_t)min(key, Skv - 1) * Hkv * D + c;
      bf16x8_t kval = *reinterpret_cast<const bf16x8_t*>(k + off);
      bf16x8_t vval = *reinterpret_cast<const bf16x8_t*>(v + off);
      if (key >= Skv) {
#pragma unroll
        for (int j = 0; j < 8; ++j) { kval[j] = (__bf16)0.f; vval[j] = (__bf16)0.f; }
      }
      *reinterpret_cast<bf16x8_t*>(&k_lds[row][c]) = kval;
      // panel (ks = row/32, n = c/16), row kk = row%32, col c0 = c%16
      const int pan = (row >> 5) * (D / 16) + (c >> 4);
      *reinterpret_cast<bf16x8_t*>(
          &v_pan[pan * PSTRIDE + (row & 31) * 16 + (c & 15)]) = vval;
    }
    __syncthreads();

    // ---- QK^T for all 4 col-subtiles (scores stay in registers) ----
    f32x4 sacc[NSUB];
#pragma unroll
    for (int ns = 0; ns < NSUB; ++ns) {
      sacc[ns] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int c = 0; c < KCH; ++c) {
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
            &k_lds[ns * 16 + col][c * 32 + quad * 8]);
        sacc[ns] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[c], bfrag,
                                                           sacc[ns], 0, 0, 0);
      }
    }
    // ---- mask + scale + ONE tile-wide row max (P for every subtile is
    // written against the same m, so no stale-P rescale hazard) ----
    float row_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) row_max[r] = -1e30f;
#pragma unroll
    for (int ns = 0; ns < NSUB; ++ns) {
      const int kcol = kt0 + ns * 16 + col;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q_off + q0 + r0 + quad * 4 + r;
        const bool masked = (kcol >= Skv) || (causal && kcol > qrow);
        sacc[ns][r] = masked ? -1e30f : sacc[ns][r] * scale;
        row_max[r] = fmaxf(row_max[r], sacc[ns][r]);
      }
    }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        row_max[r] = fmaxf(row_max[r],
                           __shfl_xor(row_max[r], off, WAVE_SIZE));
    float alpha[4], psum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], row_max[r]);
      alpha[r] = __expf(m_run[r] - m_new);
      m_run[r] = m_new;
      psum[r] = 0.f;
    }
#pragma unroll
    for (int ns = 0; ns < NSUB; ++ns)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = (sacc[ns][r] <= -1e29f)
                            ? 0.f : __expf(sacc[ns][r] - m_run[r]);
        p_lds[wave][quad * 4 + r][ns * 16 + col] = f2bf(p);
        psum[r] += p;
      }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        psum[r] += __shfl_xor(psum[r], off, WAVE_SIZE);
#pragma unroll
    for (int r = 0; r < 4; ++r)
      l_run[r] = l_run[r] * alpha[r] + psum[r];

    // ---- O = O*alpha + P V ----
#pragma unroll
    for (int n = 0; n < NS_PV; ++n)
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[n][r] *= alpha[r];
#pragma unroll
    for (int ks = 0; ks < BN / 32; ++ks) {
      bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
          &p_lds[wave][col][ks * 32 + quad * 8]);
#pragma unroll
      for (int n = 0; n < NS_PV; ++n) {
        const unsigned pan_base = (unsigned)(uintptr_t)(
            &v_pan[(ks * (D / 16) + n) * PSTRIDE]);
        bf16x8_t vb = (bf16x8_t)ds_read_tr16_frag(
            tr16_frag_addr(pan_base, lane));
        oacc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, oacc[n],
                                                          0, 0, 0);
      }
    }
    __syncthreads();  // before next tile overwrites K/VT
  }

  // ---- epilogue: divide by l, store (+ logsumexp for backward) ----
  {
    const int64_t obase = ((int64_t)b * Sq) * Hq * D + (int64_t)h * D;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = r0 + quad * 4 + r;
      const float linv = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
      if (q0 + row < Sq) {
#pragma unroll
        for (int n = 0; n < NS_PV; ++n)
          out[obase + (int64_t)(q0 + row) * Hq * D + n * 16 + col] =
              f2bf(oacc[n][r] * linv);
        if (lse && col == 0)
          lse[((int64_t)b * Hq + h) * Sq + q0 + row] =
              l_run[r] > 0.f ? m_run[r] + __logf(l_run[r]) : -1e30f;
      }
    }
  }
}

}  // namespace

extern "C" {

void tl_prefill_attn_lse(const void* q, const void* k, const void* v,
                         void* out, void* lse, int B, int Sq, int Skv,
                         int q_off, int Hq, int Hkv, int D, float scale,
                         int causal, hipStream_t stream);

void tl_prefill_attn(const void* q, const void* k, const void* v, void* out,
                     int B, int Sq, int Skv, int q_off, int Hq, int Hkv,
                     int D, float scale, int causal, hipStream_t stream) {
  tl_prefill_attn_lse(q, k, v, out, nullptr, B, Sq, Skv, q_off, Hq, Hkv, D,
                      scale, causal, stream);
}

void tl_prefill_attn_lse(const void* q, const void* k, const void* v,
                         void* out, void* lse, int B, int Sq, int Skv,
                         int q_off, int Hq, int Hkv, int D, float scale,
                         int causal, hipStream_t stream) {
  dim3 grid((Sq + BM - 1) / BM, Hq, B), block(BLOCK);
  if (D == 128)
    hipLaunchKernelGGL((prefill_attn_kernel<128>), grid, block, 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (bf16*)out, (float*)lse, B, Sq, Skv, q_off, Hq, Hkv,
                       scale, causal);
  else if (D == 64)
    hipLaunchKernelGGL((prefill_attn_kernel<64>), grid, block, 0, stream,
                       (const bf16*)q, (const bf16*)k, (const bf16*)v,
                       (bf16*)out, (float*)lse, B, Sq, Skv, q_off, Hq, Hkv,
                       scale, causal);
}

}  // extern "C"
