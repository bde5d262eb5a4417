// MFMA decode attention (v2) for gfx950.
//
// The v1 kernel (decode_attn.hip) is VALU-bound: ~85 VALU ops per 64 B of
// KV streamed caps it at ~2.6 TB/s effective (profiles/). Here the score
// and PV math run on the matrix cores instead:
//
//   grid (B, Hkv, SPLIT): ONE block covers ALL G <= 16 query heads of one
//   kv head (single pass over the K/V slab — v1 needed ceil(G/4) passes),
//   with the MFMA M-dimension = query heads.
//   Per 16-key tile: QK^T = mfma_f32_16x16x32_bf16 over D/32 k-chunks,
//   where the B-fragment IS the natural K layout (lane l reads 16 B of key
//   l&15 at d-offset (l>>4)*8 — no LDS staging for K). V is staged
//   transposed through LDS (pad stride 40 elems: 16-B aligned, 20-dword
//   row stride is conflict-free over 16 rows) so the PV B-fragment is a
//   contiguous ds_read_b128. P round-trips through a small LDS tile to
//   convert C-layout -> A-layout.
//
// Flash-decode SPLIT: for small B*Hkv the seq dim is partitioned across
// SPLIT blocks writing (m, l, o) partials; a tiny combine kernel merges.

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;
constexpr int NWAVE = 4;
constexpr int KT = 16;            // keys per MFMA tile

using f32x4 = __attribute__((ext_vector_type(4))) float;
typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

// out layout when SPLIT > 1: partial [B, Hq, SPLIT, D] float, ml [B, Hq,
// SPLIT, 2] float (m, l). SPLIT == 1 writes bf16 out directly.
// PAGED: k_cache/v_cache are page pools [n_pages, Hkv, PAGE, D] and
// block_table [B, bt_stride] maps position/PAGE -> page id (PAGE = 128).
template <int D, int SPLIT_MODE, int PAGED>
__global__ __launch_bounds__(BLOCK) void decode_attn_mfma_kernel(
    const bf16* __restrict__ q,        // [B, Hq, D]
    const bf16* __restrict__ k_cache,  // [B, Hkv, Smax, D] (or pool)
    const bf16* __restrict__ v_cache,
    const int* __restrict__ seq_lens,  // [B]
    const int* __restrict__ block_table,
    bf16* __restrict__ out,            // [B, Hq, D]
    float* __restrict__ partial,       // [B, Hq, SPLIT, D]
    float* __restrict__ partial_ml,    // [B, Hq, SPLIT, 2]
    int Hq, int Hkv, int Smax, float scale, int n_split, int bt_stride,
    int per_row) {
  constexpr int KCH = D / 32;        // MFMA k-chunks for QK^T
  constexpr int NS = D / 16;         // PV output col tiles
  const int b = blockIdx.x;
  const int hkv = blockIdx.y;
  const int split = blockIdx.z;
  const int G = Hq / Hkv;
  const int h0 = hkv * G;
  const int L = seq_lens[b];

  // per_row: this ROW's split count comes from its own length (batch
  // composition must not change a row's accumulation — exact-greedy
  // guarantee); the launch z-width n_split is a host-side upper bound
  // and surplus blocks exit.
  const int ns_row = per_row ? tl_split_for_len(L) : n_split;
  if (split >= ns_row) return;

  // this split's key range
  const int per_split = (L + ns_row - 1) / ns_row;
  const int k_begin = split * per_split;
  const int k_end = min(L, k_begin + per_split);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;         // key-in-tile / head-col
  const int quad = lane >> 4;        // 0..3

  // V as row-major [32 keys][16 dv] panels per (wave, dv-16-chunk): b128
  // staging writes + ds_read_b64_tr_b16 fragment reads (no scatter
  // transpose). Rows 16..31 stay zero — they feed the K=32 PV MFMA's
  // upper half, so the quad<2 zero-branch disappears. Same for the upper
  // half of the P tile.
  constexpr int PSTRIDE = 32 * 16 + 8;
  __shared__ bf16 v_pan[NWAVE * (D / 16) * PSTRIDE];
  __shared__ bf16 p_lds[NWAVE][16][32 + 8];
  __shared__ float red_m[NWAVE][16], red_l[NWAVE][16];
  __shared__ float red_o[NWAVE][16][D];

  // one-time zero of the never-written upper halves
  for (int i = threadIdx.x; i < NWAVE * (D / 16); i += BLOCK) {
    for (int e = 0; e < 16 * 16; ++e)
      v_pan[i * PSTRIDE + 16 * 16 + e] = f2bf(0.f);
  }
  for (int i = threadIdx.x; i < NWAVE * 16; i += BLOCK) {
    for (int e = 16; e < 32; ++e) p_lds[i / 16][i % 16][e] = f2bf(0.f);
  }
  __syncthreads();

  const bf16* kbase = k_cache + ((int64_t)b * Hkv + hkv) * Smax * D;
  const bf16* vbase = v_cache + ((int64_t)b * Hkv + hkv) * Smax * D;

  // ---- A fragments: q for this head group; rows >= G zeroed ----
  bf16x8_t afrag[KCH];
  {
    const bool valid = col < G;
    const bf16* qrow = q + ((int64_t)b * Hq + h0 + (valid ? col : 0)) * D;
#pragma unroll
    for (int c = 0; c < KCH; ++c) {
      bf16x8_t f = *reinterpret_cast<const bf16x8_t*>(
          qrow + c * 32 + quad * 8);
      if (!valid)
#pragma unroll
        for (int j = 0; j < 8; ++j) f[j] = (__bf16)0.f;
      afrag[c] = f;
    }
  }

  float m_run[4], l_run[4];
  f32x4 oacc[NS];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
#pragma unroll
  for (int n = 0; n < NS; ++n) oacc[n] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // ---- main loop: each wave takes every NWAVE'th 16-key tile ----
  for (int t0 = k_begin + wave * KT; t0 < k_end; t0 += NWAVE * KT) {
    const int key = t0 + col;
    const bool kv_ok = key < k_end;
    const int safe_key = kv_ok ? key : (k_end - 1);

    // B fragments straight from global K + stage V transposed into LDS
    f32x4 sacc = (f32x4){0.f, 0.f, 0.f, 0.f};
    {
      const bf16 *krow, *vrow;
      if (PAGED) {
        const int page = block_table[b * bt_stride + (safe_key >> 7)];
        const int64_t off =
            (((int64_t)page * Hkv + hkv) * 128 + (safe_key & 127)) * D;
        krow = k_cache + off;
        vrow = v_cache + off;
      } else {
        krow = kbase + (int64_t)safe_key * D;
        vrow = vbase + (int64_t)safe_key * D;
      }
#pragma unroll
      for (int c = 0; c < KCH; ++c) {
        bf16x8_t kf = *reinterpret_cast<const bf16x8_t*>(
            krow + c * 32 + quad * 8);
        sacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[c], kf, sacc,
                                                       0, 0, 0);
        // V: same addressing; b128 write into the (wave, dv-chunk) panel
        bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
            vrow + c * 32 + quad * 8);
        const int d0 = c * 32 + quad * 8;
        const int pan = wave * (D / 16) + (d0 >> 4);
        *reinterpret_cast<bf16x8_t*>(
            &v_pan[pan * PSTRIDE + col * 16 + (d0 & 15)]) = vf;
      }
    }

    // ---- online softmax on the 16-key score tile ----
    // lane holds S[row=quad*4+r][col] in sacc[r]; per-ROW max (a shared
    // 4-row max underflows exp when head score ranges differ by >88)
    float p[4], row_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      sacc[r] = kv_ok ? sacc[r] * scale : -1e30f;
      row_max[r] = sacc[r];
    }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        row_max[r] = fmaxf(row_max[r], __shfl_xor(row_max[r], off,
                                                  WAVE_SIZE));
    float m_new[4], alpha[4], psum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_new[r] = fmaxf(m_run[r], row_max[r]);
      alpha[r] = __expf(m_run[r] - m_new[r]);
      p[r] = (kv_ok) ? __expf(sacc[r] - m_new[r]) : 0.f;
      psum[r] = p[r];
      m_run[r] = m_new[r];
    }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        psum[r] += __shfl_xor(psum[r], off, WAVE_SIZE);
#pragma unroll
    for (int r = 0; r < 4; ++r) l_run[r] = l_run[r] * alpha[r] + psum[r];

    // P -> LDS (C-layout write), then A-fragment read. NO barriers here:
    // each wave owns its own LDS slabs ([wave] index) and same-wave
    // ds_write -> ds_read ordering is enforced by lgkmcnt; a block-wide
    // barrier would deadlock (waves run different trip counts).
#pragma unroll
    for (int r = 0; r < 4; ++r)
      p_lds[wave][quad * 4 + r][col] = f2bf(p[r]);
    // rescale O accumulators by alpha (row = quad*4+r)
#pragma unroll
    for (int n = 0; n < NS; ++n)
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[n][r] *= alpha[r];

    // ---- PV: K=32 MFMA; keys 16..31 are the pre-zeroed halves ----
    bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
        &p_lds[wave][col][quad * 8]);
#pragma unroll
    for (int n = 0; n < NS; ++n) {
      const unsigned pan_base = (unsigned)(uintptr_t)(
          &v_pan[(wave * (D / 16) + n) * PSTRIDE]);
      bf16x8_t vb = (bf16x8_t)ds_read_tr16_frag(
          tr16_frag_addr(pan_base, lane));
      oacc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, oacc[n],
                                                        0, 0, 0);
    }
  }

  // ---- wave partials -> LDS ----
  // lane holds O[row=quad*4+r][col=n*16+col]; every 16-lane group holds the
  // full row set, so groups 1..3 duplicate group 0 -> write from quad 0..3
  // cover distinct (row, col) pairs already: row=quad*4+r col covers only
  // cols where lane&15 == col. Write all lanes (distinct rows per quad).
#pragma unroll
  for (int n = 0; n < NS; ++n)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      red_o[wave][quad * 4 + r][n * 16 + col] = oacc[n][r];
  if (col == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      red_m[wave][quad * 4 + r] = m_run[r];
      red_l[wave][quad * 4 + r] = l_run[r];
    }
  }
  __syncthreads();

  // ---- final merge across waves; write out ----
  const int g_count = min(G, 16);
  for (int i = threadIdx.x; i < g_count * D; i += BLOCK) {
    const int g = i / D, d = i - g * D;
    float m_tot = -1e30f;
#pragma unroll
    for (int w = 0; w < NWAVE; ++w) m_tot = fmaxf(m_tot, red_m[w][g]);
    float l_tot = 0.f, o = 0.f;
#pragma unroll
    for (int w = 0; w < NWAVE; ++w) {
      const float f = __expf(red_m[w][g] - m_tot);
      l_tot += red_l[w][g] * f;
      o += red_o[w][g][d] * f;
    }
    if (SPLIT_MODE) {
      const int64_t base = (((int64_t)b * Hq + h0 + g) * n_split + split);
      partial[base * D + d] = o;
      if (d == 0) {
        partial_ml[base * 2] = m_tot;
        partial_ml[base * 2 + 1] = l_tot;
      }
    } else {
      out[((int64_t)b * Hq + h0 + g) * D + d] =
          f2bf(o / fmaxf(l_tot, 1e-30f));
    }
  }
}

// combine kernel: merge this row's ns_row partials per (b, h). A row
// whose ns_row == 1 reduces to f = exp(0) = 1 scales — bitwise equal to
// the direct-write path.
template <int D>
__global__ __launch_bounds__(128) void decode_attn_combine_kernel(
    const float* __restrict__ partial, const float* __restrict__ partial_ml,
    const int* __restrict__ seq_lens, bf16* __restrict__ out, int Hq,
    int n_split, int per_row) {
  const int bh = blockIdx.x;                    // b * Hq + h
  const int d = threadIdx.x;                    // [0, D)
  const int ns_row =
      per_row ? tl_split_for_len(seq_lens[bh / Hq]) : n_split;
  __shared__ float sm[64], sl[64];
  if (threadIdx.x < ns_row) {
    sm[threadIdx.x] = partial_ml[((int64_t)bh * n_split + threadIdx.x) * 2];
    sl[threadIdx.x] = partial_ml[((int64_t)bh * n_split + threadIdx.x) * 2 + 1];
  }
  __syncthreads();
  float m_tot = -1e30f;
  for (int s = 0; s < ns_row; ++s) m_tot = fmaxf(m_tot, sm[s]);
  float l_tot = 0.f, o = 0.f;
  for (int s = 0; s < ns_row; ++s) {
    const float f = __expf(sm[s] - m_tot);
    l_tot += sl[s] * f;
    o += partial[((int64_t)bh * n_split + s) * D + d] * f;
  }
  out[(int64_t)bh * D + d] = f2bf(o / fmaxf(l_tot, 1e-30f));
}

}  // namespace

extern "C" {

// scratch: partial floats [B*Hq*n_split*D] + ml [B*Hq*n_split*2]; pass
// nullptr when n_split == 1.
void tl_decode_attn_mfma(const void* q, const void* k_cache,
                         const void* v_cache, const void* seq_lens,
                         const void* block_table, void* out,
                         void* partial, void* partial_ml, int B, int Hq,
                         int Hkv, int Smax, int D, float scale, int n_split,
                         int bt_stride, int per_row, hipStream_t stream) {
  dim3 grid(B, Hkv, n_split), block(BLOCK);
#define LAUNCH(DD, SM, PG)                                                   \
  hipLaunchKernelGGL((decode_attn_mfma_kernel<DD, SM, PG>), grid, block, 0,  \
                     stream, (const bf16*)q, (const bf16*)k_cache,           \
                     (const bf16*)v_cache, (const int*)seq_lens,             \
                     (const int*)block_table, (bf16*)out,                    \
                     (float*)partial, (float*)partial_ml, Hq, Hkv, Smax,     \
                     scale, n_split, bt_stride, per_row)
#define PICK(DD)                                                             \
  do {                                                                       \
    if (block_table) {                                                       \
      if (n_split == 1) LAUNCH(DD, 0, 1); else LAUNCH(DD, 1, 1);             \
    } else {                                                                 \
      if (n_split == 1) LAUNCH(DD, 0, 0); else LAUNCH(DD, 1, 0);             \
    }                                                                        \
  } while (0)
  if (D == 128) PICK(128);
  else if (D == 64) PICK(64);
#undef PICK
#undef LAUNCH
  if (n_split > 1) {
    dim3 cgrid(B * Hq), cblock(D);
    if (D == 128)
      hipLaunchKernelGGL((decode_attn_combine_kernel<128>), cgrid, cblock, 0,
                         stream, (const float*)partial,
                         (const float*)partial_ml, (const int*)seq_lens,
                         (bf16*)out, Hq, n_split, per_row);
    else
      hipLaunchKernelGGL((decode_attn_combine_kernel<64>), cgrid, cblock, 0,
                         stream, (const float*)partial,
                         (const float*)partial_ml, (const int*)seq_lens,
                         (bf16*)out, Hq, n_split, per_row);
  }
}

}  // extern "C"
