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

// counted VMEM wait with a compile-time immediate — invisible to hipcc's
// waitcnt pass, so the glds pipeline is not drained at barriers (guide
// §5: plain __syncthreads with a glds in flight emits vmcnt(0))
template <int N>
DEVINLINE void waitcnt_vm() {
  asm volatile("s_waitcnt vmcnt(%0)" ::"n"(N) : "memory");
}

template <int MT, int HAS_BIAS, int SPLIT>
__global__ __launch_bounds__(BLOCK, 2) void gemm_tiled_kernel(
    const bf16* __restrict__ x,     // [M, K]
    const bf16* __restrict__ w,     // [N, K]
    const bf16* __restrict__ bias,  // [N] or null
    bf16* __restrict__ out,         // [M, N]
    float* __restrict__ partial,    // [SPLITK, M, N] when SPLIT
    int M, int N, int K, int n_split) {
  constexpr int MPW = MT / 4;                 // m-tiles per wave
  static_assert(MT % 4 == 0, "quadrant layout needs MT % 4 == 0");
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int col = lane & 15;
  const int quad = lane >> 4;
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

  // Counted-vmcnt double-buffer pipeline: tile t+1's glds stay in flight
  // across the barrier (each wave issues a fixed LPT loads per tile —
  // MT/2 x-chunks + 2 W-chunks, wave-uniform for MT in {8,16} — waits
  // for its OWN tile-t loads with vmcnt(LPT), then the barrier aligns
  // the workgroup, so tile t is fully in LDS without draining t+1).
  constexpr int LPT = MT / 2 + 2;
  static_assert(MT == 8 || MT == 16, "LPT assumes wave-uniform chunks");
  GT_GLDS_TILE(0, k_begin);
  int buf = 0;
  for (int k0 = k_begin; k0 < k_end; k0 += 64) {
    if (k0 + 64 < k_end) {
      GT_GLDS_TILE(buf ^ 1, k0 + 64);         // in flight under compute
      waitcnt_vm<LPT>();                      // tile t landed; t+1 out
    } else {
      waitcnt_vm<0>();                        // last tile: drain
    }
    __builtin_amdgcn_s_barrier();
    const int tile_k = min(64, k_end - k0);
    // 2D fragment reuse: this wave owns a (MT/4 x 4)-tile quadrant —
    // rows [wave*MPW*16, +MPW*16), all 64 panel columns — so each kk
    // reads MPW a-frags + 4 b-frags and issues 4*MPW mfmas (the 1D
    // col-strip layout read one a-frag PER mfma and was LDS-issue
    // bound at ~1.9 TB/s on wide-N shapes).
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      if (kk * 32 < tile_k) {
        bf16x8_t bfrag[4], afrag[MPW];
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          const int wrow = nt * 16 + col;
          const int wslot = (kk * 4 + quad) ^ (wrow & 7);
          bfrag[nt] = *reinterpret_cast<const bf16x8_t*>(
              &lds[buf][MT * 16 + wrow][wslot * 8]);
        }
#pragma unroll
        for (int mt = 0; mt < MPW; ++mt) {
          const int row = wave * (MPW * 16) + mt * 16 + col;
          const int slot = (kk * 4 + quad) ^ (row & 7);
          afrag[mt] = *reinterpret_cast<const bf16x8_t*>(
              &lds[buf][row][slot * 8]);
        }
#pragma unroll
        for (int mt = 0; mt < MPW; ++mt)
#pragma unroll
          for (int nt = 0; nt < 4; ++nt)
            acc[mt * 4 + nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[mt], bfrag[nt], acc[mt * 4 + nt], 0, 0, 0);
      }
    }
    // all waves done reading buf before the next iteration's glds
    // overwrite it (hipcc has already counted-waited the ds_reads into
    // the mfmas; lgkmcnt(0) makes that explicit before the raw barrier)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }
#undef GT_GLDS_TILE

#pragma unroll
  for (int mt = 0; mt < MPW; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int oc = blockIdx.x * 64 + nt * 16 + col;
      const float b = HAS_BIAS ? bf2f(bias[oc]) : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int orow =
            m_base + wave * (MPW * 16) + mt * 16 + quad * 4 + r;
        if (orow < M) {
          if (SPLIT)
            partial[((int64_t)split * M + orow) * N + oc] =
                acc[mt * 4 + nt][r];
          else
            out[(int64_t)orow * N + oc] = f2bf(acc[mt * 4 + nt][r] + b);
        }
      }
    }
  }
}

// Big-tile variant: 256x256 block tile, 8 waves (2M x 4N quadrants of
// 128x64), 128 KB LDS, 1 block/CU. For wide-N (gate_up, lm_head) and
// deep-K (down_proj) shapes the dominant cost in the BN=64 kernel is
// re-reading the x slab once per output panel (N/BN * M*K*2 bytes, L3-
// served at ~10 TB/s); BN=256 makes x:W traffic 1:1. Per-iteration MFMA
// work (256 mfma/block) is also long enough to hide the glds latency
// that starves the small-tile kernels. Same per-row K accumulation
// semantics as the rest of the family. Handles N % 256 != 0 (lm_head
// 151936) with clamped W loads + guarded stores.
template <int HAS_BIAS, int SPLIT>
__global__ __launch_bounds__(512, 1) void gemm_tiled_sq_kernel(
    const bf16* __restrict__ x,     // [M, K]
    const bf16* __restrict__ w,     // [N, K]
    const bf16* __restrict__ bias,  // [N] or null
    bf16* __restrict__ out,         // [M, N]
    float* __restrict__ partial,    // [SPLITK, M, N] when SPLIT
    int M, int N, int K, int n_split) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;          // 0..7
  const int col = lane & 15;
  const int quad = lane >> 4;
  const int wm = wave >> 2;                   // 0..1: 128-row band
  const int wn = wave & 3;                    // 0..3: 64-col band
  const int m_base = blockIdx.z * 256;

  const int split = SPLIT ? blockIdx.y : 0;
  const int k_per = SPLIT ? ((K / 64 + n_split - 1) / n_split) * 64 : K;
  const int k_begin = split * k_per;
  const int k_end = min(K, k_begin + k_per);

  __shared__ bf16 lds[2][256 + 256][64];

  f32x4 acc[32];
#pragma unroll
  for (int m = 0; m < 32; ++m) acc[m] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int l8 = lane >> 3, c8 = lane & 7;

  // x: 32 chunks of 8 rows -> 4 per wave; W: 32 chunks -> 4 per wave
#define GS_GLDS_TILE(bufi, k0)                                              \
  if ((k0) < k_end) {                                                       \
    _Pragma("unroll") for (int g = 0; g < 4; ++g) {                         \
      const int chunk = wave * 4 + g;                                       \
      const int row = chunk * 8 + l8;                                       \
      const int xr = min(m_base + row, M - 1);                              \
      const int kc = min((k0) + (c8 ^ (row & 7)) * 8, K - 8);               \
      auto gsrc = (const __attribute__((address_space(1))) void*)(          \
          x + (int64_t)xr * K + kc);                                        \
      auto ldst = (__attribute__((address_space(3))) void*)(                \
          &lds[bufi][chunk * 8][0]);                                        \
      __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);               \
    }                                                                       \
    _Pragma("unroll") for (int g = 0; g < 4; ++g) {                         \
      const int wchunk = wave * 4 + g;                                      \
      const int wr = wchunk * 8 + l8;                                       \
      const int gr = min((int)blockIdx.x * 256 + wr, N - 1);                \
      const int kc = min((k0) + (c8 ^ (wr & 7)) * 8, K - 8);                \
      auto gsrc = (const __attribute__((address_space(1))) void*)(          \
          w + (int64_t)gr * K + kc);                                        \
      auto ldst = (__attribute__((address_space(3))) void*)(                \
          &lds[bufi][256 + wchunk * 8][0]);                                 \
      __builtin_amdgcn_global_load_lds(gsrc, ldst, 16, 0, 0);               \
    }                                                                       \
  }

  GS_GLDS_TILE(0, k_begin);
  int buf = 0;
  for (int k0 = k_begin; k0 < k_end; k0 += 64) {
    if (k0 + 64 < k_end) {
      GS_GLDS_TILE(buf ^ 1, k0 + 64);
      waitcnt_vm<8>();                        // tile t landed; t+1 out
    } else {
      waitcnt_vm<0>();
    }
    __builtin_amdgcn_s_barrier();
    const int tile_k = min(64, k_end - k0);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      if (kk * 32 < tile_k) {
        bf16x8_t bfrag[4], afrag[8];
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          const int wrow = wn * 64 + nt * 16 + col;
          const int wslot = (kk * 4 + quad) ^ (wrow & 7);
          bfrag[nt] = *reinterpret_cast<const bf16x8_t*>(
              &lds[buf][256 + wrow][wslot * 8]);
        }
#pragma unroll
        for (int mt = 0; mt < 8; ++mt) {
          const int row = wm * 128 + mt * 16 + col;
          const int slot = (kk * 4 + quad) ^ (row & 7);
          afrag[mt] = *reinterpret_cast<const bf16x8_t*>(
              &lds[buf][row][slot * 8]);
        }
#pragma unroll
        for (int mt = 0; mt < 8; ++mt)
#pragma unroll
          for (int nt = 0; nt < 4; ++nt)
            acc[mt * 4 + nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[mt], bfrag[nt], acc[mt * 4 + nt], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    buf ^= 1;
  }
#undef GS_GLDS_TILE

#pragma unroll
  for (int mt = 0; mt < 8; ++mt) {
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int oc = blockIdx.x * 256 + wn * 64 + nt * 16 + col;
      if (oc < N) {
        const float b = HAS_BIAS ? bf2f(bias[oc]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int orow = m_base + wm * 128 + mt * 16 + quad * 4 + r;
          if (orow < M) {
            if (SPLIT)
              partial[((int64_t)split * M + orow) * N + oc] =
                  acc[mt * 4 + nt][r];
            else
              out[(int64_t)orow * N + oc] = f2bf(acc[mt * 4 + nt][r] + b);
          }
        }
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
  // very-wide-N shapes (lm_head) take the 256x256 8-wave kernel; at
  // gate_up/down widths the BN=64 kernel measures faster (see
  // profiles/gemm_family_ab.md). Dispatch is by (N, K) only -> the
  // choice is M-independent
  if (N >= 65536) {
    dim3 wgrid(cdiv(N, 256), n_split, cdiv(M, 256)), wblock(512);
#define WDISPATCH(HB, SP)                                                   \
  hipLaunchKernelGGL((gemm_tiled_sq_kernel<HB, SP>), wgrid, wblock, 0,      \
                     stream, (const bf16*)x, (const bf16*)w,                \
                     (const bf16*)bias, (bf16*)out, (float*)partial, M, N,  \
                     K, n_split)
    if (split) WDISPATCH(0, 1);
    else if (bias) WDISPATCH(1, 0);
    else WDISPATCH(0, 0);
#undef WDISPATCH
  } else {
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
  }
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
