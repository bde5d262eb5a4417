// Fused on-device token sampling for serving decode (gfx950).
//
// The reference delegates sampling to HF generate()'s host-side logits
// processors (SURVEY.md §2.4 row "LM head + sampling"); here the whole
// chain — presence/frequency penalties, temperature, top-k, top-p,
// draw — runs in ONE kernel, one block per row, so sampled decode can
// be captured in a hipGraph (counter-based RNG, no host RNG state).
//
// Algorithm per row over V logits (bf16):
//   greedy (T<=0): one penalized argmax pass (first-index tie-break,
//     matching torch.argmax).
//   sampled: pass 1 finds the penalized max M; pass 2 builds a 256-bin
//     histogram over the top 8 bits of the ORDERED float key of z
//     (count + sum of exp((z-M)/T) per bin, LDS atomics); the top-k /
//     top-p cutoff bin is refined with a second 8-bit level — exact for
//     bf16-sourced values (bf16 = top 16 f32 bits) — giving a value
//     threshold t and the kept probability mass S; pass 4 walks the row
//     in index order accumulating kept weights until u*S is crossed.
//   Ties at the threshold are all kept (the torch reference cuts ties
//   in sort order at exactly-k / crossing-p; including value-ties only
//   ever adds tokens of identical probability).
//
// RNG: splitmix64 of (seed[row] or seed_base, step counter, row) — a
// row's draw depends only on its own seed/step, never on batch
// composition, and a device counter bumped by a tiny kernel inside the
// captured graph advances replays.

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

DEVINLINE uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// monotone uint32 key for float ordering
DEVINLINE uint32_t fkey(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

struct BlockRed {
  float v[BLOCK / 64];
  int i[BLOCK / 64];
  float f[BLOCK / 64];
};

// LDS layout shared by all phases (ONE __shared__ object)
struct SampleLds {
  union {
    struct {
      int cnt[256];
      // exp-weights quantized to integers (x 2^20) and accumulated
      // with EXACT integer atomics: float atomicAdd order is
      // nondeterministic and flipped top-p thresholds between
      // identical calls (caught by the graph-reproducibility test)
      unsigned long long sum[256];
    } hist;
    BlockRed red;
    float scan[BLOCK / 64];
  } u;
  float stat[8];        // float broadcast slots
  unsigned long long statu[4];   // integer broadcast slots
};

DEVINLINE unsigned long long wquant(float e) {
  // e in [0, 1] -> fixed point x 2^20 (sums over 152k vocab stay < 2^38)
  return (unsigned long long)(e * 1048576.0f);
}

DEVINLINE float block_max_argmin(SampleLds* lds, float v, int idx,
                                 int* out_idx) {
  // max value; among equal values the smallest index (torch.argmax)
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(v, off, 64);
    int oi = __shfl_xor(idx, off, 64);
    if (ov > v || (ov == v && oi < idx)) { v = ov; idx = oi; }
  }
  if (lane == 0) { lds->u.red.v[wave] = v; lds->u.red.i[wave] = idx; }
  __syncthreads();
  float bv = lds->u.red.v[0];
  int bi = lds->u.red.i[0];
#pragma unroll
  for (int w = 1; w < BLOCK / 64; ++w) {
    float ov = lds->u.red.v[w];
    int oi = lds->u.red.i[w];
    if (ov > bv || (ov == bv && oi < bi)) { bv = ov; bi = oi; }
  }
  __syncthreads();
  *out_idx = bi;
  return bv;
}

DEVINLINE float block_sum(SampleLds* lds, float v) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  if (lane == 0) lds->u.scan[wave] = v;
  __syncthreads();
  float t = 0.f;
#pragma unroll
  for (int w = 0; w < BLOCK / 64; ++w) t += lds->u.scan[w];
  __syncthreads();
  return t;
}

__global__ __launch_bounds__(BLOCK) void sample_kernel(
    const bf16* __restrict__ logits,   // [B, V]
    const float* __restrict__ temps,   // [B]
    const float* __restrict__ top_ps,  // [B]
    const int* __restrict__ top_ks,    // [B]
    const float* __restrict__ pres,    // [B]
    const float* __restrict__ freqs,   // [B]
    int* __restrict__ counts,          // [B, V] or null
    const int64_t* __restrict__ seeds, // [B] or null
    const int64_t* __restrict__ ctr,   // [1] device counter or null
    int64_t* __restrict__ out,         // [B]
    uint64_t seed_base, int V) {
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const bf16* row = logits + (int64_t)b * V;
  int* cnt_row = counts ? counts + (int64_t)b * V : nullptr;
  const float T = temps[b];
  const float pp = pres[b], fp = freqs[b];
  const bool pen = cnt_row && (pp != 0.f || fp != 0.f);

  __shared__ SampleLds lds;

  // -------- penalized value loader --------
  auto zval = [&](int i) -> float {
    float z = bf2f(row[i]);
    if (pen) {
      int c = cnt_row[i];
      if (c > 0) z -= pp + fp * (float)c;
    }
    return z;
  };

  // -------- pass 1: max + argmax --------
  float mymax = -3.4e38f;
  int myidx = V;
  for (int i = tid; i < V; i += BLOCK) {
    float z = zval(i);
    if (z > mymax) { mymax = z; myidx = i; }
  }
  int amax;
  const float M = block_max_argmin(&lds, mymax, myidx, &amax);

  int chosen = amax;
  if (T > 0.f) {
    const float invT = 1.f / T;
    const int k_want = top_ks[b];
    const float p_want = top_ps[b];
    // -------- pass 2: level-1 histogram (key bits 31:24) --------
    for (int i = tid; i < 256; i += BLOCK) {
      lds.u.hist.cnt[i] = 0;
      lds.u.hist.sum[i] = 0ull;
    }
    __syncthreads();
    for (int i = tid; i < V; i += BLOCK) {
      float z = zval(i);
      float e = __expf((z - M) * invT);
      uint32_t bin = fkey(z) >> 24;
      atomicAdd(&lds.u.hist.cnt[bin], 1);
      atomicAdd(&lds.u.hist.sum[bin], wquant(e));
    }
    __syncthreads();
    // walk bins high->low on ONE thread (256 iterations, trivial)
    if (tid == 0) {
      unsigned long long Zi = 0ull;
      for (int i = 0; i < 256; ++i) Zi += lds.u.hist.sum[i];
      const unsigned long long pmass =
          p_want < 1.f ? (unsigned long long)(p_want * (double)Zi)
                       : ~0ull;
      long cum_n = 0;
      unsigned long long cum_e = 0ull;
      int bin_k = -1, bin_p = -1;     // boundary bins
      long above_k = 0;               // counts/mass strictly above bin
      unsigned long long above_p = 0ull;
      for (int i = 255; i >= 0; --i) {
        if (bin_k < 0 && k_want > 0 &&
            cum_n + lds.u.hist.cnt[i] >= k_want) {
          bin_k = i; above_k = cum_n;
        }
        if (bin_p < 0 && cum_e + lds.u.hist.sum[i] > pmass) {
          bin_p = i; above_p = cum_e;
        }
        cum_n += lds.u.hist.cnt[i];
        cum_e += lds.u.hist.sum[i];
        if (bin_k >= 0 && bin_p >= 0) break;
      }
      if (k_want <= 0 || k_want >= V) bin_k = 0;       // no k cut
      if (bin_k < 0) bin_k = 0;
      if (bin_p < 0) bin_p = 0;                        // no p cut
      lds.stat[1] = (float)bin_k;
      lds.stat[2] = (float)bin_p;
      lds.stat[3] = (float)above_k;   // exact up to 2^24 counts
      lds.statu[0] = above_p;
      lds.statu[1] = pmass;
    }
    __syncthreads();
    const int bin_k = (int)lds.stat[1];
    const int bin_p = (int)lds.stat[2];
    const unsigned long long pmass = lds.statu[1];

    // -------- pass 3: level-2 refinement for both cutoffs --------
    // elements whose level-1 bin == boundary get re-binned by key bits
    // 23:16; run for k and p boundaries (they often coincide)
    uint32_t thr_key = 0;        // keep keys >= thr_key
    float kept_mass = -1.f;      // computed below (full Z if no cuts)
    for (int phase = 0; phase < 2; ++phase) {
      const bool is_k = phase == 0;
      if (is_k && (k_want <= 0 || k_want >= V)) continue;
      if (!is_k && !(top_ps[b] < 1.f)) continue;
      const int bb = is_k ? bin_k : bin_p;
      for (int i = tid; i < 256; i += BLOCK) {
        lds.u.hist.cnt[i] = 0;
        lds.u.hist.sum[i] = 0ull;
      }
      __syncthreads();
      for (int i = tid; i < V; i += BLOCK) {
        float z = zval(i);
        uint32_t key = fkey(z);
        if ((int)(key >> 24) == bb) {
          float e = __expf((z - M) * invT);
          atomicAdd(&lds.u.hist.cnt[(key >> 16) & 0xFF], 1);
          atomicAdd(&lds.u.hist.sum[(key >> 16) & 0xFF], wquant(e));
        }
      }
      __syncthreads();
      if (tid == 0) {
        long cum_n = (long)lds.stat[3];
        unsigned long long cum_e = lds.statu[0];
        int sub = 0;
        if (is_k) {
          for (int i = 255; i >= 0; --i) {
            cum_n += lds.u.hist.cnt[i];
            if (cum_n >= k_want) { sub = i; break; }
          }
        } else {
          for (int i = 255; i >= 0; --i) {
            cum_e += lds.u.hist.sum[i];
            if (cum_e > pmass) { sub = i; break; }
          }
        }
        // keep keys >= boundary sub-bin start
        lds.stat[6] = (float)sub;
        lds.stat[7] = __uint_as_float(
            ((uint32_t)bb << 24) | ((uint32_t)sub << 16));
      }
      __syncthreads();
      uint32_t t = __float_as_uint(lds.stat[7]);
      if (t > thr_key) thr_key = t;
      __syncthreads();
    }
    // kept mass for the final threshold (fixed-order float pass)
    {
      float my = 0.f;
      for (int i = tid; i < V; i += BLOCK) {
        float z = zval(i);
        if (thr_key == 0 || fkey(z) >= thr_key)
          my += __expf((z - M) * invT);
      }
      kept_mass = block_sum(&lds, my);
    }

    // -------- pass 4: draw + index-ordered prefix walk --------
    uint64_t h;
    if (seeds) {
      // host passes seeds[b] = mix(request seed, request step): the
      // draw depends only on the request's own state, never on which
      // slot/batch it lands in
      h = splitmix64((uint64_t)seeds[b]);
    } else {
      uint64_t c = ctr ? (uint64_t)ctr[0] : 0ull;
      h = splitmix64(splitmix64(seed_base ^ (c * 0x9E3779B97F4A7C15ull))
                     + (uint64_t)b);
    }
    const float u = (float)((h >> 11) * (1.0 / 9007199254740992.0));
    const float target = u * kept_mass;
    // chunked block-ordered scan: 256 elements at a time
    float run = 0.f;
    chosen = -1;
    int last_kept = amax;
    for (int base = 0; base < V && chosen < 0; base += BLOCK) {
      const int i = base + tid;
      float e = 0.f;
      if (i < V) {
        float z = zval(i);
        if (thr_key == 0 || fkey(z) >= thr_key) e = __expf((z - M) * invT);
      }
      // wave-inclusive scan then wave offsets
      const int lane = tid & 63;
      const int wave = tid >> 6;
      float sc = e;
#pragma unroll
      for (int off = 1; off < 64; off <<= 1) {
        float o = __shfl_up(sc, off, 64);
        if (lane >= off) sc += o;
      }
      if (lane == 63) lds.u.scan[wave] = sc;
      __syncthreads();
      float woff = 0.f;
#pragma unroll
      for (int w = 0; w < BLOCK / 64; ++w) {
        if (w < wave) woff += lds.u.scan[w];
      }
      float tot = 0.f;
#pragma unroll
      for (int w = 0; w < BLOCK / 64; ++w) tot += lds.u.scan[w];
      __syncthreads();
      const float cum = run + woff + sc;        // inclusive prefix
      // first index whose inclusive cum >= target
      int cand = (e > 0.f && cum >= target) ? i : V;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        cand = min(cand, __shfl_xor(cand, off, 64));
      if (lane == 0) lds.u.red.i[wave] = cand;
      __syncthreads();
      int first = V;
#pragma unroll
      for (int w = 0; w < BLOCK / 64; ++w)
        first = min(first, lds.u.red.i[w]);
      __syncthreads();
      if (first < V) chosen = first;
      // remember the last kept token for roundoff fallback
      int lk = (e > 0.f) ? i : -1;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        lk = max(lk, __shfl_xor(lk, off, 64));
      if (lane == 0) lds.u.red.i[wave] = lk;
      __syncthreads();
      int blk_lk = -1;
#pragma unroll
      for (int w = 0; w < BLOCK / 64; ++w)
        blk_lk = max(blk_lk, lds.u.red.i[w]);
      __syncthreads();
      if (blk_lk >= 0) last_kept = blk_lk;
      run += tot;
    }
    if (chosen < 0) chosen = last_kept;   // fp roundoff tail
  }

  if (tid == 0) {
    out[b] = chosen;
    if (cnt_row) atomicAdd(&cnt_row[chosen], 1);
  }
}

__global__ void bump_counter_kernel(int64_t* ctr) {
  if (threadIdx.x == 0 && blockIdx.x == 0) ctr[0] += 1;
}

}  // namespace

extern "C" {

void tl_sample(const void* logits, const void* temps, const void* top_ps,
               const void* top_ks, const void* pres, const void* freqs,
               void* counts, const void* seeds, const void* ctr, void* out,
               uint64_t seed_base, int B, int V, hipStream_t stream) {
  hipLaunchKernelGGL(sample_kernel, dim3(B), dim3(BLOCK), 0, stream,
                     (const bf16*)logits, (const float*)temps,
                     (const float*)top_ps, (const int*)top_ks,
                     (const float*)pres, (const float*)freqs, (int*)counts,
                     (const int64_t*)seeds, (const int64_t*)ctr,
                     (int64_t*)out, seed_base, V);
}

void tl_bump_counter(void* ctr, hipStream_t stream) {
  hipLaunchKernelGGL(bump_counter_kernel, dim3(1), dim3(64), 0, stream,
                     (int64_t*)ctr);
}

}  // extern "C"
