// Python bindings for the tensorlink_amd CDNA4 kernel library.
// Host-side glue only — every kernel lives in the sibling .hip files and is
// reached through an extern "C" launcher taking raw pointers + hipStream_t.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

extern "C" {
void tl_rmsnorm_fwd(const void* x, const void* residual, const void* w,
                    void* y, void* r_out, void* rstd, int64_t N, int H,
                    float eps, hipStream_t stream);
void tl_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                    const void* rstd, void* dx, void* dw, void* dw_partial,
                    int n_blocks, int64_t N, int H, hipStream_t stream);
void tl_rope(void* q, void* k, const void* positions, const void* inv_freq,
             int64_t T, int Hq, int Hkv, int D, float sign,
             hipStream_t stream);
void tl_swiglu_fwd(const void* gate, const void* up, void* out, int64_t n,
                   hipStream_t stream);
void tl_swiglu_bwd(const void* dout, const void* gate, const void* up,
                   void* dgate, void* dup, int64_t n, hipStream_t stream);
void tl_adamw(void* param, const void* grad, void* m, void* v, int64_t n,
              int is_bf16, float lr, float beta1, float beta2, float eps,
              float weight_decay, int step, hipStream_t stream);
void tl_decode_attn(const void* q, const void* k_cache, const void* v_cache,
                    const void* seq_lens, void* out, int B, int Hq, int Hkv,
                    int Smax, int D, float scale, hipStream_t stream);
void tl_prefill_attn(const void* q, const void* k, const void* v, void* out,
                     int B, int Sq, int Skv, int q_off, int Hq, int Hkv,
                     int D, float scale, int causal, hipStream_t stream);
void tl_rope_append(const void* qkv, void* q_out, void* k_cache,
                    void* v_cache, const void* positions,
                    const void* inv_freq, const void* block_table,
                    int64_t T, int row_stride, int S, int Hq, int Hkv,
                    int D, int Smax, int bt_stride, hipStream_t stream);
void tl_swiglu_fused(const void* gu, void* out, int64_t N, int I,
                     hipStream_t stream);
void tl_decode_attn_mfma(const void* q, const void* k_cache,
                         const void* v_cache, const void* seq_lens,
                         const void* block_table, void* out,
                         void* partial, void* partial_ml, int B, int Hq,
                         int Hkv, int Smax, int D, float scale, int n_split,
                         int bt_stride, int per_row, hipStream_t stream);
void tl_skinny_gemm(const void* x, const void* w, const void* bias,
                    void* out, void* partial, int M, int N, int K,
                    int n_split, hipStream_t stream);
void tl_gemm_tiled(const void* x, const void* w, const void* bias, void* out,
                   void* partial, int M, int N, int K, int n_split,
                   hipStream_t stream);
void tl_sample(const void* logits, const void* temps, const void* top_ps,
               const void* top_ks, const void* pres, const void* freqs,
               void* counts, const void* seeds, const void* ctr, void* out,
               uint64_t seed_base, int B, int V, hipStream_t stream);
void tl_bump_counter(void* ctr, hipStream_t stream);
void tl_moe_gemm(const void* x, const void* tok_idx, const void* seg_off,
                 const void* w_ptrs, const void* s_ptrs, void* out, int E,
                 int N, int K, int fp8, hipStream_t stream);
void tl_prefill_attn_lse(const void* q, const void* k, const void* v,
                         void* out, void* lse, int B, int Sq, int Skv,
                         int q_off, int Hq, int Hkv, int D, float scale,
                         int causal, hipStream_t stream);
void tl_attn_bwd(const void* q, const void* k, const void* v,
                 const void* dout, const void* lse, const void* delta,
                 void* dq, void* dk, void* dv, int B, int S, int Hq,
                 int Hkv, int D, float scale, int causal,
                 hipStream_t stream);
}

namespace {

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// host copy of tl_split_for_len (common.hpp — keep in sync; this file is
// compiled by plain g++ so it cannot include the device header)
int tl_split_for_len(int L) {
  int ns = 1;
  while (ns < 16 && L > 1024 * ns) ns <<= 1;
  return ns;
}

#define CHECK_IN(t, d)                                              \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                 \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");       \
  TORCH_CHECK((t).scalar_type() == (d), #t " wrong dtype")

using torch::Tensor;

std::vector<Tensor> rmsnorm_fwd(Tensor x, c10::optional<Tensor> residual,
                                Tensor w, double eps, bool save_rstd) {
  CHECK_IN(x, torch::kBFloat16);
  CHECK_IN(w, torch::kBFloat16);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H must be divisible by 8");
  const int64_t N = x.numel() / H;
  auto y = torch::empty_like(x);
  Tensor rstd, r_out;
  if (save_rstd) rstd = torch::empty({N}, x.options().dtype(torch::kFloat));
  const void* res_ptr = nullptr;
  void* rout_ptr = nullptr;
  if (residual.has_value()) {
    CHECK_IN(residual.value(), torch::kBFloat16);
    r_out = torch::empty_like(x);
    res_ptr = residual->data_ptr();
    rout_ptr = r_out.data_ptr();
  }
  tl_rmsnorm_fwd(x.data_ptr(), res_ptr, w.data_ptr(), y.data_ptr(), rout_ptr,
                 save_rstd ? rstd.data_ptr() : nullptr, N, H, (float)eps,
                 cur_stream());
  std::vector<Tensor> outs = {y};
  if (residual.has_value()) outs.push_back(r_out);
  if (save_rstd) outs.push_back(rstd);
  return outs;
}

std::vector<Tensor> rmsnorm_bwd(Tensor dy, Tensor x, Tensor w, Tensor rstd) {
  CHECK_IN(dy, torch::kBFloat16);
  CHECK_IN(x, torch::kBFloat16);
  CHECK_IN(w, torch::kBFloat16);
  CHECK_IN(rstd, torch::kFloat);
  const int H = x.size(-1);
  const int64_t N = x.numel() / H;
  const int n_blocks = (int)std::min<int64_t>(N, 512);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto partial = torch::empty({n_blocks, (int64_t)H},
                              x.options().dtype(torch::kFloat));
  tl_rmsnorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr(), rstd.data_ptr(),
                 dx.data_ptr(), dw.data_ptr(), partial.data_ptr(), n_blocks,
                 N, H, cur_stream());
  return {dx, dw};
}

void rope_(Tensor q, Tensor k, Tensor positions, Tensor inv_freq,
           double sign) {
  CHECK_IN(q, torch::kBFloat16);
  CHECK_IN(k, torch::kBFloat16);
  CHECK_IN(positions, torch::kInt);
  CHECK_IN(inv_freq, torch::kFloat);
  const int D = q.size(-1);
  const int Hq = q.size(-2);
  const int Hkv = k.size(-2);
  const int64_t T = q.numel() / ((int64_t)Hq * D);
  TORCH_CHECK(k.numel() / ((int64_t)Hkv * D) == T, "q/k token mismatch");
  TORCH_CHECK(positions.numel() == T, "positions size mismatch");
  tl_rope(q.data_ptr(), k.data_ptr(), positions.data_ptr(),
          inv_freq.data_ptr(), T, Hq, Hkv, D, (float)sign, cur_stream());
}

Tensor swiglu_fwd(Tensor gate, Tensor up) {
  CHECK_IN(gate, torch::kBFloat16);
  CHECK_IN(up, torch::kBFloat16);
  TORCH_CHECK(gate.numel() % 8 == 0, "numel must be divisible by 8");
  auto out = torch::empty_like(gate);
  tl_swiglu_fwd(gate.data_ptr(), up.data_ptr(), out.data_ptr(), gate.numel(),
                cur_stream());
  return out;
}

std::vector<Tensor> swiglu_bwd(Tensor dout, Tensor gate, Tensor up) {
  CHECK_IN(dout, torch::kBFloat16);
  auto dg = torch::empty_like(gate);
  auto du = torch::empty_like(up);
  tl_swiglu_bwd(dout.data_ptr(), gate.data_ptr(), up.data_ptr(), dg.data_ptr(),
                du.data_ptr(), gate.numel(), cur_stream());
  return {dg, du};
}

void adamw_(Tensor param, Tensor grad, Tensor m, Tensor v, double lr,
            double beta1, double beta2, double eps, double weight_decay,
            int64_t step) {
  TORCH_CHECK(param.is_cuda() && param.is_contiguous());
  TORCH_CHECK(m.scalar_type() == torch::kFloat &&
              v.scalar_type() == torch::kFloat);
  TORCH_CHECK(grad.scalar_type() == param.scalar_type());
  const bool is_bf16 = param.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(is_bf16 || param.scalar_type() == torch::kFloat);
  tl_adamw(param.data_ptr(), grad.data_ptr(), m.data_ptr(), v.data_ptr(),
           param.numel(), is_bf16 ? 1 : 0, (float)lr, (float)beta1,
           (float)beta2, (float)eps, (float)weight_decay, (int)step,
           cur_stream());
}

Tensor decode_attn(Tensor q, Tensor k_cache, Tensor v_cache, Tensor seq_lens,
                   double scale, int64_t n_split,
                   c10::optional<Tensor> block_table) {
  CHECK_IN(q, torch::kBFloat16);
  CHECK_IN(k_cache, torch::kBFloat16);
  CHECK_IN(v_cache, torch::kBFloat16);
  CHECK_IN(seq_lens, torch::kInt);
  const int B = q.size(0), Hq = q.size(1), D = q.size(2);
  const int Hkv = k_cache.size(1), Smax = k_cache.size(2);
  const int G = Hq / Hkv;
  const void* bt = nullptr;
  int bt_stride = 0;
  if (block_table.has_value()) {
    CHECK_IN(block_table.value(), torch::kInt);
    TORCH_CHECK(G <= 16, "paged decode requires GQA group <= 16");
    TORCH_CHECK(k_cache.size(2) == 128, "paged pool PAGE must be 128");
    bt = block_table->data_ptr();
    bt_stride = block_table->size(1);
  }
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  auto out = torch::empty_like(q);
  const bool legacy = (getenv("TL_DECODE_LEGACY") != nullptr || G > 16) &&
                      bt == nullptr;
  if (legacy) {
    tl_decode_attn(q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
                   seq_lens.data_ptr(), out.data_ptr(), B, Hq, Hkv, Smax, D,
                   (float)scale, cur_stream());
    return out;
  }
  int per_row = 0;
  if (n_split <= 0) {
    // auto: per-ROW split from each row's own length (tl_split_for_len,
    // common.hpp) so a sequence's numerics are batch-independent; the
    // launch z-width comes from a batch-shape-free host bound on L
    // (cache capacity), no device sync needed.
    per_row = 1;
    const int upper = bt ? bt_stride * 128 : Smax;
    n_split = tl_split_for_len(upper);
  }
  TORCH_CHECK(n_split <= 64, "n_split too large");
  Tensor partial, partial_ml;
  void *pp = nullptr, *pml = nullptr;
  if (n_split > 1) {
    partial = torch::empty({(int64_t)B * Hq * n_split * D},
                           q.options().dtype(torch::kFloat));
    partial_ml = torch::empty({(int64_t)B * Hq * n_split * 2},
                              q.options().dtype(torch::kFloat));
    pp = partial.data_ptr();
    pml = partial_ml.data_ptr();
  }
  tl_decode_attn_mfma(q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
                      seq_lens.data_ptr(), bt, out.data_ptr(), pp, pml, B,
                      Hq, Hkv, Smax, D, (float)scale, (int)n_split,
                      bt_stride, per_row, cur_stream());
  return out;
}

Tensor prefill_attn(Tensor q, Tensor k, Tensor v, double scale, bool causal,
                    int64_t q_off) {
  CHECK_IN(q, torch::kBFloat16);
  CHECK_IN(k, torch::kBFloat16);
  CHECK_IN(v, torch::kBFloat16);
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(q_off + Sq <= Skv || !causal, "q_off + Sq must fit Skv");
  auto out = torch::empty_like(q);
  tl_prefill_attn(q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
                  B, Sq, Skv, (int)q_off, Hq, Hkv, D, (float)scale,
                  causal ? 1 : 0, cur_stream());
  return out;
}

std::vector<Tensor> prefill_attn_lse(Tensor q, Tensor k, Tensor v,
                                     double scale, bool causal) {
  CHECK_IN(q, torch::kBFloat16);
  CHECK_IN(k, torch::kBFloat16);
  CHECK_IN(v, torch::kBFloat16);
  const int B = q.size(0), Sq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Skv = k.size(1);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  auto out = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, Sq}, q.options().dtype(torch::kFloat));
  tl_prefill_attn_lse(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                      out.data_ptr(), lse.data_ptr(), B, Sq, Skv, 0, Hq,
                      Hkv, D, (float)scale, causal ? 1 : 0, cur_stream());
  return {out, lse};
}

std::vector<Tensor> attn_bwd(Tensor q, Tensor k, Tensor v, Tensor dout,
                             Tensor lse, Tensor delta, double scale,
                             bool causal) {
  CHECK_IN(q, torch::kBFloat16);
  CHECK_IN(k, torch::kBFloat16);
  CHECK_IN(v, torch::kBFloat16);
  CHECK_IN(dout, torch::kBFloat16);
  CHECK_IN(lse, torch::kFloat);
  CHECK_IN(delta, torch::kFloat);
  const int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(k.size(1) == S, "training attention needs Sq == Skv");
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  tl_attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), dout.data_ptr(),
              lse.data_ptr(), delta.data_ptr(), dq.data_ptr(),
              dk.data_ptr(), dv.data_ptr(), B, S, Hq, Hkv, D, (float)scale,
              causal ? 1 : 0, cur_stream());
  return {dq, dk, dv};
}

// qkv: [T, row_stride] fused projection output (q | k | v per row).
// Returns rotated q as a contiguous [T, Hq, D] tensor; k/v land in the
// caches.
Tensor rope_append_(Tensor qkv, Tensor k_cache, Tensor v_cache,
                    Tensor positions, Tensor inv_freq, int64_t S,
                    int64_t Hq, int64_t Hkv,
                    c10::optional<Tensor> block_table) {
  CHECK_IN(qkv, torch::kBFloat16);
  CHECK_IN(k_cache, torch::kBFloat16);
  CHECK_IN(v_cache, torch::kBFloat16);
  CHECK_IN(positions, torch::kInt);
  CHECK_IN(inv_freq, torch::kFloat);
  const int D = k_cache.size(3);
  const int row_stride = qkv.size(-1);
  TORCH_CHECK(row_stride == (Hq + 2 * Hkv) * D, "qkv width mismatch");
  const int64_t T = qkv.numel() / row_stride;
  const int Smax = k_cache.size(2);
  TORCH_CHECK(T % S == 0, "T must be divisible by S");
  TORCH_CHECK(D % 16 == 0, "D must be divisible by 16");
  const void* bt = nullptr;
  int bt_stride = 0;
  if (block_table.has_value()) {
    CHECK_IN(block_table.value(), torch::kInt);
    TORCH_CHECK(k_cache.size(2) == 128, "paged pool PAGE must be 128");
    bt = block_table->data_ptr();
    bt_stride = block_table->size(1);
    TORCH_CHECK(block_table->size(0) * S == T, "table batch mismatch");
  } else {
    TORCH_CHECK(k_cache.size(0) * S == T, "cache batch mismatch");
  }
  auto q_out = torch::empty({T, Hq, (int64_t)D}, qkv.options());
  tl_rope_append(qkv.data_ptr(), q_out.data_ptr(), k_cache.data_ptr(),
                 v_cache.data_ptr(), positions.data_ptr(),
                 inv_freq.data_ptr(), bt, T, row_stride, (int)S, (int)Hq,
                 (int)Hkv, D, Smax, bt_stride, cur_stream());
  return q_out;
}

Tensor swiglu_fused(Tensor gu) {
  CHECK_IN(gu, torch::kBFloat16);
  const int64_t I2 = gu.size(-1);
  TORCH_CHECK(I2 % 16 == 0, "fused width must be divisible by 16");
  const int64_t I = I2 / 2;
  const int64_t N = gu.numel() / I2;
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = torch::empty(sizes, gu.options());
  tl_swiglu_fused(gu.data_ptr(), out.data_ptr(), N, (int)I, cur_stream());
  return out;
}

// Split-K policy SHARED by the streaming (M<=64) and tiled (M>64) GEMM
// kernels. It depends only on (N, K) — never M — so a row's accumulation
// (split boundaries + s-ordered reduce) is bitwise identical at every M:
// the serving engine's exact-greedy guarantee rests on this.
// split-block target (~blocks before M/z tiling); env-tunable for
// in-pipeline A/B — read ONCE so the policy stays (N,K)-pure in-process
static int gemm_split_target() {
  static int t = [] {
    const char* e = getenv("TL_GEMM_SPLIT_TARGET");
    return e ? atoi(e) : 256;
  }();
  return t;
}

static int gemm_n_split(int N, int K) {
  if (N >= 65536) {
    // 256x256-tile class (gemm_tiled_sq, lm_head widths): enough
    // column panels to fill the chip without splitting
    int n_split = 1;
    while (((N + 255) / 256) * n_split < gemm_split_target() &&
           n_split < 16 && (K / 64) / (n_split * 2) >= 16)
      n_split <<= 1;
    return n_split;
  }
  // split K so (N/64)*n_split lands near the target block count
  int n_split = 1;
  while ((N / 64) * n_split < gemm_split_target() && n_split < 8 &&
         (K / 32) / (n_split * 2) >= 2)
    n_split <<= 1;
  return n_split;
}

Tensor skinny_gemm(Tensor x, Tensor w, c10::optional<Tensor> bias) {
  CHECK_IN(x, torch::kBFloat16);
  CHECK_IN(w, torch::kBFloat16);
  const int K = x.size(-1);
  const int64_t M = x.numel() / K;
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  TORCH_CHECK(K % 32 == 0 && N % 64 == 0, "unsupported shape");
  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto out = torch::empty(sizes, x.options());
  const void* bp = nullptr;
  if (bias.has_value()) {
    CHECK_IN(bias.value(), torch::kBFloat16);
    bp = bias->data_ptr();
  }
  const int n_split = gemm_n_split(N, K);
  Tensor partial;
  void* pp = nullptr;
  if (n_split > 1) {
    partial = torch::empty({(int64_t)n_split * M * N},
                           x.options().dtype(torch::kFloat));
    pp = partial.data_ptr();
  }
  if (M <= 64)
    tl_skinny_gemm(x.data_ptr(), w.data_ptr(), bp, out.data_ptr(), pp,
                   (int)M, N, K, n_split, cur_stream());
  else
    tl_gemm_tiled(x.data_ptr(), w.data_ptr(), bp, out.data_ptr(), pp,
                  (int)M, N, K, n_split, cur_stream());
  return out;
}

Tensor sample_tokens(Tensor logits, Tensor temps, Tensor top_ps,
                     Tensor top_ks, Tensor pres, Tensor freqs,
                     c10::optional<Tensor> counts,
                     c10::optional<Tensor> seeds,
                     c10::optional<Tensor> counter, int64_t seed_base) {
  CHECK_IN(logits, torch::kBFloat16);
  CHECK_IN(temps, torch::kFloat);
  CHECK_IN(top_ps, torch::kFloat);
  CHECK_IN(top_ks, torch::kInt);
  CHECK_IN(pres, torch::kFloat);
  CHECK_IN(freqs, torch::kFloat);
  const int B = logits.size(0);
  const int V = logits.size(1);
  void* cnt = nullptr;
  if (counts.has_value()) {
    CHECK_IN(counts.value(), torch::kInt);
    TORCH_CHECK(counts->size(0) == B && counts->size(1) == V,
                "counts must be [B, V]");
    cnt = counts->data_ptr();
  }
  const void* sd = nullptr;
  if (seeds.has_value()) {
    CHECK_IN(seeds.value(), torch::kLong);
    sd = seeds->data_ptr();
  }
  const void* ct = nullptr;
  if (counter.has_value()) {
    CHECK_IN(counter.value(), torch::kLong);
    ct = counter->data_ptr();
  }
  auto out = torch::empty({B}, logits.options().dtype(torch::kLong));
  tl_sample(logits.data_ptr(), temps.data_ptr(), top_ps.data_ptr(),
            top_ks.data_ptr(), pres.data_ptr(), freqs.data_ptr(), cnt, sd,
            ct, out.data_ptr(), (uint64_t)seed_base, B, V, cur_stream());
  return out;
}

void bump_sample_counter(Tensor ctr) {
  CHECK_IN(ctr, torch::kLong);
  tl_bump_counter(ctr.data_ptr(), cur_stream());
}

Tensor moe_gemm(Tensor x, c10::optional<Tensor> tok_idx, Tensor seg_off,
                Tensor w_ptrs, c10::optional<Tensor> s_ptrs, int64_t N,
                bool fp8) {
  CHECK_IN(x, torch::kBFloat16);
  CHECK_IN(seg_off, torch::kInt);
  CHECK_IN(w_ptrs, torch::kLong);
  const int K = x.size(-1);
  TORCH_CHECK(K % 32 == 0 && N % 64 == 0, "unsupported MoE shape");
  const int E = w_ptrs.size(0);
  TORCH_CHECK(seg_off.size(0) == E + 1, "seg_off must be [E+1]");
  const void* ti = nullptr;
  int64_t P;
  if (tok_idx.has_value()) {
    CHECK_IN(tok_idx.value(), torch::kInt);
    ti = tok_idx->data_ptr();
    P = tok_idx->size(0);
  } else {
    P = x.size(0);
  }
  const void* sp = nullptr;
  if (fp8) {
    TORCH_CHECK(s_ptrs.has_value(), "fp8 needs scale pointers");
    CHECK_IN(s_ptrs.value(), torch::kLong);
    sp = s_ptrs->data_ptr();
  }
  auto out = torch::empty({P, N}, x.options());
  tl_moe_gemm(x.data_ptr(), ti, seg_off.data_ptr(), w_ptrs.data_ptr(), sp,
              out.data_ptr(), E, (int)N, K, fp8 ? 1 : 0, cur_stream());
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("rope_append_", &rope_append_,
          "fused RoPE + KV-cache append from fused-QKV rows");
  mod.def("swiglu_fused", &swiglu_fused, "silu-mul on fused [.,2I] rows");
  mod.def("skinny_gemm", &skinny_gemm, "decode-M GEMM: x @ W^T + bias");
  mod.def("rmsnorm_fwd", &rmsnorm_fwd, "fused RMSNorm fwd (+residual)");
  mod.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm bwd");
  mod.def("rope_", &rope_, "in-place RoPE apply");
  mod.def("swiglu_fwd", &swiglu_fwd, "silu(gate)*up");
  mod.def("swiglu_bwd", &swiglu_bwd, "SwiGLU bwd");
  mod.def("adamw_", &adamw_, "fused AdamW step");
  mod.def("decode_attn", &decode_attn, "GQA decode attention over KV cache");
  mod.def("prefill_attn", &prefill_attn, "causal GQA prefill attention");
  mod.def("sample_tokens", &sample_tokens,
          "fused penalties/temperature/top-k/top-p sampling, one block/row");
  mod.def("bump_sample_counter", &bump_sample_counter,
          "advance the graph-safe sampling RNG counter");
  mod.def("moe_gemm", &moe_gemm,
          "grouped expert GEMM over expert-sorted pair rows");
  mod.def("prefill_attn_lse", &prefill_attn_lse,
          "causal GQA prefill attention returning (out, logsumexp)");
  mod.def("attn_bwd", &attn_bwd,
          "flash-attention backward: (dq, dk, dv) from saved lse/delta");
}
