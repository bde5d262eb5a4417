"""Native decoder model zoo (Llama / Qwen2 / Qwen3 dense families).

The reference runs HuggingFace ``transformers`` modules eagerly per stage
(``tensorlink/ml/worker.py:326-335``) and ships rotary buffers / KV caches
over the network. Here the decoder is defined natively on top of the
CDNA4 op library (:mod:`tensorlink_amd.ops`): fused residual+RMSNorm, RoPE
computed on device, flash prefill + KV-cache decode attention, fused
SwiGLU — GEMMs go through torch.matmul (hipBLASLt on ROCm).

A model instance holds a contiguous *slice* of layers plus optionally the
embedding (first stage) and final-norm/lm_head (last stage) — the unit the
planner assigns to one GPU rank (cf. reference grouped entries
``model.layers.N-M``, ``ml/graphing.py:64-128``).
"""

from __future__ import annotations

import dataclasses

import math
from typing import List, Optional

import torch
import torch.nn as nn

from tensorlink_amd import ops
from tensorlink_amd.models.configs import ModelConfig


class KVCache:
    """Device-resident contiguous KV cache for one stage.

    Layout [B, Hkv, Smax, D] per layer (head-major so decode streams a
    contiguous [L, D] slab per (batch, head) — see decode_attn.hip).
    The reference serializes its HF DynamicCache over TCP every step
    (``ml/utils.py:210-221``); this cache never leaves the GPU.
    """

    def __init__(self, n_layers: int, batch: int, max_seq: int, config,
                 device, dtype=torch.bfloat16):
        self.k = [torch.zeros(batch, config.num_key_value_heads, max_seq,
                              config.head_dim, device=device, dtype=dtype)
                  for _ in range(n_layers)]
        self.v = [torch.zeros_like(self.k[0]) for _ in range(n_layers)]
        self.seq_lens = torch.zeros(batch, device=device, dtype=torch.int32)
        self.max_seq = max_seq
        self.batch = batch

    def reset(self):
        self.seq_lens.zero_()

    def append(self, layer: int, k_new: torch.Tensor, v_new: torch.Tensor,
               positions: torch.Tensor):
        """Write new K/V at given positions. k_new [B, S_new, Hkv, D];
        positions [B, S_new] int."""
        B, S_new, Hkv, D = k_new.shape
        idx = positions.to(self.k[layer].device, torch.long)
        idx = idx.view(B, 1, S_new, 1).expand(B, Hkv, S_new, D)
        self.k[layer].scatter_(2, idx, k_new.transpose(1, 2))
        self.v[layer].scatter_(2, idx, v_new.transpose(1, 2))

    def advance(self, n: int):
        self.seq_lens += n


class TLLinear(nn.Linear):
    """nn.Linear whose inference forward routes decode-M shapes to the
    hand-written MFMA skinny GEMM (ops/csrc/skinny_gemm.hip)."""

    def forward(self, x):
        return ops.linear(x, self.weight, self.bias)


def compute_inv_freq(config: ModelConfig) -> torch.Tensor:
    """RoPE inverse frequencies, honoring HF rope_scaling. Implements
    the Llama-3.1 "llama3" spectrum scaling (long wavelengths divided by
    `factor`, short kept, smooth ramp between — matches transformers'
    _compute_llama3_parameters); the kernels consume the resulting
    buffer directly, so scaled-rope models need no kernel change."""
    D = config.head_dim
    inv = 1.0 / (config.rope_theta ** (
        torch.arange(0, D, 2, dtype=torch.float32) / D))
    rs = config.rope_scaling
    if rs and rs.get("rope_type", rs.get("type")) == "llama3":
        factor = rs["factor"]
        lo, hi = rs["low_freq_factor"], rs["high_freq_factor"]
        orig = rs["original_max_position_embeddings"]
        wavelen = 2 * math.pi / inv
        smooth = (orig / wavelen - lo) / (hi - lo)
        smoothed = (1 - smooth) / factor * inv + smooth * inv
        inv = torch.where(wavelen > orig / lo, inv / factor, inv)
        mid = (wavelen <= orig / lo) & (wavelen >= orig / hi)
        inv = torch.where(mid, smoothed, inv)
    return inv


class Attention(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        self.config = config
        h, q, kv = config.hidden_size, config.q_size, config.kv_size
        bias = config.qkv_bias
        # fused qkv: ONE [q+2kv, h] GEMM instead of three (checkpoint
        # loading maps HF q/k/v rows into slices — models/loader.py)
        self.qkv_proj = TLLinear(h, q + 2 * kv, bias=bias)
        self.o_proj = TLLinear(q, h, bias=False)
        self.q_size, self.kv_size = q, kv
        self.n_heads = config.num_attention_heads
        self.n_kv = config.num_key_value_heads
        self.head_dim = config.head_dim
        self.scale = 1.0 / math.sqrt(self.head_dim)
        # Qwen3 applies RMSNorm to q/k per head
        self.use_qk_norm = config.architecture == "qwen3"
        if self.use_qk_norm:
            self.q_norm = nn.Parameter(torch.ones(self.head_dim))
            self.k_norm = nn.Parameter(torch.ones(self.head_dim))
        self.register_buffer("inv_freq", compute_inv_freq(config),
                             persistent=False)

    def _split_qkv(self, qkv, B, S):
        q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size],
                            dim=-1)
        q = q.reshape(B, S, self.n_heads, self.head_dim)
        k = k.reshape(B, S, self.n_kv, self.head_dim)
        v = v.reshape(B, S, self.n_kv, self.head_dim)
        return q, k, v

    def forward(self, x: torch.Tensor, positions: torch.Tensor,
                kv_cache: Optional[KVCache] = None, layer_idx: int = 0,
                training: bool = False) -> torch.Tensor:
        B, S, H = x.shape
        eps = self.config.rms_norm_eps
        qkv = self.qkv_proj(x)
        flat_pos = positions.reshape(-1)

        if training or kv_cache is None:
            q, k, v = self._split_qkv(qkv, B, S)
            if self.use_qk_norm:
                q = ops.rmsnorm(q.contiguous(), self.q_norm.to(q.dtype), eps)
                k = ops.rmsnorm(k.contiguous(), self.k_norm.to(k.dtype), eps)
            if training:
                q2, k2 = ops.apply_rope(q.reshape(B * S, self.n_heads, -1),
                                        k.reshape(B * S, self.n_kv, -1),
                                        flat_pos, self.inv_freq)
                q = q2.view(B, S, self.n_heads, -1)
                k = k2.view(B, S, self.n_kv, -1)
                out = ops.attention_train(q, k, v, causal=True,
                                          scale=self.scale)
            else:
                q = q.contiguous()
                k = k.contiguous()
                ops.apply_rope_(q.view(B * S, self.n_heads, -1),
                                k.view(B * S, self.n_kv, -1), flat_pos,
                                self.inv_freq)
                out = ops.attention_prefill(q, k, v.contiguous(),
                                            causal=True, scale=self.scale)
            return self.o_proj(out.reshape(B, S, -1))

        # inference with KV cache: fused rope+append straight off the
        # fused-QKV rows (q comes back contiguous, k/v land in the cache;
        # paged caches pass their block table through to the kernels)
        table = getattr(kv_cache, "table", None)
        if self.use_qk_norm:
            q, k, v = self._split_qkv(qkv, B, S)
            q = ops.rmsnorm(q.contiguous(), self.q_norm.to(q.dtype), eps)
            k = ops.rmsnorm(k.contiguous(), self.k_norm.to(k.dtype), eps)
            qkv = torch.cat([q.reshape(B, S, -1), k.reshape(B, S, -1),
                             v.reshape(B, S, -1)], dim=-1)
        q = ops.rope_append_(qkv.reshape(B * S, -1), kv_cache.k[layer_idx],
                             kv_cache.v[layer_idx], flat_pos, self.inv_freq,
                             S, self.n_heads, self.n_kv, block_table=table)
        q = q.view(B, S, self.n_heads, self.head_dim)
        if S == 1:
            out = ops.attention_decode(
                q, kv_cache.k[layer_idx], kv_cache.v[layer_idx],
                kv_cache.seq_lens + 1, scale=self.scale, block_table=table)
        else:
            # prefill with cache write; rotated k / raw v are read back from
            # the cache. off > 0 = chunked prefill continuing an existing
            # sequence (the chunk's queries sit at global rows off..off+S-1
            # and attend to all off+S cached keys). A single q_off serves
            # the whole view, so cached lengths must agree per row (ragged
            # multi-token work goes through the decode path instead).
            if B > 1:
                assert int(kv_cache.seq_lens.min()) == \
                    int(kv_cache.seq_lens.max()), \
                    "ragged cached prefill needs per-row q_off"
            off = int(kv_cache.seq_lens.max())
            kv_len = off + S
            if table is None:
                k_attn = kv_cache.k[layer_idx][:, :, :kv_len].permute(
                    0, 2, 1, 3).contiguous()
                v_attn = kv_cache.v[layer_idx][:, :, :kv_len].permute(
                    0, 2, 1, 3).contiguous()
            else:
                k_attn, v_attn = kv_cache.gather_contiguous(layer_idx, kv_len)
            out = ops.attention_prefill(q, k_attn, v_attn, causal=True,
                                        scale=self.scale, q_off=off)
        return self.o_proj(out.reshape(B, S, -1))


class MLP(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        self.inter = i
        # fused gate+up: ONE [2i, h] GEMM (HF gate/up rows map to slices)
        self.gate_up_proj = TLLinear(h, 2 * i, bias=False)
        self.down_proj = TLLinear(i, h, bias=False)

    def forward(self, x):
        gu = self.gate_up_proj(x)
        if torch.is_grad_enabled() and gu.requires_grad:
            gate, up = gu.split([self.inter, self.inter], dim=-1)
            return self.down_proj(ops.swiglu(gate.contiguous(),
                                             up.contiguous()))
        return self.down_proj(ops.swiglu_fused(gu))


class MoEMLP(nn.Module):
    """Mixtral-style sparse MoE block (new capability — the reference has no
    MoE-aware code)."""

    def __init__(self, config: ModelConfig):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        self.num_experts = config.num_local_experts
        self.top_k = config.num_experts_per_tok
        self.gate = nn.Linear(h, self.num_experts, bias=False)
        expert_cfg = (config if config.expert_intermediate_size == i
                      else dataclasses.replace(
                          config,
                          intermediate_size=config.expert_intermediate_size))
        self.experts = nn.ModuleList([MLP(expert_cfg)
                                      for _ in range(self.num_experts)])

    def _fused_kind(self):
        """"bf16" / "fp8" when every expert is uniform and the grouped
        kernel (ops/csrc/moe_gemm.hip) can serve this block; None falls
        back to the per-expert loop (CPU, training, fp4)."""
        from tensorlink_amd.models.quant import Fp8Linear
        e0 = self.experts[0]
        if isinstance(e0.gate_up_proj, Fp8Linear):
            return "fp8"
        if isinstance(e0.gate_up_proj, nn.Linear) \
                and e0.gate_up_proj.bias is None:
            return "bf16"
        return None

    def _tables(self, dev, fp8):
        """Device pointer tables into the per-expert weights (no stacked
        copy); rebuilt if weights moved."""
        gus = [(e.gate_up_proj.weight_fp8 if fp8 else e.gate_up_proj.weight)
               for e in self.experts]
        key = (str(dev), fp8, gus[0].data_ptr())
        if getattr(self, "_tbl_key", None) != key:
            downs = [(e.down_proj.weight_fp8 if fp8 else e.down_proj.weight)
                     for e in self.experts]
            mk = lambda ts: torch.tensor([t.data_ptr() for t in ts],
                                         dtype=torch.int64, device=dev)
            self._gu_ptrs = mk(gus)
            self._down_ptrs = mk(downs)
            if fp8:
                self._gu_scales = mk([e.gate_up_proj.scale
                                      for e in self.experts])
                self._down_scales = mk([e.down_proj.scale
                                        for e in self.experts])
            else:
                self._gu_scales = self._down_scales = None
            self._tbl_key = key
        return (self._gu_ptrs, self._down_ptrs, self._gu_scales,
                self._down_scales)

    def _fused_forward(self, flat, rw, idx, fp8):
        """Grouped expert GEMMs over expert-sorted (token, slot) pairs;
        deterministic combine: inverse-permutation gather then a fixed
        slot-order sum (no atomics — batcher greedy equality holds for
        MoE models too). hipGraph-capture-safe end to end."""
        C = ops._require_ext()
        E, k = self.num_experts, self.top_k
        T, H = flat.shape
        P = T * k
        dev = flat.device
        gu_p, down_p, gu_s, down_s = self._tables(dev, fp8)
        fi = idx.reshape(-1)
        order = fi.argsort(stable=True)
        # counts via scatter_add (torch.bincount device-syncs, which
        # breaks hipGraph capture of quantized-MoE decode)
        counts = torch.zeros(E, device=dev, dtype=torch.long)
        counts.scatter_add_(0, fi, torch.ones_like(fi))
        seg = torch.zeros(E + 1, device=dev, dtype=torch.int32)
        seg[1:] = counts.cumsum(0)
        pair_tok = (order // k).to(torch.int32)
        Ie = self.experts[0].inter
        gu = C.moe_gemm(flat.contiguous(), pair_tok, seg, gu_p, gu_s,
                        2 * Ie, fp8)
        act = ops.swiglu_fused(gu)
        y_pairs = C.moe_gemm(act, None, seg, down_p, down_s, H, fp8)
        y_pairs = y_pairs * rw.reshape(-1)[order].unsqueeze(1)
        inv = torch.empty_like(order)
        inv[order] = torch.arange(P, device=dev, dtype=order.dtype)
        return y_pairs[inv].view(T, k, H).sum(1)

    def forward(self, x):
        B, S, H = x.shape
        flat = x.reshape(-1, H)
        weights, idx = ops.moe_topk_router(self.gate(flat), self.top_k)
        weights = weights.to(x.dtype)
        if (flat.is_cuda and flat.dtype == torch.bfloat16
                and not torch.is_grad_enabled() and ops.extension_loaded()):
            kind = self._fused_kind()
            if kind is not None:
                out = self._fused_forward(flat, weights, idx,
                                          kind == "fp8")
                return out.reshape(B, S, H)
        out = torch.zeros_like(flat)
        for e in range(self.num_experts):
            mask = (idx == e)
            tok, slot = mask.nonzero(as_tuple=True)
            if tok.numel() == 0:
                continue
            out.index_add_(0, tok,
                           self.experts[e](flat[tok]) * weights[tok, slot, None])
        return out.reshape(B, S, H)


class DecoderLayer(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        self.config = config
        self.input_layernorm = nn.Parameter(torch.ones(config.hidden_size))
        self.post_attention_layernorm = nn.Parameter(
            torch.ones(config.hidden_size))
        self.self_attn = Attention(config)
        self.mlp = MoEMLP(config) if config.is_moe else MLP(config)

    def forward(self, hidden, positions, kv_cache=None, layer_idx=0,
                training=False):
        eps = self.config.rms_norm_eps
        w_in = self.input_layernorm.to(hidden.dtype)
        w_post = self.post_attention_layernorm.to(hidden.dtype)
        if training or not hidden.is_cuda:
            h = ops.rmsnorm(hidden, w_in, eps)
            hidden = hidden + self.self_attn(h, positions, kv_cache,
                                             layer_idx, training)
            h = ops.rmsnorm(hidden, w_post, eps)
            hidden = hidden + self.mlp(h)
            return hidden
        # inference: fused residual + norm (one HBM round trip)
        h = ops.rmsnorm(hidden, w_in, eps)
        attn_out = self.self_attn(h, positions, kv_cache, layer_idx, False)
        h, hidden = ops.rmsnorm_residual(attn_out, hidden, w_post, eps)
        return hidden + self.mlp(h)


class StageModel(nn.Module):
    """A contiguous slice of the decoder living on one rank."""

    def __init__(self, config: ModelConfig, layer_start: int, layer_end: int,
                 has_embedding: bool, has_head: bool):
        super().__init__()
        self.config = config
        self.layer_start = layer_start
        self.layer_end = layer_end
        self.has_embedding = has_embedding
        self.has_head = has_head
        if has_embedding:
            self.embed_tokens = nn.Embedding(config.vocab_size,
                                             config.hidden_size)
        self.layers = nn.ModuleList(
            [DecoderLayer(config) for _ in range(layer_end - layer_start)])
        if has_head:
            self.norm = nn.Parameter(torch.ones(config.hidden_size))
            # Tied embeddings share storage only when the embedding lives on
            # this same stage; with PP>1 the last stage materializes its own
            # copy of the tied weight (the reference instead pins tied
            # modules to the host — graphing.py:532-537).
            if not (config.tie_word_embeddings and has_embedding):
                self.lm_head = TLLinear(config.hidden_size,
                                        config.vocab_size, bias=False)

    @property
    def num_layers(self):
        return len(self.layers)

    def make_kv_cache(self, batch: int, max_seq: int, device, dtype=None,
                      kv_mode: str = "contiguous"):
        if dtype is None:
            dtype = next(self.parameters()).dtype
        if kv_mode == "paged":
            from tensorlink_amd.models.paged import PagedKVCache
            return PagedKVCache(self.num_layers, batch, max_seq,
                                self.config, device, dtype)
        return KVCache(self.num_layers, batch, max_seq, self.config, device,
                       dtype)

    def embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        w = self.embed_tokens.weight
        return self.embed_tokens(input_ids).to(w.dtype)

    def head(self, hidden: torch.Tensor) -> torch.Tensor:
        h = ops.rmsnorm(hidden, self.norm.to(hidden.dtype),
                        self.config.rms_norm_eps)
        if self.config.tie_word_embeddings and self.has_embedding:
            return h @ self.embed_tokens.weight.t()
        return self.lm_head(h)

    def forward(self, hidden_or_ids, positions, kv_cache: Optional[KVCache]
                = None, training: bool = False,
                return_logits: bool = True) -> torch.Tensor:
        if self.has_embedding and hidden_or_ids.dtype in (torch.int32,
                                                          torch.int64):
            hidden = self.embed(hidden_or_ids)
        else:
            hidden = hidden_or_ids
        ckpt = (training and getattr(self, "grad_checkpointing", False)
                and torch.is_grad_enabled())
        for i, layer in enumerate(self.layers):
            if ckpt:
                # recompute the layer in backward instead of stashing
                # activations (no dropout anywhere -> exact recompute)
                hidden = torch.utils.checkpoint.checkpoint(
                    layer, hidden, positions, kv_cache, i, training,
                    use_reentrant=False)
            else:
                hidden = layer(hidden, positions, kv_cache, i, training)
        if kv_cache is not None:
            kv_cache.advance(hidden.shape[1])
        if self.has_head and return_logits:
            return self.head(hidden)
        return hidden


def build_full_model(config: ModelConfig):
    """Whole model as a single stage (PP=1 / whole-model offload — the
    reference's ``entire_model`` path, ``ml/module.py:894-897``)."""
    cls = _stage_class(config)
    return cls(config, 0, config.num_hidden_layers, True, True)


def build_stage(config: ModelConfig, spec):
    cls = _stage_class(config)
    return cls(config, spec.layer_start, spec.layer_end,
               spec.has_embedding, spec.has_head)


def _stage_class(config: ModelConfig):
    if config.architecture == "gpt2":
        from tensorlink_amd.models.gpt2 import Gpt2StageModel
        return Gpt2StageModel
    if config.architecture == "neox":
        from tensorlink_amd.models.neox import NeoxStageModel
        return NeoxStageModel
    return StageModel
