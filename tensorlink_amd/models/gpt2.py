"""GPT-2 model family (LayerNorm + learned positions + GELU MLP).

Covers the reference's CI models (sshleifer/tiny-gpt2 — reference
``tests/test_distributed_model.py:24`` — and the "GPT-2-small
DistributedModel on 2 local CPU worker procs" plumbing config,
BASELINE.json #1). Same stage interface as the llama-family
:class:`StageModel` so the pipeline runtime drives either.

Attention uses the shared CDNA4 kernels on GPU (head_dim 64);
LayerNorm/GELU are torch-native (GPT-2 is the plumbing family, not the
flagship perf path).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from tensorlink_amd import ops
from tensorlink_amd.models.configs import ModelConfig
from tensorlink_amd.models.dense import KVCache


def gpt2_config(name: str = "gpt2-small") -> ModelConfig:
    sizes = {
        "gpt2-small": (768, 12, 12),
        "gpt2-medium": (1024, 24, 16),
        "tiny-gpt2": (64, 2, 2),     # sshleifer/tiny-gpt2 scale
    }
    h, layers, heads = sizes[name]
    return ModelConfig(
        name=name, vocab_size=50257, hidden_size=h,
        intermediate_size=4 * h, num_hidden_layers=layers,
        num_attention_heads=heads, num_key_value_heads=heads,
        max_position_embeddings=1024, tie_word_embeddings=True,
        architecture="gpt2", rms_norm_eps=1e-5)


class Gpt2Attention(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        h = config.hidden_size
        self.c_attn = nn.Linear(h, 3 * h)
        self.c_proj = nn.Linear(h, h)
        self.n_heads = config.num_attention_heads
        self.head_dim = h // self.n_heads
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, x, kv_cache: Optional[KVCache], layer_idx: int,
                positions, training: bool):
        B, S, H = x.shape
        qkv = self.c_attn(x)
        q, k, v = qkv.split(H, dim=-1)
        q = q.view(B, S, self.n_heads, self.head_dim)
        k = k.view(B, S, self.n_heads, self.head_dim)
        v = v.view(B, S, self.n_heads, self.head_dim)
        if training:
            out = ops.attention_train(q, k, v, causal=True, scale=self.scale)
        elif kv_cache is None:
            out = ops.attention_prefill(q, k, v, causal=True,
                                        scale=self.scale)
        else:
            kv_cache.append(layer_idx, k, v, positions)
            table = getattr(kv_cache, "table", None)
            if S == 1:
                out = ops.attention_decode(
                    q, kv_cache.k[layer_idx], kv_cache.v[layer_idx],
                    kv_cache.seq_lens + 1, scale=self.scale,
                    block_table=table)
            else:
                # prefill with cache; off > 0 = chunked continuation
                # (same contract as models/dense.py Attention)
                off = int(kv_cache.seq_lens.max())
                kv_len = off + S
                if table is None:
                    k_attn = kv_cache.k[layer_idx][:, :, :kv_len].permute(
                        0, 2, 1, 3).contiguous()
                    v_attn = kv_cache.v[layer_idx][:, :, :kv_len].permute(
                        0, 2, 1, 3).contiguous()
                else:
                    k_attn, v_attn = kv_cache.gather_contiguous(layer_idx,
                                                                kv_len)
                out = ops.attention_prefill(q, k_attn, v_attn, causal=True,
                                            scale=self.scale, q_off=off)
        return self.c_proj(out.reshape(B, S, H))


class Gpt2Block(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        h = config.hidden_size
        self.ln_1 = nn.LayerNorm(h, eps=config.rms_norm_eps)
        self.attn = Gpt2Attention(config)
        self.ln_2 = nn.LayerNorm(h, eps=config.rms_norm_eps)
        self.c_fc = nn.Linear(h, config.intermediate_size)
        self.c_proj = nn.Linear(config.intermediate_size, h)

    def forward(self, x, kv_cache, layer_idx, positions, training):
        x = x + self.attn(self.ln_1(x), kv_cache, layer_idx, positions,
                          training)
        x = x + self.c_proj(F.gelu(self.c_fc(self.ln_2(x)),
                                   approximate="tanh"))
        return x


class Gpt2StageModel(nn.Module):
    """GPT-2 stage with the StageModel interface."""

    def __init__(self, config: ModelConfig, layer_start: int, layer_end: int,
                 has_embedding: bool, has_head: bool):
        super().__init__()
        self.config = config
        self.layer_start = layer_start
        self.layer_end = layer_end
        self.has_embedding = has_embedding
        self.has_head = has_head
        if has_embedding:
            self.wte = nn.Embedding(config.vocab_size, config.hidden_size)
            self.wpe = nn.Embedding(config.max_position_embeddings,
                                    config.hidden_size)
        self.layers = nn.ModuleList(
            [Gpt2Block(config) for _ in range(layer_end - layer_start)])
        if has_head:
            self.ln_f = nn.LayerNorm(config.hidden_size,
                                     eps=config.rms_norm_eps)
            if not (config.tie_word_embeddings and has_embedding):
                self.lm_head = nn.Linear(config.hidden_size,
                                         config.vocab_size, bias=False)

    @property
    def num_layers(self):
        return len(self.layers)

    def make_kv_cache(self, batch, max_seq, device, dtype=None,
                      kv_mode: str = "contiguous") -> KVCache:
        if dtype is None:
            dtype = next(self.parameters()).dtype
        # GPT-2 (plumbing family) always uses the contiguous cache
        return KVCache(self.num_layers, batch, max_seq, self.config, device,
                       dtype)

    def head(self, hidden):
        h = self.ln_f(hidden)
        if self.config.tie_word_embeddings and self.has_embedding:
            return h @ self.wte.weight.t()
        return self.lm_head(h)

    def forward(self, hidden_or_ids, positions, kv_cache=None,
                training=False, return_logits=True):
        if self.has_embedding and hidden_or_ids.dtype in (torch.int32,
                                                          torch.int64):
            hidden = self.wte(hidden_or_ids) + self.wpe(positions.long())
        else:
            hidden = hidden_or_ids
        for i, layer in enumerate(self.layers):
            hidden = layer(hidden, kv_cache, i, positions, training)
        if kv_cache is not None:
            kv_cache.advance(hidden.shape[1])
        if self.has_head and return_logits:
            return self.head(hidden)
        return hidden


def load_gpt2_hf_weights(stage: Gpt2StageModel, ckpt_dir: str,
                         dtype=torch.float32) -> int:
    """Map HF GPT-2 checkpoint keys (transformer.h.N.*, Conv1D transposed
    weights) onto the native stage."""
    from safetensors import safe_open
    import os
    path = os.path.join(ckpt_dir, "model.safetensors")
    params = dict(stage.named_parameters())
    loaded = 0
    conv1d = ("c_attn.weight", "c_proj.weight", "c_fc.weight")
    with safe_open(path, framework="pt", device="cpu") as f:
        for key in f.keys():
            k = key[len("transformer."):] if key.startswith("transformer.") \
                else key
            native = None
            if k == "wte.weight" and stage.has_embedding:
                native = "wte.weight"
            elif k == "wpe.weight" and stage.has_embedding:
                native = "wpe.weight"
            elif k.startswith("ln_f.") and stage.has_head:
                native = k
            elif k.startswith("h."):
                _, idx, rest = k.split(".", 2)
                idx = int(idx)
                if stage.layer_start <= idx < stage.layer_end:
                    local = idx - stage.layer_start
                    rest = rest.replace("mlp.c_fc", "c_fc").replace(
                        "mlp.c_proj", "c_proj")
                    native = f"layers.{local}.{rest}"
            if native and native in params:
                t = f.get_tensor(key).to(dtype)
                if any(k.endswith(c) for c in conv1d):
                    t = t.t().contiguous()  # HF Conv1D stores [in, out]
                with torch.no_grad():
                    params[native].copy_(t)
                loaded += 1
    return loaded
