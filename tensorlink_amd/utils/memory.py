"""Memory estimation driving pipeline-stage planning.

Re-derivation of the reference's memory model (``tensorlink/ml/utils.py:36-124``):
params + grads + optimizer state + activations + KV cache, with an overhead
factor — but computed analytically from :class:`ModelConfig` instead of
walking an instantiated meta-device module, and calibrated for MI355X
(288 GB HBM3E per GPU, bf16 compute, fp32 Adam moments).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

DTYPE_BYTES = {
    "float32": 4, "fp32": 4, "bfloat16": 2, "bf16": 2, "float16": 2,
    "fp16": 2, "fp8": 1, "float8_e4m3fn": 1,
}

# The reference applies a flat ×1.20 overhead (ml/utils.py:121). We keep a
# smaller fudge because activations are accounted explicitly and the HIP
# allocator is ours; plus a fixed per-rank reserve for RCCL buffers,
# workspace and fragmentation.
OVERHEAD_FACTOR = 1.10
FIXED_RESERVE_BYTES = 4 << 30

MI355X_HBM_BYTES = 288 << 30


@dataclass
class MemoryEstimate:
    params: int
    grads: int
    optimizer: int
    activations: int
    kv_cache: int

    @property
    def total(self) -> int:
        raw = (self.params + self.grads + self.optimizer
               + self.activations + self.kv_cache)
        return int(raw * OVERHEAD_FACTOR)


def layer_param_bytes(config, dtype_bytes: int) -> int:
    """Weight bytes of one decoder layer."""
    h = config.hidden_size
    attn = h * config.q_size + 2 * h * config.kv_size + config.q_size * h
    if config.qkv_bias:
        attn += config.q_size + 2 * config.kv_size
    if config.is_moe:
        mlp = (config.num_local_experts * 3 * h
               * config.expert_intermediate_size)
        mlp += h * config.num_local_experts
    else:
        gated = getattr(config, "gated_mlp", True)
        mlp = (3 if gated else 2) * h * config.intermediate_size
    return (attn + mlp + 2 * h) * dtype_bytes


def embedding_param_bytes(config, dtype_bytes: int) -> int:
    return config.vocab_size * config.hidden_size * dtype_bytes


def head_param_bytes(config, dtype_bytes: int) -> int:
    # final norm + lm_head (zero extra weight bytes when tied)
    n = config.hidden_size
    if not config.tie_word_embeddings:
        n += config.vocab_size * config.hidden_size
    return n * dtype_bytes


def activation_bytes_per_layer(config, batch: int, seq: int,
                               dtype_bytes: int, training: bool) -> int:
    """Activation working set per decoder layer.

    Reference model: B*S*H*dtype × 4 (eval) / × 7 (train)
    (``ml/utils.py:95-99``). Training stores residual streams + attn/MLP
    intermediates for backward; eval only needs transient buffers that the
    caching allocator reuses across layers, so eval activations are NOT
    multiplied by layer count by the caller.
    """
    base = batch * seq * config.hidden_size * dtype_bytes
    if training:
        # saved tensors per layer: 2 norms, qkv, attn out, gate/up (I/H ratio),
        # softmax stats — ≈ (6 + 2*I/H) hidden-sized tensors
        ratio = config.intermediate_size / config.hidden_size
        return int(base * (6 + 2 * ratio))
    return base * 4


def kv_cache_bytes_per_layer(config, batch: int, seq: int, dtype_bytes: int) -> int:
    # B*S*kv_heads*head_dim*2*dtype (reference ml/utils.py:110-118)
    return batch * seq * config.num_key_value_heads * config.head_dim * 2 * dtype_bytes


def estimate_memory(config, *, batch_size: int = 1, seq_len: int = 4096,
                    training: bool = False, optimizer: str = "adamw",
                    dtype: str = "bfloat16",
                    n_layers: int | None = None,
                    include_embedding: bool = False,
                    include_head: bool = False) -> MemoryEstimate:
    """Estimate bytes for a contiguous slice of the model on one rank."""
    db = DTYPE_BYTES[dtype]
    nl = config.num_hidden_layers if n_layers is None else n_layers
    params = nl * layer_param_bytes(config, db)
    if include_embedding:
        params += embedding_param_bytes(config, db)
    if include_head:
        params += head_param_bytes(config, db)
    grads = params if training else 0
    # AdamW: exp_avg + exp_avg_sq in fp32 (reference: 2*params*(4/dtype) —
    # ml/utils.py:77 — same formula)
    opt = 2 * params * (4 // db if db <= 4 else 1) if training and optimizer else 0
    if training:
        act = nl * activation_bytes_per_layer(config, batch_size, seq_len, db, True)
    else:
        act = activation_bytes_per_layer(config, batch_size, seq_len, db, False)
    kv = 0 if training else nl * kv_cache_bytes_per_layer(config, batch_size, seq_len, db)
    return MemoryEstimate(params, grads, opt, act, kv)


def get_gpu_memory(device: int | None = None) -> int:
    """Free device bytes; host RAM as CPU fallback like the reference
    (``ml/utils.py:127-149``)."""
    if torch.cuda.is_available():
        free, _total = torch.cuda.mem_get_info(device)
        return free
    try:
        import psutil
        return psutil.virtual_memory().available
    except Exception:
        return 16 << 30
