"""Pure-PyTorch reference implementations of every custom op.

These serve three purposes:
1. numerics baseline for the HIP kernel unit tests (fp32 reference, cf. the
   gradient-hash validation idea in reference ``tensorlink/ml/proofs.py:6``);
2. CPU execution path for the no-GPU test tier (the reference's CI is also
   CPU-only — ``.github/workflows/pytest.yaml``);
3. documentation of each kernel's contract.

They are NOT used on a GPU box: :mod:`tensorlink_amd.ops` raises if the HIP
extension is missing there.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


# --------------------------------------------------------------------------
# RMSNorm
# --------------------------------------------------------------------------
def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """y = x / sqrt(mean(x^2) + eps) * weight, stats in fp32."""
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps)
    return (y * weight.float()).to(dtype)


def rmsnorm_residual(x: torch.Tensor, residual: torch.Tensor,
                     weight: torch.Tensor, eps: float = 1e-6
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fused residual-add + RMSNorm: r = x + residual; y = rmsnorm(r).

    Returns (y, r). The fused form is what the HIP kernel implements — one
    HBM round trip instead of two (the residual stream is the memory-bound
    hot path on MI355X).
    """
    r = (x.float() + residual.float())
    var = r.pow(2).mean(-1, keepdim=True)
    y = r * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype), r.to(x.dtype)


# --------------------------------------------------------------------------
# RoPE
# --------------------------------------------------------------------------
def rope_cos_sin(head_dim: int, positions: torch.Tensor, theta: float = 10000.0,
                 device=None, dtype=torch.float32) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [n_pos, head_dim//2], fp32. Computed on device — the
    reference ships HF rotary buffers over the network instead
    (``ml/module.py:1320-1422``)."""
    device = device or positions.device
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, device=device,
                                             dtype=torch.float32) / head_dim))
    freqs = positions.to(device).float()[:, None] * inv_freq[None, :]
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def apply_rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
               sin: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rotate q,k. Shapes: q/k [B, S, H, D] (D even), cos/sin [S, D/2].

    Uses the HF 'rotate_half' convention (first/second half pairing), so HF
    checkpoints produce identical outputs.
    """
    def rot(x):
        d2 = x.shape[-1] // 2
        x1, x2 = x[..., :d2].float(), x[..., d2:].float()
        c = cos.view(1, cos.shape[0], 1, d2).float()
        s = sin.view(1, sin.shape[0], 1, d2).float()
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)
    return rot(q), rot(k)


# --------------------------------------------------------------------------
# SwiGLU
# --------------------------------------------------------------------------
def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up, computed in fp32."""
    g = gate.float()
    return (g * torch.sigmoid(g) * up.float()).to(gate.dtype)


# --------------------------------------------------------------------------
# Attention
# --------------------------------------------------------------------------
def attention_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      causal: bool = True, scale: Optional[float] = None,
                      q_off: int = 0) -> torch.Tensor:
    """Causal GQA attention. q [B,Sq,Hq,D], k/v [B,Skv,Hkv,D] →
    [B,Sq,Hq,D]. q_off: query row i sits at global position q_off+i
    (chunked prefill against Skv = q_off + Sq history+chunk keys)."""
    B, Sq, Hq, D = q.shape
    Skv = k.shape[1]
    Hkv = k.shape[2]
    scale = scale or 1.0 / math.sqrt(D)
    rep = Hq // Hkv
    qf = q.float().transpose(1, 2)                       # [B,Hq,Sq,D]
    kf = k.float().transpose(1, 2).repeat_interleave(rep, dim=1)
    vf = v.float().transpose(1, 2).repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        rows = torch.arange(Sq, device=q.device).unsqueeze(1) + q_off
        cols = torch.arange(Skv, device=q.device).unsqueeze(0)
        scores = scores.masked_fill(cols > rows, float("-inf"))
    p = scores.softmax(-1)
    out = torch.matmul(p, vf)
    return out.transpose(1, 2).to(q.dtype)


def attention_decode(q: torch.Tensor, k_cache: torch.Tensor,
                     v_cache: torch.Tensor, seq_lens: torch.Tensor,
                     scale: Optional[float] = None) -> torch.Tensor:
    """Single-token decode over a contiguous KV cache.

    q [B,1,Hq,D]; k_cache/v_cache [B,Smax,Hkv,D]; seq_lens [B] = valid kv
    length per sequence (incl. the current token already written).
    """
    B, _, Hq, D = q.shape
    Hkv = k_cache.shape[2]
    scale = scale or 1.0 / math.sqrt(D)
    rep = Hq // Hkv
    out = torch.empty_like(q)
    for b in range(B):
        L = int(seq_lens[b])
        kf = k_cache[b, :L].float().repeat_interleave(rep, dim=1)   # [L,Hq,D]
        vf = v_cache[b, :L].float().repeat_interleave(rep, dim=1)
        qf = q[b, 0].float()                                        # [Hq,D]
        scores = torch.einsum("hd,lhd->hl", qf, kf) * scale
        p = scores.softmax(-1)
        out[b, 0] = torch.einsum("hl,lhd->hd", p, vf).to(q.dtype)
    return out


# --------------------------------------------------------------------------
# Sampling
# --------------------------------------------------------------------------
def sample_token(logits: torch.Tensor, *, temperature: float = 1.0,
                 top_p: float = 1.0, top_k: int = 0,
                 generator: Optional[torch.Generator] = None,
                 token_counts: Optional[dict] = None,
                 presence_penalty: float = 0.0,
                 frequency_penalty: float = 0.0) -> torch.Tensor:
    """Sample next tokens from [B, V] logits. temperature<=0 → greedy.

    token_counts (B==1 path): {token_id: count} of tokens generated so
    far; OpenAI-style penalties subtract presence_penalty once per seen
    token plus frequency_penalty * count (the reference only DECLARES
    these fields — ``api/models.py:73-74`` — and never applies them)."""
    if token_counts and (presence_penalty or frequency_penalty):
        logits = logits.float().clone()
        for t, c in token_counts.items():
            logits[..., t] -= presence_penalty + frequency_penalty * c
    if temperature <= 0.0:
        return logits.argmax(-1)
    logits = logits.float() / temperature
    if top_k and top_k > 0 and top_k < logits.shape[-1]:
        kth = logits.topk(top_k, dim=-1).values[..., -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if top_p < 1.0:
        sorted_logits, idx = logits.sort(-1, descending=True)
        probs = sorted_logits.softmax(-1)
        cum = probs.cumsum(-1)
        cut = cum - probs > top_p          # keep tokens until cum mass > p
        sorted_logits = sorted_logits.masked_fill(cut, float("-inf"))
        logits = torch.full_like(logits, float("-inf")).scatter(
            -1, idx, sorted_logits)
    probs = logits.softmax(-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


# --------------------------------------------------------------------------
# Fused AdamW
# --------------------------------------------------------------------------
def adamw_step(param: torch.Tensor, grad: torch.Tensor,
               exp_avg: torch.Tensor, exp_avg_sq: torch.Tensor, *,
               lr: float, beta1: float, beta2: float, eps: float,
               weight_decay: float, step: int) -> None:
    """In-place AdamW matching torch.optim.AdamW semantics.

    param bf16/fp32; moments fp32. Decoupled weight decay."""
    p32 = param.float()
    g32 = grad.float()
    exp_avg.mul_(beta1).add_(g32, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g32, g32, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    p32 = p32 * (1 - lr * weight_decay) - lr * (exp_avg / bc1) / denom
    param.copy_(p32.to(param.dtype))


# --------------------------------------------------------------------------
# Cross-entropy (vocab-parallel ready)
# --------------------------------------------------------------------------
def causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor,
                   ignore_index: int = -100) -> torch.Tensor:
    """Shifted CE loss like HF CausalLM (logits [B,S,V], labels [B,S])."""
    lg = logits[:, :-1].reshape(-1, logits.shape[-1]).float()
    lb = labels[:, 1:].reshape(-1)
    return F.cross_entropy(lg, lb, ignore_index=ignore_index)


# --------------------------------------------------------------------------
# Chunked cross-entropy (memory-bounded loss)
# --------------------------------------------------------------------------
class _ChunkedCELoss(torch.autograd.Function):
    """Shifted causal-LM CE without materializing the full [B*S, V]
    logits: the head GEMM + CE run chunk-by-chunk over tokens, and
    backward recomputes each chunk's logits to produce d_hidden and
    d_weight. Peak extra memory is one [chunk, V] block instead of
    B*S*V*4 bytes (Qwen2.5-7B: 152k vocab x 4 B = 0.6 MB/token — a
    4k-token micro-batch saves ~2.4 GB). Exactly equals
    F.cross_entropy(head(hidden), labels) in fp32.
    """

    @staticmethod
    def forward(ctx, hidden, weight, labels, chunk):
        # hidden [N, H] fp-any; weight [V, H]; labels [N] with -100 pads
        N = hidden.shape[0]
        valid = (labels != -100)
        n_valid = int(valid.sum())
        total = hidden.new_zeros((), dtype=torch.float32)
        for s in range(0, N, chunk):
            e = min(s + chunk, N)
            lg = (hidden[s:e].float() @ weight.float().t())
            total = total + F.cross_entropy(lg, labels[s:e],
                                            ignore_index=-100,
                                            reduction="sum")
        ctx.save_for_backward(hidden, weight, labels)
        ctx.chunk = chunk
        ctx.n_valid = max(n_valid, 1)
        return total / ctx.n_valid

    @staticmethod
    def backward(ctx, dloss):
        hidden, weight, labels = ctx.saved_tensors
        chunk, n_valid = ctx.chunk, ctx.n_valid
        N, H = hidden.shape
        dh = torch.zeros_like(hidden)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        scale = dloss.float() / n_valid
        for s in range(0, N, chunk):
            e = min(s + chunk, N)
            hf = hidden[s:e].float()
            lg = hf @ weight.float().t()
            p = lg.softmax(-1)
            lb = labels[s:e]
            ok = lb != -100
            safe = lb.clamp(min=0)
            p[torch.arange(e - s, device=p.device), safe] -= 1.0
            p = p * ok.unsqueeze(1)
            p = p * scale
            dh[s:e] = (p @ weight.float()).to(hidden.dtype)
            dw += p.t() @ hf
        return dh, dw.to(weight.dtype), None, None


def chunked_causal_lm_loss(hidden: torch.Tensor, head_weight: torch.Tensor,
                           labels: torch.Tensor, chunk: int = 1024,
                           ignore_index: int = -100) -> torch.Tensor:
    """Shifted CE loss from the pre-head hidden states, chunked over
    tokens (see _ChunkedCELoss). hidden [B,S,H], labels [B,S]."""
    B, S, H = hidden.shape
    h = hidden[:, :-1].reshape(-1, H)
    lb = labels[:, 1:].reshape(-1).clone()
    lb[lb == ignore_index] = -100
    return _ChunkedCELoss.apply(h, head_weight, lb, chunk)


# --------------------------------------------------------------------------
# MoE routing
# --------------------------------------------------------------------------
def moe_topk_router(router_logits: torch.Tensor, top_k: int
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Softmax-then-topk router (Mixtral convention).

    router_logits [T, E] → (weights [T,k] normalized, indices [T,k])."""
    probs = router_logits.float().softmax(-1)
    weights, idx = probs.topk(top_k, dim=-1)
    weights = weights / weights.sum(-1, keepdim=True)
    return weights, idx
