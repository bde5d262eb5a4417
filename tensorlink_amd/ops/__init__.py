"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, PyTorch reference on CPU.

Policy (required by the build contract): on a GPU box the HIP extension MUST
be loadable — a missing extension raises instead of silently falling back to
eager PyTorch. On CPU-only hosts (the test tier) the fp32 reference
implementations in :mod:`tensorlink_amd.ops.reference` run instead.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from tensorlink_amd.ops import reference as ref

_C = None
_IMPORT_ERROR: Optional[BaseException] = None
try:
    from tensorlink_amd import _C  # type: ignore  # built by setup.py
except Exception as e:  # pragma: no cover - absent on CPU CI before build
    _IMPORT_ERROR = e


def extension_loaded() -> bool:
    return _C is not None


def _require_ext():
    if _C is None:
        raise RuntimeError(
            "tensorlink_amd._C HIP extension is not built but a CUDA/HIP "
            "tensor reached the op layer. Build it in-tree with "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            f"(import error: {_IMPORT_ERROR!r})")
    return _C


def _on_gpu(*tensors: torch.Tensor) -> bool:
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


# ---------------------------------------------------------------------------
# RMSNorm (autograd-aware)
# ---------------------------------------------------------------------------
class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        C = _require_ext()
        need_grad = x.requires_grad or weight.requires_grad
        outs = C.rmsnorm_fwd(x.contiguous(), None, weight.contiguous(), eps,
                             need_grad)
        y = outs[0]
        if need_grad:
            ctx.save_for_backward(x, weight, outs[1])
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _require_ext()
        x, weight, rstd = ctx.saved_tensors
        dx, dw = C.rmsnorm_bwd(dy.contiguous(), x.contiguous(),
                               weight.contiguous(), rstd)
        return dx, dw, None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6):
    if not _on_gpu(x):
        return ref.rmsnorm(x, weight, eps)
    if torch.is_grad_enabled() and (x.requires_grad or weight.requires_grad):
        return _RMSNormFn.apply(x, weight, eps)
    return _require_ext().rmsnorm_fwd(x.contiguous(), None,
                                      weight.contiguous(), eps, False)[0]


def rmsnorm_residual(x: torch.Tensor, residual: torch.Tensor,
                     weight: torch.Tensor, eps: float = 1e-6):
    """Fused residual add + RMSNorm (inference path). Returns (y, r)."""
    if not _on_gpu(x):
        return ref.rmsnorm_residual(x, residual, weight, eps)
    outs = _require_ext().rmsnorm_fwd(x.contiguous(), residual.contiguous(),
                                      weight.contiguous(), eps, False)
    return outs[0], outs[1]


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------
def apply_rope_(q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
                inv_freq: torch.Tensor, sign: float = 1.0) -> None:
    """In-place RoPE on q [T,Hq,D] / k [T,Hkv,D] with positions [T].

    Inference path (no autograd). sign=-1 applies the inverse rotation.
    """
    if not _on_gpu(q):
        # compute cos/sin from the PASSED inv_freq (NOT a default theta:
        # Qwen uses rope_theta=1e6, Llama-3.1 a scaled spectrum)
        freqs = positions.float()[:, None] * inv_freq.float()[None, :]
        cos, sin = freqs.cos(), freqs.sin()
        # reference expects [B,S,H,D]
        qq, kk = ref.apply_rope(q.unsqueeze(0), k.unsqueeze(0), cos,
                                sin * sign)
        q.copy_(qq[0])
        k.copy_(kk[0])
        return
    _require_ext().rope_(q, k, positions.int(), inv_freq.float(), sign)


def rope_append_(qkv: torch.Tensor, k_cache: torch.Tensor,
                 v_cache: torch.Tensor, positions: torch.Tensor,
                 inv_freq: torch.Tensor, S: int, Hq: int, Hkv: int,
                 block_table: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Fused RoPE + KV-cache append from the fused-QKV projection output.

    qkv [T, (Hq+2*Hkv)*D] with T = B*S (q | k | v per row); caches
    [B,Hkv,Smax,D] (or paged pools [n_pages,Hkv,128,D] with block_table
    [B, pages]); positions [T] gives the RoPE angle and cache slot.
    Returns the rotated q as a contiguous [T, Hq, D] tensor.
    """
    if _on_gpu(qkv):
        return _require_ext().rope_append_(
            qkv.contiguous(), k_cache, v_cache, positions.int(),
            inv_freq.float(), S, Hq, Hkv,
            block_table.int() if block_table is not None else None)
    D = k_cache.shape[-1]
    T = qkv.numel() // qkv.shape[-1]
    q = qkv[..., :Hq * D].reshape(T, Hq, D).contiguous()
    k = qkv[..., Hq * D:(Hq + Hkv) * D].reshape(T, Hkv, D).contiguous()
    v = qkv[..., (Hq + Hkv) * D:].reshape(T, Hkv, D).contiguous()
    apply_rope_(q, k, positions, inv_freq)
    B = T // S
    pos = positions.view(B, S).long()
    kB = k.view(B, S, Hkv, D)
    vB = v.view(B, S, Hkv, D)
    if block_table is None:
        for b in range(B):
            k_cache[b, :, pos[b]] = kB[b].transpose(0, 1)
            v_cache[b, :, pos[b]] = vB[b].transpose(0, 1)
    else:
        for b in range(B):
            for s_i in range(S):
                p = int(pos[b, s_i])
                page = int(block_table[b, p // 128])
                k_cache[page, :, p % 128] = kB[b, s_i]
                v_cache[page, :, p % 128] = vB[b, s_i]
    return q


class _RopeFn(torch.autograd.Function):
    """Autograd RoPE for the training path (operates out-of-place)."""

    @staticmethod
    def forward(ctx, q, k, positions, inv_freq):
        q2, k2 = q.contiguous().clone(), k.contiguous().clone()
        apply_rope_(q2, k2, positions, inv_freq, 1.0)
        ctx.save_for_backward(positions, inv_freq)
        return q2, k2

    @staticmethod
    def backward(ctx, dq, dk):
        positions, inv_freq = ctx.saved_tensors
        dq2, dk2 = dq.contiguous().clone(), dk.contiguous().clone()
        apply_rope_(dq2, dk2, positions, inv_freq, -1.0)
        return dq2, dk2, None, None


def apply_rope(q, k, positions, inv_freq):
    return _RopeFn.apply(q, k, positions, inv_freq)


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------
class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        C = _require_ext()
        ctx.save_for_backward(gate, up)
        return C.swiglu_fwd(gate.contiguous(), up.contiguous())

    @staticmethod
    def backward(ctx, dout):
        C = _require_ext()
        gate, up = ctx.saved_tensors
        dg, du = C.swiglu_bwd(dout.contiguous(), gate.contiguous(),
                              up.contiguous())
        return dg, du


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if not _on_gpu(gate):
        return ref.swiglu(gate, up)
    if torch.is_grad_enabled() and (gate.requires_grad or up.requires_grad):
        return _SwiGLUFn.apply(gate, up)
    return _require_ext().swiglu_fwd(gate.contiguous(), up.contiguous())


def swiglu_fused(gate_up: torch.Tensor) -> torch.Tensor:
    """silu(gu[..., :I]) * gu[..., I:] on the fused gate_up GEMM output
    (inference path — avoids two .contiguous() splits)."""
    if not _on_gpu(gate_up):
        I = gate_up.shape[-1] // 2
        return ref.swiglu(gate_up[..., :I], gate_up[..., I:])
    return _require_ext().swiglu_fused(gate_up.contiguous())


# ---------------------------------------------------------------------------
# Attention
# ---------------------------------------------------------------------------
def attention_prefill(q, k, v, causal: bool = True,
                      scale: Optional[float] = None,
                      q_off: int = 0) -> torch.Tensor:
    """q [B,Sq,Hq,D], k/v [B,Skv,Hkv,D] -> [B,Sq,Hq,D]. Inference path.
    q_off: query row i is at global position q_off+i (chunked prefill)."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if not _on_gpu(q):
        return ref.attention_prefill(q, k, v, causal, scale, q_off)
    if (q.shape[1] >= 4096 and q_off == 0 and causal
            and q.shape[1] == k.shape[1]):
        # long-context full prefill: ROCm flash SDPA measured faster
        # than the hand-written kernel at S >= 4k (376 vs 259 TF,
        # profiles/ r1); chunked/q_off windows stay on the HIP kernel
        out = torch.nn.functional.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, scale=scale, enable_gqa=True)
        return out.transpose(1, 2)
    return _require_ext().prefill_attn(q.contiguous(), k.contiguous(),
                                       v.contiguous(), scale, causal,
                                       q_off)


class _FlashAttnFn(torch.autograd.Function):
    """Hand-written flash attention for training (gfx950): forward =
    prefill kernel saving the logsumexp; backward = the two MFMA kernels
    in ops/csrc/flash_bwd.hip (dKV over key tiles, dQ over query tiles)
    with Delta = rowsum(dO*O) precomputed in fp32."""

    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        C = _require_ext()
        out, lse = C.prefill_attn_lse(q.contiguous(), k.contiguous(),
                                      v.contiguous(), scale, causal)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.causal = causal
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        C = _require_ext()
        q, k, v, out, lse = ctx.saved_tensors
        delta = (dout.float() * out.float()).sum(-1)          # [B,S,Hq]
        delta = delta.permute(0, 2, 1).contiguous()           # [B,Hq,S]
        dq, dk, dv = C.attn_bwd(q.contiguous(), k.contiguous(),
                                v.contiguous(), dout.contiguous(), lse,
                                delta, ctx.scale, ctx.causal)
        return dq, dk, dv, None, None


def attention_train(q, k, v, causal: bool = True,
                    scale: Optional[float] = None) -> torch.Tensor:
    """Training attention with autograd. GPU bf16 runs the hand-written
    flash forward+backward kernels; CPU (and unsupported head dims)
    composes torch SDPA."""
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if (q.is_cuda and q.dtype == torch.bfloat16 and _C is not None
            and D in (64, 128) and k.shape[1] == S):
        return _FlashAttnFn.apply(q, k, v, causal, scale)
    qt = q.transpose(1, 2)
    kt = k.transpose(1, 2)
    vt = v.transpose(1, 2)
    try:
        out = torch.nn.functional.scaled_dot_product_attention(
            qt, kt, vt, is_causal=causal, scale=scale, enable_gqa=True)
    except (TypeError, RuntimeError):
        # older torch / unsupported backend: materialize repeated heads
        rep = Hq // Hkv
        out = torch.nn.functional.scaled_dot_product_attention(
            qt, kt.repeat_interleave(rep, dim=1),
            vt.repeat_interleave(rep, dim=1), is_causal=causal, scale=scale)
    return out.transpose(1, 2)


def attention_decode(q, k_cache, v_cache, seq_lens,
                     scale: Optional[float] = None, n_split: int = 0,
                     block_table: Optional[torch.Tensor] = None
                     ) -> torch.Tensor:
    """q [B,1,Hq,D] or [B,Hq,D]; caches [B,Hkv,Smax,D] (or paged pools +
    block_table); seq_lens [B]. n_split: flash-decode splits (0 = auto)."""
    squeeze = q.dim() == 4
    if squeeze:
        q3 = q.squeeze(1)
    else:
        q3 = q
    scale = scale if scale is not None else 1.0 / math.sqrt(q3.shape[-1])
    if not _on_gpu(q3):
        if block_table is not None:
            # gather pages -> contiguous [B, L, Hkv, D] for the reference
            B = q3.shape[0]
            L = int(seq_lens.max())
            np_ = (L + 127) // 128
            idx = block_table[:, :np_].long()
            kc = k_cache[idx].permute(0, 1, 3, 2, 4)
            vc = v_cache[idx].permute(0, 1, 3, 2, 4)
            kc = kc.reshape(B, np_ * 128, *kc.shape[3:])[:, :L]
            vc = vc.reshape(B, np_ * 128, *vc.shape[3:])[:, :L]
            out = ref.attention_decode(q3.unsqueeze(1), kc, vc, seq_lens,
                                       scale)
        else:
            out = ref.attention_decode(q3.unsqueeze(1),
                                       k_cache.permute(0, 2, 1, 3),
                                       v_cache.permute(0, 2, 1, 3),
                                       seq_lens, scale)
        return out if squeeze else out.squeeze(1)
    out = _require_ext().decode_attn(
        q3.contiguous(), k_cache, v_cache, seq_lens.int(), scale, n_split,
        block_table.int() if block_table is not None else None)
    return out.unsqueeze(1) if squeeze else out


# ---------------------------------------------------------------------------
# Linear (hand-written serving GEMM family)
# ---------------------------------------------------------------------------
import os as _os

# Inference GEMMs at M <= this route to the deterministic MFMA family
# (streaming kernel at M<=64, tiled all-glds kernel above). The family
# guarantees bitwise M-INDEPENDENT per-row results (same K chunk order,
# same (N,K)-only split policy), which is what makes chunked prefill /
# speculative verify / ragged batch decode / plain generate emit
# identical greedy tokens. Above the threshold (large prefill) hipBLASLt
# wins on throughput and exact cross-path equality is not claimed.
GEMM_M_MAX = int(_os.environ.get("TL_GEMM_M_MAX", "512"))

# incremented whenever the hand-written family handles a linear — GPU
# tests assert on it so a silent hipBLASLt fallback cannot pass as
# kernel coverage
gemm_dispatch_count = 0


# (N,K)-pure shape classes; TL_GEMM_LIB_CLASSES routes whole classes to
# hipBLASLt for in-pipeline A/B (a class flips for ALL M at once, so the
# bitwise M-independence guarantee is preserved per process)
# default: the gateup class (16k <= N < 64k) routes to hipBLASLt — the
# BN=64/256 custom structures both lose ~40 us there in-pipeline
# (profiles/gemm_family_ab.md); all other classes are custom, keeping
# bitwise M-independence by construction where the tests assert it
_LIB_CLASSES = frozenset(
    c for c in _os.environ.get("TL_GEMM_LIB_CLASSES",
                               "gateup").split(",") if c)


def _gemm_class(N: int, K: int) -> str:
    if N >= 65536:
        return "wide"
    if N >= 16384:
        return "gateup"
    if K >= 8192:
        return "deepk"
    return "narrow"


def _use_tl_gemm(M: int, N: int, K: int) -> bool:
    return (K % 32 == 0 and N % 64 == 0 and M <= GEMM_M_MAX
            and _gemm_class(N, K) not in _LIB_CLASSES)


def linear(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """x @ W^T + bias. Serving-M shapes route to the hand-written MFMA
    family (skinny_gemm.hip / gemm_tiled.hip); large prefill M and
    training go to torch/hipBLASLt."""
    global gemm_dispatch_count
    if (x.is_cuda and not torch.is_grad_enabled()
            and x.dtype == torch.bfloat16
            and not _os.environ.get("TL_NO_SKINNY")):
        K = x.shape[-1]
        M = x.numel() // K
        N = weight.shape[0]
        if _use_tl_gemm(M, N, K):
            gemm_dispatch_count += 1
            return _require_ext().skinny_gemm(
                x.contiguous(), weight,
                bias.to(torch.bfloat16) if bias is not None else None)
    return torch.nn.functional.linear(x, weight, bias)


# ---------------------------------------------------------------------------
# AdamW
# ---------------------------------------------------------------------------
def adamw_(param: torch.Tensor, grad: torch.Tensor, exp_avg: torch.Tensor,
           exp_avg_sq: torch.Tensor, *, lr: float, beta1: float = 0.9,
           beta2: float = 0.999, eps: float = 1e-8,
           weight_decay: float = 0.0, step: int = 1) -> None:
    if not _on_gpu(param):
        ref.adamw_step(param, grad, exp_avg, exp_avg_sq, lr=lr, beta1=beta1,
                       beta2=beta2, eps=eps, weight_decay=weight_decay,
                       step=step)
        return
    _require_ext().adamw_(param, grad.contiguous(), exp_avg, exp_avg_sq, lr,
                          beta1, beta2, eps, weight_decay, step)


# ---------------------------------------------------------------------------
# Sampling
# ---------------------------------------------------------------------------
def _splitmix64(x: int) -> int:
    x = (x + 0x9E3779B97F4A7C15) & (2 ** 64 - 1)
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & (2 ** 64 - 1)
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & (2 ** 64 - 1)
    return x ^ (x >> 31)


def request_seed(seed: int, step: int) -> int:
    """Per-draw seed for a seeded request: depends only on (seed, step),
    so a request's samples are reproducible wherever it is batched.
    Returned as a signed 63-bit value (torch.long seeds tensor)."""
    return _splitmix64((seed & (2 ** 63 - 1)) ^ (step * 0xA24BAED4963EE407)) \
        & (2 ** 63 - 1)


def sample_tokens(logits: torch.Tensor, *, temps, top_ps, top_ks, pres,
                  freqs, counts: Optional[torch.Tensor] = None,
                  seeds: Optional[torch.Tensor] = None,
                  counter: Optional[torch.Tensor] = None,
                  seed_base: int = 0) -> torch.Tensor:
    """Batched fused sampling (ops/csrc/sampling.hip): penalties +
    temperature + top-k + top-p + draw, one kernel, graph-capture-safe.
    logits [B, V] bf16 on GPU; per-row param tensors; counts [B, V] i32
    is read for penalties and the chosen token's count is incremented."""
    C = _require_ext()
    return C.sample_tokens(logits.contiguous(), temps, top_ps, top_ks,
                           pres, freqs, counts, seeds, counter, seed_base)


def sample_token(logits, *, temperature=1.0, top_p=1.0, top_k=0,
                 generator=None, token_counts=None,
                 presence_penalty=0.0, frequency_penalty=0.0):
    """Single-request sampling with the reference (torch) semantics.
    The serving engine's hot path uses :func:`sample_tokens`; this stays
    torch-composed for API parity (dict token_counts, torch.Generator)
    and for greedy argmax."""
    return ref.sample_token(logits, temperature=temperature, top_p=top_p,
                            top_k=top_k, generator=generator,
                            token_counts=token_counts,
                            presence_penalty=presence_penalty,
                            frequency_penalty=frequency_penalty)


moe_topk_router = ref.moe_topk_router
causal_lm_loss = ref.causal_lm_loss
rope_cos_sin = ref.rope_cos_sin


chunked_causal_lm_loss = ref.chunked_causal_lm_loss
