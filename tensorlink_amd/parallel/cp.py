"""Context (sequence) parallelism for prefill — ring attention (opt-in,
beyond reference parity; SURVEY.md §2.2 lists SP/CP as a stretch goal).

The sequence is sharded across ranks (weights replicated); each rank
computes attention for its query chunk while K/V chunks rotate around the
ring over xGMI P2P. Causality prunes the ring: chunk j's K/V only visits
ranks i >= j, and partial results merge with the standard online-softmax
(m, l, o) combination, so the result equals single-rank attention.

Scope: inference prefill (logits), decode after a CP prefill (each rank
keeps the K/V of ITS sequence chunk — the cache is sequence-sharded, so
cache memory divides by cp; decode tokens append to the last rank and
every step merges the per-rank partial attentions with one all-gather of
(o, m, l)), and TRAINING: the ring rotation is an autograd Function
(:class:`_RingPass` — backward rotates gradients the opposite way), the
online-softmax partials/merge are plain differentiable torch math with
the row max detached (softmax is shift-invariant, so detaching is
exact), and per-chunk losses + an all-reduce grad sum reproduce the
full-sequence gradient. The kernel-grade version would fuse the merge
into prefill_attn.hip.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist

from tensorlink_amd import ops
from tensorlink_amd.models.configs import ModelConfig, get_config
from tensorlink_amd.models.dense import build_full_model
from tensorlink_amd.models.loader import init_random_stage
from tensorlink_amd.parallel.comm import device_for_rank


def _partial_attn(q, k, v, scale, causal_diag):
    """Unnormalized attention partial of q [B,Sq,H,D] against one K/V
    chunk. Returns (o_unnorm [B,Sq,H,D], m [B,H,Sq], l [B,H,Sq]).
    causal_diag=True applies the in-chunk causal mask (q and k are the
    same chunk); False means the whole chunk precedes q (no mask)."""
    B, Sq, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    qf = q.float().permute(0, 2, 1, 3)
    kf = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    vf = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal_diag:
        Sk = k.shape[1]
        mask = torch.triu(torch.ones(Sq, Sk, dtype=torch.bool,
                                     device=q.device), 1)
        scores = scores.masked_fill(mask, float("-inf"))
    m = scores.amax(-1)                                   # [B,H,Sq]
    p = torch.exp(scores - m.unsqueeze(-1))
    p = torch.nan_to_num(p, nan=0.0)                      # fully-masked rows
    l = p.sum(-1)
    o = torch.matmul(p, vf)                               # unnormalized
    return o.permute(0, 2, 1, 3), m, l


def _merge(acc, nxt):
    """Online-softmax merge of two (o, m, l) partials."""
    if acc is None:
        return nxt
    o1, m1, l1 = acc
    o2, m2, l2 = nxt
    m = torch.maximum(m1, m2)
    a1 = torch.exp(m1 - m)
    a2 = torch.exp(m2 - m)
    o = o1 * a1.permute(0, 2, 1).unsqueeze(-1) + \
        o2 * a2.permute(0, 2, 1).unsqueeze(-1)
    l = l1 * a1 + l2 * a2
    return o, m, l


class CPRunner:
    """Sequence-sharded prefill across a CP group (weights replicated,
    same seeded init on every rank)."""

    def __init__(self, model, rank: int, cp: int, device=None, seed: int = 0):
        self.rank, self.cp = rank, cp
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        dtype = (torch.bfloat16 if self.device.type == "cuda"
                 else torch.float32)
        config = (model if isinstance(model, ModelConfig)
                  else get_config(model))
        self.stage = build_full_model(config)
        init_random_stage(self.stage, device=self.device, dtype=dtype,
                          seed=seed)
        self.stage.eval()
        self.config = config
        self.dtype = dtype

    def _ring_attention(self, attn, x, positions, layer_idx=None):
        """One attention layer with K/V rotating around the CP ring.
        layer_idx is not None => cache this rank's (rotated) k/v chunk for
        subsequent decode."""
        B, Sq, H = x.shape
        qkv = attn.qkv_proj(x)
        q, k, v = attn._split_qkv(qkv, B, Sq)
        if attn.use_qk_norm:
            eps = attn.config.rms_norm_eps
            q = ops.rmsnorm(q.contiguous(), attn.q_norm.to(q.dtype), eps)
            k = ops.rmsnorm(k.contiguous(), attn.k_norm.to(k.dtype), eps)
        q = q.contiguous()
        k = k.contiguous()
        v = v.contiguous()
        flat_pos = positions.reshape(-1)
        ops.apply_rope_(q.view(B * Sq, attn.n_heads, -1),
                        k.view(B * Sq, attn.n_kv, -1), flat_pos,
                        attn.inv_freq)
        scale = attn.scale
        if layer_idx is not None:
            self._kv[layer_idx] = (k.clone(), v.clone())

        acc = None
        cur_k, cur_v = k, v
        for r in range(self.cp):
            src_chunk = (self.rank - r) % self.cp
            if src_chunk == self.rank:
                part = _partial_attn(q, cur_k, cur_v, scale,
                                     causal_diag=True)
                acc = _merge(acc, part)
            elif src_chunk < self.rank:
                part = _partial_attn(q, cur_k, cur_v, scale,
                                     causal_diag=False)
                acc = _merge(acc, part)
            # rotate K/V to the next rank (skip after the last step)
            if r < self.cp - 1 and self.cp > 1:
                nk = torch.empty_like(cur_k)
                nv = torch.empty_like(cur_v)
                nxt = (self.rank + 1) % self.cp
                prv = (self.rank - 1) % self.cp
                reqs = [dist.isend(cur_k.contiguous(), nxt),
                        dist.isend(cur_v.contiguous(), nxt)]
                dist.recv(nk, prv)
                dist.recv(nv, prv)
                for w in reqs:
                    w.wait()
                cur_k, cur_v = nk, nv
        o, m, l = acc
        out = (o / l.permute(0, 2, 1).unsqueeze(-1).clamp(min=1e-30))
        return attn.o_proj(out.to(x.dtype).reshape(B, Sq, -1))

    @torch.no_grad()
    def forward_logits(self, input_ids: torch.Tensor,
                       keep_kv: bool = False) -> torch.Tensor:
        """Prefill logits for this rank's sequence chunk. input_ids is the
        FULL [B, S] prompt on every rank (S divisible by cp). keep_kv
        caches this rank's K/V chunk for subsequent decode."""
        B, S = input_ids.shape
        assert S % self.cp == 0, "S must divide cp"
        sc = S // self.cp
        ids = input_ids.to(self.device)[:, self.rank * sc:(self.rank + 1) * sc]
        positions = (torch.arange(sc, device=self.device,
                                  dtype=torch.int32) + self.rank * sc) \
            .unsqueeze(0).expand(B, -1).contiguous()

        stage = self.stage
        if keep_kv:
            self._kv = [None] * len(stage.layers)
        hidden = stage.embed(ids)
        eps = self.config.rms_norm_eps
        for li, layer in enumerate(stage.layers):
            h = ops.rmsnorm(hidden, layer.input_layernorm.to(hidden.dtype),
                            eps)
            hidden = hidden + self._ring_attention(
                layer.self_attn, h, positions,
                layer_idx=li if keep_kv else None)
            h = ops.rmsnorm(hidden,
                            layer.post_attention_layernorm.to(hidden.dtype),
                            eps)
            hidden = hidden + layer.mlp(h)
        return stage.head(hidden)       # [B, sc, V] — this rank's chunk

    # ------------------------------------------------------------------
    # decode over the sequence-sharded cache
    # ------------------------------------------------------------------
    def _decode_attention(self, attn, x, pos: int, layer_idx: int):
        """One decode step of one attention layer: partial attention over
        this rank's K/V shard, merged across the group with one all-gather
        of (o, m, l). New tokens' K/V append to the LAST rank."""
        B = x.shape[0]
        qkv = attn.qkv_proj(x)
        q, k, v = attn._split_qkv(qkv, B, 1)
        if attn.use_qk_norm:
            eps = attn.config.rms_norm_eps
            q = ops.rmsnorm(q.contiguous(), attn.q_norm.to(q.dtype), eps)
            k = ops.rmsnorm(k.contiguous(), attn.k_norm.to(k.dtype), eps)
        q = q.contiguous()
        k = k.contiguous()
        positions = torch.full((B,), pos, device=self.device,
                               dtype=torch.int32)
        ops.apply_rope_(q.view(B, attn.n_heads, -1),
                        k.view(B, attn.n_kv, -1), positions, attn.inv_freq)
        if self.rank == self.cp - 1:        # owner of all decode tokens
            k_loc = torch.cat([self._kv[layer_idx][0], k], dim=1)
            v_loc = torch.cat([self._kv[layer_idx][1], v], dim=1)
            self._kv[layer_idx] = (k_loc, v_loc)
        k_loc, v_loc = self._kv[layer_idx]
        o, m, l = _partial_attn(q, k_loc, v_loc, attn.scale,
                                causal_diag=False)
        if self.cp > 1:
            gath = []
            for t in (o, m, l):
                t = t.contiguous()
                bufs = [torch.empty_like(t) for _ in range(self.cp)]
                dist.all_gather(bufs, t)
                gath.append(bufs)
            acc = None
            for r in range(self.cp):
                acc = _merge(acc, (gath[0][r], gath[1][r], gath[2][r]))
            o, m, l = acc
        out = o / l.permute(0, 2, 1).unsqueeze(-1).clamp(min=1e-30)
        return attn.o_proj(out.to(x.dtype).reshape(B, 1, -1))

    @torch.no_grad()
    def _decode_forward(self, tok: torch.Tensor, pos: int) -> torch.Tensor:
        """Replicated single-token forward over the sharded cache."""
        stage = self.stage
        eps = self.config.rms_norm_eps
        hidden = stage.embed(tok.view(-1, 1))
        for li, layer in enumerate(stage.layers):
            h = ops.rmsnorm(hidden, layer.input_layernorm.to(hidden.dtype),
                            eps)
            hidden = hidden + self._decode_attention(layer.self_attn, h,
                                                     pos, li)
            h = ops.rmsnorm(hidden,
                            layer.post_attention_layernorm.to(hidden.dtype),
                            eps)
            hidden = hidden + layer.mlp(h)
        return stage.head(hidden).squeeze(1)        # [B, V]

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor,
                 max_new_tokens: int = 16) -> torch.Tensor:
        """Greedy generation: CP ring prefill, then sharded-cache decode.
        Every rank returns the full [B, max_new_tokens] output (sampling
        is replicated — the merged logits agree on all ranks; the prompt's
        last-token logits live on the last rank and are broadcast)."""
        B, S = input_ids.shape
        logits = self.forward_logits(input_ids, keep_kv=True)
        cur = logits[:, -1].argmax(-1)              # correct on last rank
        dist.broadcast(cur, self.cp - 1)
        out = torch.empty(B, max_new_tokens, device=self.device,
                          dtype=torch.int64)
        out[:, 0] = cur
        for t in range(1, max_new_tokens):
            lg = self._decode_forward(cur, S + t - 1)
            cur = lg.argmax(-1)
            out[:, t] = cur
        return out


class _RingPass(torch.autograd.Function):
    """Differentiable ring rotation: forward sends x to rank+1 and
    receives from rank-1; backward routes the received tensor's gradient
    back to its origin (send to rank-1, receive own grad from rank+1).
    SPMD-symmetric: every rank executes the same pass sequence, so the
    blocking recv always has a matching isend."""

    @staticmethod
    def forward(ctx, x, rank, world):
        ctx.ring = (rank, world)
        y = torch.empty_like(x)
        w = dist.isend(x.contiguous(), (rank + 1) % world)
        dist.recv(y, (rank - 1) % world)
        w.wait()
        return y

    @staticmethod
    def backward(ctx, dy):
        rank, world = ctx.ring
        dy = dy.contiguous()
        dx = torch.empty_like(dy)
        w = dist.isend(dy, (rank - 1) % world)
        dist.recv(dx, (rank + 1) % world)
        w.wait()
        return dx, None, None


def _partial_attn_train(q, k, v, scale, causal_diag):
    """Differentiable version of :func:`_partial_attn` (row max
    detached — exact, since softmax is shift-invariant)."""
    B, Sq, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    qf = q.float().permute(0, 2, 1, 3)
    kf = k.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    vf = v.float().permute(0, 2, 1, 3).repeat_interleave(rep, 1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal_diag:
        Sk = k.shape[1]
        mask = torch.triu(torch.ones(Sq, Sk, dtype=torch.bool,
                                     device=q.device), 1)
        scores = scores.masked_fill(mask, float("-inf"))
    m = scores.amax(-1).detach()
    p = torch.exp(scores - m.unsqueeze(-1))
    p = torch.nan_to_num(p, nan=0.0)
    l = p.sum(-1)
    o = torch.matmul(p, vf)
    return o.permute(0, 2, 1, 3), m, l


class CPTrainer:
    """Context-parallel training (weights replicated, sequence sharded).

    Every rank holds the full model and trains on its sequence chunk;
    attention runs the differentiable K/V ring. Losses are per-chunk
    sums normalized by the GLOBAL token count, so after backward each
    rank holds the partial dW that flows through its chunk's compute —
    one all-reduce SUM over the flat gradient completes dL/dW exactly
    (weights replicated => gradients must be summed, not averaged).
    """

    def __init__(self, model, rank: int, cp: int, device=None, seed: int = 0,
                 lr: float = 1e-3, weight_decay: float = 0.01):
        from tensorlink_amd.optim import FusedAdamW
        self.rank, self.cp = rank, cp
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        self.dtype = (torch.bfloat16 if self.device.type == "cuda"
                      else torch.float32)
        config = (model if isinstance(model, ModelConfig)
                  else get_config(model))
        self.stage = build_full_model(config)
        init_random_stage(self.stage, device=self.device, dtype=self.dtype,
                          seed=seed)
        self.stage.train()
        for p in self.stage.parameters():
            p.requires_grad_(True)
        self.config = config
        self.opt = FusedAdamW(self.stage.parameters(), lr=lr,
                              weight_decay=weight_decay)

    def _ring_attention_train(self, attn, x, positions):
        B, sc, H = x.shape
        qkv = attn.qkv_proj(x)
        q, k, v = attn._split_qkv(qkv, B, sc)
        if attn.use_qk_norm:
            eps = attn.config.rms_norm_eps
            q = ops.rmsnorm(q.contiguous(), attn.q_norm.to(q.dtype), eps)
            k = ops.rmsnorm(k.contiguous(), attn.k_norm.to(k.dtype), eps)
        flat_pos = positions.reshape(-1)
        q2, k2 = ops.apply_rope(q.reshape(B * sc, attn.n_heads, -1),
                                k.reshape(B * sc, attn.n_kv, -1),
                                flat_pos, attn.inv_freq)
        q = q2.view(B, sc, attn.n_heads, -1)
        k = k2.view(B, sc, attn.n_kv, -1)
        acc = None
        cur_k, cur_v = k, v
        for r in range(self.cp):
            src = (self.rank - r) % self.cp
            if src == self.rank:
                acc = _merge(acc, _partial_attn_train(q, cur_k, cur_v,
                                                      attn.scale, True))
            elif src < self.rank:
                acc = _merge(acc, _partial_attn_train(q, cur_k, cur_v,
                                                      attn.scale, False))
            if r < self.cp - 1 and self.cp > 1:
                cur_k = _RingPass.apply(cur_k, self.rank, self.cp)
                cur_v = _RingPass.apply(cur_v, self.rank, self.cp)
        o, m, l = acc
        out = o / l.permute(0, 2, 1).unsqueeze(-1).clamp(min=1e-30)
        out = out.to(x.dtype).reshape(B, sc, -1)
        # anchor the final (possibly unused) ring buffers so every rank
        # runs the SAME backward pass sequence (a pruned chunk's
        # _RingPass would otherwise never fire its backward and deadlock
        # the symmetric grad rotation); the added grads are exactly zero
        if self.cp > 1:
            out = out + 0.0 * (cur_k.sum() + cur_v.sum()).to(out.dtype)
        return attn.o_proj(out)

    def _forward_chunk(self, ids_chunk, positions):
        stage = self.stage
        eps = self.config.rms_norm_eps
        hidden = stage.embed(ids_chunk)
        for layer in stage.layers:
            h = ops.rmsnorm(hidden, layer.input_layernorm.to(hidden.dtype),
                            eps)
            hidden = hidden + self._ring_attention_train(layer.self_attn,
                                                         h, positions)
            h = ops.rmsnorm(hidden,
                            layer.post_attention_layernorm.to(hidden.dtype),
                            eps)
            hidden = hidden + layer.mlp(h)
        return stage.head(hidden)

    def train_step(self, input_ids: torch.Tensor) -> float:
        """One optimizer step; input_ids [B, S] (full sequence, identical
        on every rank; S divisible by cp). Returns the global mean loss."""
        import torch.nn.functional as F
        B, S = input_ids.shape
        assert S % self.cp == 0
        sc = S // self.cp
        ids = input_ids.to(self.device)
        t0 = self.rank * sc
        chunk = ids[:, t0:t0 + sc]
        positions = (torch.arange(sc, device=self.device,
                                  dtype=torch.int32) + t0) \
            .unsqueeze(0).expand(B, -1).contiguous()
        self.opt.zero_grad()
        logits = self._forward_chunk(chunk, positions)
        # shifted CE with one-token lookahead across the chunk boundary;
        # the global last position has no label
        if self.rank == self.cp - 1:
            lg = logits[:, :-1]
            lb = ids[:, t0 + 1:t0 + sc]
        else:
            lg = logits
            lb = ids[:, t0 + 1:t0 + sc + 1]
        n_total = B * (S - 1)
        loss = F.cross_entropy(lg.reshape(-1, lg.shape[-1]).float(),
                               lb.reshape(-1), reduction="sum") / n_total
        loss.backward()
        if self.cp > 1:
            dist.all_reduce(self.opt.flat_grad)     # SUM of partial dW
            total = loss.detach().clone()
            dist.all_reduce(total)
        else:
            total = loss.detach()
        self.opt.step()
        return float(total)
