"""Tensor parallelism (opt-in, beyond reference parity).

The reference has no intra-layer sharding ("TP: NO").
This module adds Megatron-style TP for inference: attention heads and MLP
intermediate are sharded column-parallel across ranks; the o/down
projections are row-parallel and their outputs are summed with ONE RCCL
all-reduce each over xGMI — so a pure-TP decode step needs exactly two
all-reduces per layer and no point-to-point traffic (every rank holds the
full residual stream and computes identical logits after the reductions).

Design choice: sharding happens by *construction* — a rank builds the
model from a head-sharded config (the layer code in models/dense.py is
untouched), the o/down linears are swapped for all-reducing variants, and
weights are sliced from a full-model state so TP output is comparable to
the single-rank reference. KV caches hold only the local heads, so cache
memory also divides by TP.
"""

from __future__ import annotations

import dataclasses
from typing import Optional

import torch
import torch.distributed as dist

from tensorlink_amd.models.configs import ModelConfig, get_config
from tensorlink_amd.models.dense import TLLinear, build_full_model
from tensorlink_amd.models.loader import init_random_stage
from tensorlink_amd.parallel.comm import device_for_rank
from tensorlink_amd.parallel.pipeline import SamplingParams


class _ReduceFromTP(torch.autograd.Function):
    """Megatron's g: all-reduce forward, identity backward (row-parallel
    outputs — each rank holds a partial sum, gradients are replicated)."""

    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, dy):
        return dy, None


class _CopyToTP(torch.autograd.Function):
    """Megatron's f: identity forward, all-reduce backward (column-parallel
    inputs — x is replicated, each rank produces a partial dx)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous()
        dist.all_reduce(dy, group=ctx.group)
        return dy, None


def _tp_active(group=None) -> bool:
    return dist.is_initialized() and dist.get_world_size(group) > 1


class AllReduceLinear(TLLinear):
    """Row-parallel output projection: local GEMM then sum across the TP
    group (``self.group``; None = world). Differentiable — the training
    path routes the all-reduce through :class:`_ReduceFromTP` so
    gradients flow (identity backward)."""

    group = None            # TP subgroup (TP x PP grids set it per-stage)

    def forward(self, x):
        y = super().forward(x)
        if _tp_active(self.group):
            if torch.is_grad_enabled() and y.requires_grad:
                y = _ReduceFromTP.apply(y, self.group)
            else:
                y = y.contiguous()
                dist.all_reduce(y, group=self.group)
        return y


class ColumnParallelLinear(TLLinear):
    """Column-parallel input projection (qkv / gate_up): weight rows are
    sharded, x is replicated. Forward is the plain local GEMM; in training
    the input passes through :class:`_CopyToTP` so each rank's partial dx
    is summed in backward (inference: pure identity, zero overhead)."""

    group = None

    def forward(self, x):
        if _tp_active(self.group) and torch.is_grad_enabled() \
                and x.requires_grad:
            x = _CopyToTP.apply(x, self.group)
        return super().forward(x)


def local_config(config: ModelConfig, tp: int) -> ModelConfig:
    assert config.num_attention_heads % tp == 0, "heads must divide tp"
    assert config.num_key_value_heads % tp == 0, "kv heads must divide tp"
    assert config.intermediate_size % tp == 0
    return dataclasses.replace(
        config,
        num_attention_heads=config.num_attention_heads // tp,
        num_key_value_heads=config.num_key_value_heads // tp,
        intermediate_size=config.intermediate_size // tp,
        head_dim=config.head_dim)


def shard_state(full_state: dict, config: ModelConfig, tp_rank: int,
                tp: int, vocab_parallel: bool = False) -> dict:
    """Slice a full-model state dict into this rank's TP shard.
    vocab_parallel additionally shards the LM head over the vocab dim
    (Megatron-style — training path; inference keeps it replicated)."""
    q, kv, i = config.q_size, config.kv_size, config.intermediate_size
    ql, kvl, il = q // tp, kv // tp, i // tp
    out = {}
    for name, w in full_state.items():
        if vocab_parallel and name == "lm_head.weight":
            vl = w.shape[0] // tp
            out[name] = w[tp_rank * vl:(tp_rank + 1) * vl]
            continue
        if "qkv_proj" in name:
            qs = w[tp_rank * ql:(tp_rank + 1) * ql]
            ks = w[q + tp_rank * kvl:q + (tp_rank + 1) * kvl]
            vs = w[q + kv + tp_rank * kvl:q + kv + (tp_rank + 1) * kvl]
            out[name] = torch.cat([qs, ks, vs], dim=0)
        elif "gate_up_proj" in name:
            gs = w[tp_rank * il:(tp_rank + 1) * il]
            us = w[i + tp_rank * il:i + (tp_rank + 1) * il]
            out[name] = torch.cat([gs, us], dim=0)
        elif "o_proj.weight" in name:
            out[name] = w[:, tp_rank * ql:(tp_rank + 1) * ql]
        elif "down_proj.weight" in name:
            out[name] = w[:, tp_rank * il:(tp_rank + 1) * il]
        else:
            out[name] = w          # norms, embeddings, head: replicated
    return {k: v.contiguous() for k, v in out.items()}


def build_tp_model(config_or_name, tp_rank: int, tp: int, device=None,
                   dtype=None, seed: int = 0, vocab_parallel: bool = False):
    """Full model (PP=1) sharded TP-ways. Weights come from slicing the
    SAME seeded full-model init every rank, so a TP group reproduces the
    single-rank reference."""
    config = (config_or_name if isinstance(config_or_name, ModelConfig)
              else get_config(config_or_name))
    device = device if device is not None else device_for_rank()
    dtype = dtype or (torch.bfloat16 if device.type == "cuda"
                      else torch.float32)
    assert not config.is_moe, \
        "TP shards dense projections; shard MoE experts with EP " \
        "(parallel/ep.py) instead"
    if vocab_parallel:
        assert not config.tie_word_embeddings, \
            "vocab-parallel head requires untied embeddings"
        assert config.vocab_size % tp == 0

    full = build_full_model(config)
    init_random_stage(full, device="cpu", dtype=dtype, seed=seed)
    shards = shard_state(full.state_dict(), config, tp_rank, tp,
                         vocab_parallel=vocab_parallel)
    del full

    stage = build_full_model(local_config(config, tp))
    if vocab_parallel:
        # column-parallel: output (vocab) dim sharded, so backward must
        # all-reduce the input gradient across the group
        stage.lm_head = ColumnParallelLinear(config.hidden_size,
                                             config.vocab_size // tp,
                                             bias=False)
    # swap row-parallel outputs for all-reducing variants and
    # column-parallel inputs for the f-op (backward all-reduce) variants
    for layer in stage.layers:
        for holder, name, cls in (
                (layer.self_attn, "o_proj", AllReduceLinear),
                (layer.mlp, "down_proj", AllReduceLinear),
                (layer.self_attn, "qkv_proj", ColumnParallelLinear),
                (layer.mlp, "gate_up_proj", ColumnParallelLinear)):
            old = getattr(holder, name)
            new = cls(old.in_features, old.out_features,
                      bias=old.bias is not None)
            setattr(holder, name, new)
    stage.load_state_dict(shards)
    stage.to(device=device, dtype=dtype)
    stage.eval()
    return stage


class TPRunner:
    """Pure-TP generation (SPMD across the TP group): every rank runs the
    same decode loop on its head shard; logits agree after the per-layer
    all-reduces, so sampling is local and no tokens travel."""

    def __init__(self, model, rank: int, tp: int, device=None, seed: int = 0):
        self.rank, self.tp = rank, tp
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        self.stage = build_tp_model(model, rank, tp, device=self.device,
                                    seed=seed)
        self.config = self.stage.config

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor,
                 sampling: Optional[SamplingParams] = None) -> torch.Tensor:
        sp = sampling or SamplingParams()
        B, S = input_ids.shape
        ids = input_ids.to(self.device)
        cache = self.stage.make_kv_cache(B, S + sp.max_new_tokens,
                                         self.device)
        pos = torch.arange(S, device=self.device,
                           dtype=torch.int32).unsqueeze(0).expand(B, -1)
        hidden = self.stage(ids, pos.contiguous(), kv_cache=cache,
                            return_logits=False)
        logits = self.stage.head(hidden[:, -1:]).squeeze(1)
        out = torch.empty(B, sp.max_new_tokens, device=self.device,
                          dtype=torch.int64)
        cur = logits.argmax(-1)
        out[:, 0] = cur
        positions = torch.full((B,), S, device=self.device,
                               dtype=torch.int32)
        for t in range(1, sp.max_new_tokens):
            lg = self.stage(cur.unsqueeze(1), positions.unsqueeze(1),
                            kv_cache=cache).squeeze(1)
            cur = lg.argmax(-1)
            out[:, t] = cur
            positions += 1
        return out


class _VocabParallelCE(torch.autograd.Function):
    """Cross-entropy over vocab-sharded logits without gathering them
    (Megatron algorithm): all-reduce MAX for the row max, SUM for the
    exp-denominator and the target logit; backward is purely local
    (softmax_local minus the local one-hot)."""

    @staticmethod
    def forward(ctx, logits, labels, v0, group):
        # logits [N, Vl] fp32; labels [N] global ids (-100 = ignored,
        # matching the non-vp chunked/plain CE paths)
        valid = labels != -100
        lmax = logits.max(-1).values.contiguous()
        dist.all_reduce(lmax, op=dist.ReduceOp.MAX, group=group)
        z = (logits - lmax.unsqueeze(1)).exp()
        denom = z.sum(-1).contiguous()
        dist.all_reduce(denom, group=group)
        Vl = logits.shape[1]
        in_shard = (labels >= v0) & (labels < v0 + Vl)
        idx = (labels - v0).clamp(0, Vl - 1)
        tgt = (logits.gather(1, idx.unsqueeze(1)).squeeze(1)
               - lmax) * in_shard
        tgt = tgt.contiguous()
        dist.all_reduce(tgt, group=group)
        loss = (denom.log() - tgt) * valid
        n_valid = valid.sum().clamp(min=1)
        ctx.save_for_backward(z, denom, idx, in_shard, valid, n_valid)
        ctx.group = group
        return loss.sum() / n_valid

    @staticmethod
    def backward(ctx, dloss):
        z, denom, idx, in_shard, valid, n_valid = ctx.saved_tensors
        dlogits = z / denom.unsqueeze(1)
        dlogits.scatter_add_(
            1, idx.unsqueeze(1),
            -in_shard.to(dlogits.dtype).unsqueeze(1))
        dlogits.mul_(valid.to(dlogits.dtype).unsqueeze(1))
        dlogits.mul_(dloss / n_valid)
        return dlogits, None, None, None


def vocab_parallel_ce(logits_local, labels, vocab_start, group=None):
    return _VocabParallelCE.apply(logits_local.float(), labels,
                                  vocab_start, group)


class TPTrainer:
    """SPMD tensor-parallel training (PP=1): every rank holds a head/
    intermediate shard, runs the full replicated batch, and computes the
    identical loss after the per-layer reductions (row-parallel forward
    all-reduce; column-parallel backward all-reduce via _CopyToTP).

    Replicated parameters (norms, embeddings, lm head) receive full
    gradients on every rank from identical replicated activations; their
    grads are all-reduce-averaged each step only to pin down fp
    drift across ranks (exact arithmetic would make it a no-op).
    """

    def __init__(self, model, rank: int, tp: int, device=None, seed: int = 0,
                 lr: float = 1e-3, weight_decay: float = 0.01,
                 vocab_parallel: bool = False):
        from tensorlink_amd.optim import FusedAdamW
        self.rank, self.tp = rank, tp
        self.vocab_parallel = vocab_parallel
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        self.stage = build_tp_model(model, rank, tp, device=self.device,
                                    seed=seed,
                                    vocab_parallel=vocab_parallel)
        self.stage.train()
        for p in self.stage.parameters():
            p.requires_grad_(True)
        self.config = self.stage.config
        sharded = ("qkv_proj", "gate_up_proj", "o_proj", "down_proj")
        if vocab_parallel:
            sharded = sharded + ("lm_head",)
        self._replicated = [
            p for n, p in self.stage.named_parameters()
            if not any(t in n for t in sharded)]
        self._v0 = rank * (self.config.vocab_size // tp
                           if vocab_parallel else 0)
        self.opt = FusedAdamW(self.stage.parameters(), lr=lr,
                              weight_decay=weight_decay)

    def train_step(self, input_ids: torch.Tensor,
                   labels: Optional[torch.Tensor] = None) -> float:
        from tensorlink_amd import ops as tl_ops
        ids = input_ids.to(self.device)
        labels = ids if labels is None else labels.to(self.device)
        B, S = ids.shape
        pos = torch.arange(S, device=self.device,
                           dtype=torch.int32).unsqueeze(0).expand(B, -1)
        self.opt.zero_grad()
        hidden = self.stage(ids, pos.contiguous(), training=True,
                            return_logits=False)
        logits = self.stage.head(hidden)     # [B,S,V] or [B,S,V/tp] (vp)
        if self.vocab_parallel:
            lg = logits[:, :-1].reshape(-1, logits.shape[-1])
            lb = labels[:, 1:].reshape(-1)
            loss = vocab_parallel_ce(lg, lb, self._v0)
        else:
            loss = tl_ops.causal_lm_loss(logits, labels)
        loss.backward()
        if _tp_active():
            for p in self._replicated:
                if p.grad is not None:
                    g = p.grad.contiguous()
                    dist.all_reduce(g)
                    p.grad = g.div_(self.tp)
        self.opt.step()
        return float(loss.detach())


# ---------------------------------------------------------------------------
# TP x PP grid
# ---------------------------------------------------------------------------
def _swap_tp_linears(stage, tp_group):
    """Swap a StageModel's projections for TP variants bound to
    tp_group."""
    for layer in stage.layers:
        for holder, name, cls in (
                (layer.self_attn, "o_proj", AllReduceLinear),
                (layer.mlp, "down_proj", AllReduceLinear),
                (layer.self_attn, "qkv_proj", ColumnParallelLinear),
                (layer.mlp, "gate_up_proj", ColumnParallelLinear)):
            old = getattr(holder, name)
            new = cls(old.in_features, old.out_features,
                      bias=old.bias is not None)
            new.group = tp_group
            setattr(holder, name, new)


class TPPPRunner:
    """2-D TP x PP inference grid on a world of pp*tp ranks.

    Grid: ``rank = tp_rank * pp + stage`` — each pipeline replica's ranks
    are consecutive (P2P's rank_base offset maps pipeline-local ranks),
    and the TP group for stage s is the stride-pp set {t*pp + s}. Stage
    weights come from the SAME per-stage seeded init a pure-PP run uses,
    sliced by tp_rank, so a tp x pp grid reproduces the pp-only output
    exactly. The pipeline replicas are NOT independent: every o/down
    projection all-reduces across the tp group, which keeps all replicas
    in lockstep layer by layer. Sampling must therefore be deterministic
    across replicas: greedy, or an identical SamplingParams.seed.

    The planner runs on the TP-LOCAL config, so per-stage memory checks
    see the sharded sizes — folding TP into the stage-assignment grid.
    """

    def __init__(self, model, rank: int, world: int, tp: int, device=None,
                 dtype=None, seed: int = 0):
        import dataclasses as _dc

        from tensorlink_amd.models.dense import build_stage
        from tensorlink_amd.parallel.pipeline import PipelineRunner
        from tensorlink_amd.parallel.planner import plan_for_world
        config = (model if isinstance(model, ModelConfig)
                  else get_config(model))
        assert world % tp == 0, "world must divide tp"
        pp = world // tp
        self.pp, self.tp = pp, tp
        self.stage_idx, self.tp_rank = rank % pp, rank // pp
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        dtype = dtype or (torch.bfloat16 if self.device.type == "cuda"
                          else torch.float32)
        # same group-creation order on every rank (dist.new_group is
        # collective over the world)
        pp_groups = [dist.new_group(list(range(t * pp, (t + 1) * pp)))
                     for t in range(tp)] if pp > 1 and tp > 1 else \
            [None] * tp
        tp_groups = [dist.new_group([t * pp + s for t in range(tp)])
                     for s in range(pp)] if tp > 1 and pp > 1 else \
            [None] * pp
        self.tp_group = tp_groups[self.stage_idx]

        local_cfg = local_config(config, tp)
        plan = plan_for_world(local_cfg, pp)
        self.runner = PipelineRunner(
            plan, self.stage_idx, pp, device=self.device, init="empty",
            dtype=dtype, group=pp_groups[self.tp_rank],
            rank_base=self.tp_rank * pp)

        # seed the FULL stage exactly as a pure-PP rank would, then slice
        spec = plan.stage_for_rank(self.stage_idx)
        full = build_stage(config, spec)
        init_random_stage(full, device="cpu", dtype=dtype,
                          seed=seed + self.stage_idx)
        shards = shard_state(full.state_dict(), config, self.tp_rank, tp)
        del full
        if tp > 1:
            _swap_tp_linears(self.runner.stage, self.tp_group)
        self.runner.stage.load_state_dict(shards)
        self.runner.stage.to(device=self.device, dtype=dtype)
        self.runner.stage.eval()
        self.config = self.runner.config

    def generate(self, input_ids, sampling=None, **kw):
        """SPMD over the whole grid; input_ids significant on every
        pipeline replica's first rank (stage 0); tokens return on them.
        Stochastic sampling must be seeded so the replicas stay in
        lockstep (their per-layer TP all-reduces pair up)."""
        if sampling is not None and sampling.temperature > 0 \
                and sampling.seed is None:
            raise ValueError(
                "TP x PP sampling needs SamplingParams.seed: unseeded "
                "draws diverge across pipeline replicas and break the "
                "per-layer TP collectives")
        return self.runner.generate(input_ids, sampling, **kw)


class TPPPTrainer:
    """2-D TP x PP training grid (same rank layout as TPPPRunner).

    Each pipeline replica runs the 1F1B schedule on TP-sharded stages;
    the f/g collectives inside the swapped projections synchronize the
    tp group layer by layer (every replica executes the identical
    micro-batch schedule, so the collectives pair). Replicated
    parameters (norms, embeddings, head) get full gradients on every
    tp peer from identical activations; a pre-step all-reduce-average
    over the tp group pins down fp drift (exact arithmetic would make
    it a no-op).
    """

    def __init__(self, model, rank: int, world: int, tp: int, device=None,
                 dtype=None, seed: int = 0, lr: float = 1e-4,
                 grid_base: int = 0, pp_groups=None, tp_groups=None,
                 vocab_parallel: bool = False, **opt_kwargs):
        """rank/world are GRID-local (0..tp*pp). grid_base offsets them
        into the global rank space, and pp_groups/tp_groups may be
        pre-built by an outer 3-D (DP) wrapper — dist.new_group is
        collective over the WORLD, so nested grids must hoist group
        creation."""
        from tensorlink_amd.models.dense import build_stage
        from tensorlink_amd.optim import FusedAdamW
        from tensorlink_amd.parallel.pipeline import PipelineTrainer
        from tensorlink_amd.parallel.planner import plan_for_world
        config = (model if isinstance(model, ModelConfig)
                  else get_config(model))
        assert world % tp == 0
        pp = world // tp
        self.pp, self.tp = pp, tp
        self.stage_idx, self.tp_rank = rank % pp, rank // pp
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        dtype = dtype or (torch.bfloat16 if self.device.type == "cuda"
                          else torch.float32)
        if pp_groups is None:
            pp_groups = [dist.new_group(
                [grid_base + t * pp + s for s in range(pp)])
                for t in range(tp)] if pp > 1 and tp > 1 else [None] * tp
        if tp_groups is None:
            tp_groups = [dist.new_group(
                [grid_base + t * pp + s for t in range(tp)])
                for s in range(pp)] if tp > 1 and pp > 1 else [None] * pp
        self.tp_group = tp_groups[self.stage_idx]

        local_cfg = local_config(config, tp)
        plan = plan_for_world(local_cfg, pp, training=True)
        self.trainer = PipelineTrainer(
            plan, self.stage_idx, pp, device=self.device, init="empty",
            dtype=dtype, seed=seed, lr=lr, group=pp_groups[self.tp_rank],
            rank_base=grid_base + self.tp_rank * pp, **opt_kwargs)
        stage = self.trainer.stage

        spec = plan.stage_for_rank(self.stage_idx)
        full = build_stage(config, spec)
        init_random_stage(full, device="cpu", dtype=dtype,
                          seed=seed + self.stage_idx)
        vp = vocab_parallel and stage.has_head and tp > 1
        if vocab_parallel:
            assert not config.tie_word_embeddings
        shards = shard_state(full.state_dict(), config, self.tp_rank, tp,
                             vocab_parallel=vp)
        del full
        if tp > 1:
            _swap_tp_linears(stage, self.tp_group)
        if vp:
            head = ColumnParallelLinear(config.hidden_size,
                                        config.vocab_size // tp,
                                        bias=False)
            head.group = self.tp_group
            stage.lm_head = head
            self.trainer._vp = (self.tp_rank * (config.vocab_size // tp),
                                self.tp_group)
        stage.load_state_dict(shards)
        stage.to(device=self.device, dtype=dtype)
        stage.train()
        for p in stage.parameters():
            p.requires_grad_(True)
        # the swap replaced modules AFTER PipelineTrainer built its
        # optimizer — rebuild over the final parameter set
        self.trainer.optimizer = FusedAdamW(stage.parameters(), lr=lr,
                                            **opt_kwargs)
        sharded = ("qkv_proj", "gate_up_proj", "o_proj", "down_proj")
        if vp:
            sharded = sharded + ("lm_head",)
        self._replicated = [
            p for n, p in stage.named_parameters()
            if not any(t in n for t in sharded)]

        def _sync_replicated(tr):
            if self.tp <= 1:
                return
            for p in self._replicated:
                if p.grad is not None:
                    dist.all_reduce(p.grad, group=self.tp_group)
                    p.grad.div_(self.tp)
        self.trainer.grad_hook = _sync_replicated
        self.config = stage.config

    def train_step(self, input_ids=None, labels=None, n_micro=None):
        """SPMD over the whole grid; input_ids/labels significant on
        every replica's first rank (pass the same batch to each)."""
        return self.trainer.train_step(input_ids, labels, n_micro=n_micro)


class Hybrid3DTrainer:
    """DP x TP x PP 3-D training grid on world = dp * tp * pp ranks.

    Rank layout: ``rank = dp_rank * (tp*pp) + tp_rank * pp + stage``.
    Each DP replica is a TP x PP grid training its shard of the global
    batch; before the optimizer step the gradients are (1) tp-synced
    for replicated params (inside TPPPTrainer's hook), then (2)
    all-reduce-AVERAGED across the dp peers holding the same
    (tp_rank, stage) shard. Averaging matches mean-reduction losses, so
    the trajectory equals a single replica training on the concatenated
    batch. All dist.new_group calls are hoisted here in one
    deterministic order (the call is collective over the world).
    """

    def __init__(self, model, rank: int, world: int, dp: int, tp: int,
                 device=None, dtype=None, seed: int = 0, lr: float = 1e-4,
                 **opt_kwargs):
        rep = world // dp
        pp = rep // tp
        assert dp * tp * pp == world
        self.dp, self.tp, self.pp = dp, tp, pp
        self.dp_rank = rank // rep
        lrank = rank % rep
        g = lambda ranks: dist.new_group(ranks) if len(ranks) > 1 else None
        pp_groups_all = [[g([d * rep + t * pp + s for s in range(pp)])
                          for t in range(tp)] for d in range(dp)]
        tp_groups_all = [[g([d * rep + t * pp + s for t in range(tp)])
                          for s in range(pp)] for d in range(dp)]
        dp_groups = [[g([d * rep + t * pp + s for d in range(dp)])
                      for s in range(pp)] for t in range(tp)]
        self.inner = TPPPTrainer(
            model, lrank, rep, tp, device=device, dtype=dtype, seed=seed,
            lr=lr, grid_base=self.dp_rank * rep,
            pp_groups=pp_groups_all[self.dp_rank],
            tp_groups=tp_groups_all[self.dp_rank], **opt_kwargs)
        self.dp_group = dp_groups[self.inner.tp_rank][self.inner.stage_idx]
        self.config = self.inner.config

        tp_hook = self.inner.trainer.grad_hook

        def _hook(tr):
            if tp_hook is not None:
                tp_hook(tr)
            if self.dp > 1:
                dist.all_reduce(tr.optimizer.flat_grad,
                                group=self.dp_group)
                tr.optimizer.flat_grad.div_(self.dp)
        self.inner.trainer.grad_hook = _hook

    def train_step(self, input_ids=None, labels=None, n_micro=None):
        """input_ids/labels = THIS dp replica's batch shard, significant
        on the replica's first rank. Returns the replica-shard loss."""
        return self.inner.train_step(input_ids, labels, n_micro=n_micro)
