"""Expert parallelism for MoE layers (opt-in, beyond reference parity).

The reference has no MoE-aware code at all; this build's
default places each layer's full expert set on its pipeline stage. EP
additionally shards the EXPERTS of every MoE layer across ranks: each
rank stores and computes only its num_experts/ep local experts, the
router (tiny) is replicated, and the layer output is the all-reduce sum
of per-rank partial outputs — token t's contribution comes only from the
rank(s) owning its routed experts, so the sum reconstructs the dense
result exactly.

Two exchange modes per MoE layer (``mode=`` on EPMoEMLP/EPRunner):

- ``"allreduce"`` (default): every rank runs the router over ALL tokens,
  computes its local experts' contribution and the layer output is the
  all-reduce sum of partials. One hidden-sized all-reduce (2·T·H words
  on a ring); simple and overlap-friendly on xGMI.
- ``"alltoall"``: tokens are sharded across the EP group; each rank
  routes its T/ep tokens and dispatches them to their experts' owner
  ranks with an uneven ``all_to_all_single`` (dispatch), gets expert
  outputs back (combine), and the token shards are all-gathered to
  rebuild the replicated stream. Traffic ≈ (2·top_k + 1)·T/ep·H words —
  cheaper than the all-reduce when ep > 2·top_k + 1, i.e. for large EP
  groups with few active experts per token (the Mixtral regime at ep=8,
  top_k=2 is the break-even point on paper; measure on xGMI).
"""

from __future__ import annotations

import dataclasses
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from tensorlink_amd import ops
from tensorlink_amd.models.configs import ModelConfig, get_config
from tensorlink_amd.models.dense import MoEMLP, build_full_model
from tensorlink_amd.models.loader import init_random_stage
from tensorlink_amd.parallel.comm import device_for_rank
from tensorlink_amd.parallel.pipeline import SamplingParams


class EPMoEMLP(nn.Module):
    """MoE block holding a shard of the experts (see module docstring for
    the two exchange modes)."""

    def __init__(self, full: MoEMLP, ep_rank: int, ep: int,
                 mode: str = "allreduce"):
        super().__init__()
        n = full.num_experts
        assert n % ep == 0, "experts must divide ep"
        assert mode in ("allreduce", "alltoall")
        self.num_experts = n
        self.top_k = full.top_k
        self.ep_rank, self.ep = ep_rank, ep
        self.mode = mode
        self.local_n = n // ep
        self.local_base = ep_rank * self.local_n
        self.gate = full.gate                       # replicated router
        self.experts = nn.ModuleList(
            full.experts[self.local_base:self.local_base + self.local_n])

    def _run_local_experts(self, x_recv: torch.Tensor,
                           eid_recv: torch.Tensor) -> torch.Tensor:
        """Apply the owning local expert to each received row."""
        y = torch.empty_like(x_recv)
        for le, expert in enumerate(self.experts):
            m = (eid_recv == self.local_base + le).nonzero(as_tuple=True)[0]
            if m.numel():
                y[m] = expert(x_recv[m])
        return y

    def _forward_alltoall(self, x):
        B, S, H = x.shape
        flat = x.reshape(-1, H)
        T = flat.shape[0]
        ep, r = self.ep, self.ep_rank
        shard = (T + ep - 1) // ep                   # padded shard size
        t0, t1 = r * shard, min((r + 1) * shard, T)
        my = flat[t0:t1]                             # [t, H] this rank's
        weights, idx = ops.moe_topk_router(self.gate(my), self.top_k)
        weights = weights.to(x.dtype)

        # (token, slot) pairs sorted by destination rank
        t = my.shape[0]
        pair_tok = torch.arange(t, device=x.device
                                ).unsqueeze(1).expand(-1, self.top_k)
        dest = idx // self.local_n                   # [t, k]
        order = dest.reshape(-1).argsort(stable=True)
        send_eid = idx.reshape(-1)[order].contiguous()
        send_tok = pair_tok.reshape(-1)[order].contiguous()
        send_x = my[send_tok].contiguous()
        send_counts = torch.bincount(dest.reshape(-1),
                                     minlength=ep).tolist()

        # exchange pair counts, then dispatch tokens + expert ids
        # (tensors live on x.device — NCCL requires device-resident
        # collectives on GPU; gloo accepts CPU)
        cnt_in = torch.tensor(send_counts, dtype=torch.int64,
                              device=x.device)
        cnt_out = torch.empty_like(cnt_in)
        dist.all_to_all_single(cnt_out, cnt_in)
        recv_counts = cnt_out.tolist()
        n_recv = int(sum(recv_counts))
        x_recv = torch.empty(n_recv, H, device=x.device, dtype=x.dtype)
        eid_recv = torch.empty(n_recv, device=x.device, dtype=send_eid.dtype)
        dist.all_to_all_single(x_recv, send_x, recv_counts, send_counts)
        dist.all_to_all_single(eid_recv, send_eid, recv_counts, send_counts)

        # expert compute on owned rows, then combine (reverse exchange)
        y_recv = self._run_local_experts(x_recv, eid_recv)
        y_back = torch.empty_like(send_x)
        dist.all_to_all_single(y_back, y_recv, send_counts, recv_counts)

        out_my = torch.zeros(shard, H, device=x.device, dtype=x.dtype)
        w_pairs = weights.reshape(-1)[order]
        out_my.index_add_(0, send_tok, y_back * w_pairs[:, None])

        # rebuild the replicated stream from the token shards
        gathered = [torch.empty_like(out_my) for _ in range(ep)]
        dist.all_gather(gathered, out_my)
        return torch.cat(gathered, 0)[:T].reshape(B, S, H)

    def forward(self, x):
        if (self.mode == "alltoall" and dist.is_initialized()
                and dist.get_world_size() > 1):
            return self._forward_alltoall(x)
        B, S, H = x.shape
        flat = x.reshape(-1, H)
        if (dist.is_initialized() and dist.get_world_size() > 1
                and torch.is_grad_enabled() and flat.requires_grad):
            # f-collective on the block input (training): each rank's
            # graph holds only its local experts' (and its own gate
            # copy's) dL/dx terms, so dx must be SUMMED across the EP
            # group in backward
            from tensorlink_amd.parallel.tp import _CopyToTP
            flat = _CopyToTP.apply(flat, None)
        weights, idx = ops.moe_topk_router(self.gate(flat), self.top_k)
        weights = weights.to(x.dtype)
        out = torch.zeros_like(flat)
        for le, expert in enumerate(self.experts):
            e = self.local_base + le
            mask = (idx == e)
            tok, slot = mask.nonzero(as_tuple=True)
            if tok.numel() == 0:
                continue
            out = out.index_add(0, tok, expert(flat[tok]) * weights[tok,
                                                                    slot,
                                                                    None])
        if dist.is_initialized() and dist.get_world_size() > 1:
            if torch.is_grad_enabled() and out.requires_grad:
                from tensorlink_amd.parallel.tp import _ReduceFromTP
                out = _ReduceFromTP.apply(out, None)
            else:
                out = out.contiguous()
                dist.all_reduce(out)
        return out.reshape(B, S, H)


def build_ep_model(config_or_name, ep_rank: int, ep: int, device=None,
                   dtype=None, seed: int = 0, quantize: Optional[str] = None,
                   mode: str = "allreduce"):
    """Full MoE model (PP=1) with experts sharded EP-ways; weights come
    from the same seeded full init on every rank so the EP group
    reproduces the single-rank reference."""
    config = (config_or_name if isinstance(config_or_name, ModelConfig)
              else get_config(config_or_name))
    assert config.is_moe, "EP applies to MoE configs"
    device = device if device is not None else device_for_rank()
    dtype = dtype or (torch.bfloat16 if device.type == "cuda"
                      else torch.float32)
    stage = build_full_model(config)
    init_random_stage(stage, device="cpu", dtype=dtype, seed=seed)
    for layer in stage.layers:
        layer.mlp = EPMoEMLP(layer.mlp, ep_rank, ep, mode=mode)
    if quantize == "fp8":
        from tensorlink_amd.models.quant import Fp8Linear
        for layer in stage.layers:
            for expert in layer.mlp.experts:
                for name in ("gate_up_proj", "down_proj"):
                    setattr(expert, name,
                            Fp8Linear.from_linear(getattr(expert, name)))
    stage.to(device=device, dtype=dtype)
    stage.eval()
    return stage


class EPRunner:
    """SPMD MoE generation with expert-sharded layers (every rank holds
    the full attention/embedding weights and 1/ep of the experts)."""

    def __init__(self, model, rank: int, ep: int, device=None, seed: int = 0,
                 quantize: Optional[str] = None, mode: str = "allreduce"):
        self.rank, self.ep = rank, ep
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        self.stage = build_ep_model(model, rank, ep, device=self.device,
                                    seed=seed, quantize=quantize, mode=mode)
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
        out = torch.empty(B, sp.max_new_tokens, device=self.device,
                          dtype=torch.int64)
        cur = self.stage.head(hidden[:, -1:]).squeeze(1).argmax(-1)
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


class EPTrainer:
    """Expert-parallel training (partial-sum mode).

    Every rank runs the full replicated batch; the MoE partial sums pass
    through the differentiable all-reduce (identity backward), so after
    ``loss.backward()``:

    - expert grads are complete locally (only this rank computed those
      experts' contributions);
    - non-MoE grads (attention, norms, embeddings, head) are complete
      AND identical on every rank (the computation is replicated);
    - router/gate grads are PARTIAL (each rank's graph only saw its own
      experts' weight usages) — one all-reduce SUM per gate completes
      them.
    """

    def __init__(self, model, rank: int, ep: int, device=None, seed: int = 0,
                 lr: float = 1e-3, weight_decay: float = 0.01):
        from tensorlink_amd.optim import FusedAdamW
        self.rank, self.ep = rank, ep
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        self.stage = build_ep_model(model, rank, ep, device=self.device,
                                    seed=seed)
        self.stage.train()
        for p in self.stage.parameters():
            p.requires_grad_(True)
        self.config = self.stage.config
        self.gate_params = [layer.mlp.gate.weight
                            for layer in self.stage.layers
                            if isinstance(layer.mlp, EPMoEMLP)]
        self.opt = FusedAdamW(self.stage.parameters(), lr=lr,
                              weight_decay=weight_decay)

    def train_step(self, input_ids: torch.Tensor) -> float:
        from tensorlink_amd import ops as tl_ops
        ids = input_ids.to(self.device)
        B, S = ids.shape
        pos = torch.arange(S, device=self.device,
                           dtype=torch.int32).unsqueeze(0).expand(B, -1)
        self.opt.zero_grad()
        hidden = self.stage(ids, pos.contiguous(), training=True,
                            return_logits=False)
        logits = self.stage.head(hidden)
        loss = tl_ops.causal_lm_loss(logits, ids)
        loss.backward()
        if dist.is_initialized() and dist.get_world_size() > 1:
            for g in self.gate_params:
                if g.grad is not None:
                    # in-place: .grad is a view into the optimizer's
                    # flat gradient buffer
                    dist.all_reduce(g.grad)  # SUM completes router grads
        self.opt.step()
        return float(loss.detach())
