"""Expert parallelism for MoE layers (opt-in, beyond reference parity).

The reference has no MoE-aware code at all; this build's
default places each layer's full expert set on its pipeline stage. EP
additionally shards the EXPERTS of every MoE layer across ranks: each
rank stores and computes only its num_experts/ep local experts, the
router (tiny) is replicated, and the layer output is the all-reduce sum
of per-rank partial outputs — token t's contribution comes only from the
rank(s) owning its routed experts, so the sum reconstructs the dense
result exactly.

This "partial-sum" EP trades the classic all-to-all token exchange for
one hidden-sized all-reduce per MoE layer — the right starting point on
xGMI where all-reduce is per-link bound but simple and overlap-friendly;
token-routing all-to-all is the roadmap follow-up.
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
    """MoE block holding a shard of the experts; output is this rank's
    partial sum, all-reduced across the EP group."""

    def __init__(self, full: MoEMLP, ep_rank: int, ep: int):
        super().__init__()
        n = full.num_experts
        assert n % ep == 0, "experts must divide ep"
        self.num_experts = n
        self.top_k = full.top_k
        self.ep_rank, self.ep = ep_rank, ep
        self.local_n = n // ep
        self.local_base = ep_rank * self.local_n
        self.gate = full.gate                       # replicated router
        self.experts = nn.ModuleList(
            full.experts[self.local_base:self.local_base + self.local_n])

    def forward(self, x):
        B, S, H = x.shape
        flat = x.reshape(-1, H)
        weights, idx = ops.moe_topk_router(self.gate(flat), self.top_k)
        weights = weights.to(x.dtype)
        out = torch.zeros_like(flat)
        for le, expert in enumerate(self.experts):
            e = self.local_base + le
            mask = (idx == e)
            tok, slot = mask.nonzero(as_tuple=True)
            if tok.numel() == 0:
                continue
            out.index_add_(0, tok, expert(flat[tok]) * weights[tok, slot,
                                                              None])
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(out)
        return out.reshape(B, S, H)


def build_ep_model(config_or_name, ep_rank: int, ep: int, device=None,
                   dtype=None, seed: int = 0, quantize: Optional[str] = None):
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
        layer.mlp = EPMoEMLP(layer.mlp, ep_rank, ep)
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
                 quantize: Optional[str] = None):
        self.rank, self.ep = rank, ep
        self.device = device if device is not None else device_for_rank()
        self.stage = build_ep_model(model, rank, ep, device=self.device,
                                    seed=seed, quantize=quantize)
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
