"""Hybrid data-parallel × pipeline-parallel training.

The reference plumbs a ``dp_factor`` through its job requests but never
implements replication (``nodes/user_thread.py:139``, SURVEY.md §2.2 —
"Parameter only — NOT implemented"). Here DP is real: world = dp × pp,
contiguous pipeline groups (adjacent stages stay on adjacent xGMI
neighbors), and gradients are averaged across replicas with ONE RCCL
all-reduce over each rank's flat gradient buffer (the FusedAdamW flat
layout makes the whole stage a single bucket) right before the fused
optimizer step.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from tensorlink_amd.parallel.pipeline import PipelineTrainer
from tensorlink_amd.parallel.planner import StagePlan, plan_for_world


class HybridTrainer:
    """dp × pp trainer. SPMD: construct and call on every global rank."""

    def __init__(self, model, global_rank: int, world: int, dp: int = 1,
                 device=None, lr: float = 1e-4, **kw):
        assert world % dp == 0, "world must be divisible by dp"
        self.dp = dp
        self.pp = world // dp
        self.global_rank = global_rank
        self.dp_rank = global_rank // self.pp
        self.pp_rank = global_rank % self.pp

        # subgroups (every rank must construct every group)
        self.pp_group = None
        self.dp_group = None
        if dist.is_initialized() and world > 1:
            for d in range(dp):
                ranks = list(range(d * self.pp, (d + 1) * self.pp))
                g = dist.new_group(ranks) if self.pp > 1 else None
                if d == self.dp_rank:
                    self.pp_group = g
            for p in range(self.pp):
                ranks = list(range(p, world, self.pp))
                g = dist.new_group(ranks) if dp > 1 else None
                if p == self.pp_rank:
                    self.dp_group = g

        plan = model if isinstance(model, StagePlan) else \
            plan_for_world(model, self.pp, training=True)
        self.trainer = PipelineTrainer(
            plan, self.pp_rank, self.pp, device=device, lr=lr,
            group=self.pp_group, rank_base=self.dp_rank * self.pp, **kw)
        if dp > 1:
            self.trainer.grad_hook = self._sync_grads
        self.device = self.trainer.device

    def _sync_grads(self, trainer):
        flat = trainer.optimizer.flat_grad
        dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.dp_group)
        flat.div_(self.dp)

    def train_step(self, input_ids: Optional[torch.Tensor],
                   labels: Optional[torch.Tensor],
                   n_micro: Optional[int] = None) -> float:
        """input_ids/labels significant on GLOBAL rank 0: the global batch
        is sharded across replicas (rank 0 scatters to each replica's
        first pipeline rank)."""
        if self.dp > 1:
            # global shape agreement, then scatter shards to replica heads
            meta = None
            if self.global_rank == 0:
                meta = (tuple(input_ids.shape), n_micro)
            holder = [meta]
            dist.broadcast_object_list(holder, src=0)
            (B, S), n_micro = holder[0]
            assert B % self.dp == 0, "batch must divide dp"
            b = B // self.dp
            shard_ids = shard_labels = None
            # NCCL send/recv needs device tensors
            comm_dev = self.device if self.device.type == "cuda" else "cpu"
            if self.global_rank == 0:
                input_ids = input_ids.to(torch.int64)
                labels = labels.to(torch.int64)
                for d in range(1, self.dp):
                    dist.send(torch.stack(
                        [input_ids[d * b:(d + 1) * b],
                         labels[d * b:(d + 1) * b]]).contiguous().to(comm_dev),
                        d * self.pp)
                shard_ids = input_ids[:b]
                shard_labels = labels[:b]
            elif self.pp_rank == 0:
                buf = torch.empty(2, b, S, dtype=torch.int64, device=comm_dev)
                dist.recv(buf, 0)
                shard_ids, shard_labels = buf[0], buf[1]
            loss = self.trainer.train_step(shard_ids, shard_labels,
                                           n_micro=n_micro)
            # average reported loss across replicas
            t = torch.tensor([loss], dtype=torch.float64,
                             device=self.device if self.device.type == "cuda"
                             else "cpu")
            dist.all_reduce(t, group=self.dp_group)
            return float(t.item() / self.dp)
        return self.trainer.train_step(input_ids, labels, n_micro=n_micro)

    @property
    def optimizer(self):
        return self.trainer.optimizer
