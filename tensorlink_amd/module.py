"""User-facing DistributedModel API.

Keeps the reference's public surface (``tensorlink/ml/module.py:237-1020``):
``DistributedModel(model, training=...)`` is an ``nn.Module`` whose
``forward`` returns logits, whose ``loss.backward()`` routes the gradient
back through the pipeline (reference ``CustomAutogradRouter``,
``ml/module.py:126-144``), plus ``generate``, ``create_optimizer``
(``ml/optim.py:81``), ``train``/``eval``, ``state_dict`` retrieval and
checkpoint saving (``ml/module.py:577-670``).

Execution backends:
- world_size == 1 (default): the whole model runs in-process on this
  rank's device (the reference's ``entire_model`` path).
- mode="local", world_size=N: N-1 local worker *processes* are spawned
  (torch.distributed gloo/nccl over 127.0.0.1) and this process becomes
  rank 0 — the MI355X collapse of the reference's User-parent /
  worker-network topology (``nodes/nodes.py:103-173``); BASELINE.json
  config #1 ("GPT-2-small DistributedModel on 2 local CPU worker procs")
  runs exactly this path.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn as nn

from tensorlink_amd.parallel.pipeline import (PipelineRunner,
                                              PipelineTrainer,
                                              SamplingParams)
from tensorlink_amd.parallel.planner import plan_for_world


def _cluster_worker(rank: int, world: int, port: int, model: str,
                    training: bool, init: str, ckpt_dir: Optional[str],
                    seed: int, lr: float):
    """Entry point for spawned worker processes (ranks 1..N-1)."""
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    from tensorlink_amd.parallel.comm import init_distributed
    init_distributed()
    plan = plan_for_world(model, world, training=training)
    if training:
        holder = PipelineTrainer(plan, rank, world, init=init,
                                 ckpt_dir=ckpt_dir, seed=seed, lr=lr)
        runner = holder.runner
    else:
        holder = None
        runner = PipelineRunner(plan, rank, world, init=init,
                                ckpt_dir=ckpt_dir, seed=seed)
    p2p = runner.p2p
    while True:
        cmd = p2p.broadcast_obj(None, src=0)
        op = cmd[0]
        if op == "shutdown":
            break
        elif op == "generate":
            runner.generate(None, None)
        elif op == "fwd_eval":
            runner.forward_logits(None)
        elif op == "forward":
            holder.spmd_forward(None)
        elif op == "backward":
            holder.spmd_backward(None)
        elif op == "train_step":
            holder.train_step(None, None, n_micro=cmd[1])
        elif op == "opt_step":
            holder.optimizer.step()
        elif op == "opt_zero":
            holder.optimizer.zero_grad()
        elif op == "save":
            from tensorlink_amd.models.loader import save_stage_to_safetensors
            save_stage_to_safetensors(runner.stage, cmd[1], rank)
    torch.distributed.destroy_process_group()


class _Router(torch.autograd.Function):
    """Routes loss.backward() on rank-0 logits into the pipeline backward
    (reference CustomAutogradRouter, ml/module.py:126-144)."""

    @staticmethod
    def forward(ctx, logits, model):
        ctx.model = model
        return logits

    @staticmethod
    def backward(ctx, grad):
        ctx.model._pipeline_backward(grad)
        return grad, None


class DistributedModel(nn.Module):
    def __new__(cls, model=None, *args, **kwargs):
        # reference trusted mode: DistributedModel(model=CustomModel(),
        # trusted=True) distributes an arbitrary nn.Module
        # (ml/module.py:259, docs/examples/EXAMPLES.md)
        if isinstance(model, nn.Module):
            from tensorlink_amd.custom import CustomDistributedModel
            return CustomDistributedModel(
                model, world_size=kwargs.get("world_size", 2),
                trusted=kwargs.get("trusted", False),
                lr=kwargs.get("lr", 1e-3))
        return super().__new__(cls)

    def __init__(self, model: str, training: bool = False,
                 world_size: int = 1, mode: str = "auto",
                 device=None, init: str = "random",
                 ckpt_dir: Optional[str] = None, seed: int = 0,
                 lr: float = 1e-4, n_pipelines: int = 1, tp: int = 1,
                 **trainer_kw):
        super().__init__()
        self.model_name = model
        self.training_mode = training
        self.world_size = world_size
        self.n_pipelines = n_pipelines
        self.tp = tp
        self._procs = []
        self._p2p = None
        self.device = (torch.device(device)
                       if device is not None else device)

        launched_world = int(os.environ.get("WORLD_SIZE", "1"))
        if world_size > 1 and (mode == "torchrun" or
                               (mode == "auto" and
                                launched_world == world_size)):
            # already launched SPMD (torchrun): join the existing group
            # instead of spawning local workers
            self._join_torchrun(model, training, world_size, init, ckpt_dir,
                                seed, lr)
        elif world_size > 1 and mode in ("auto", "local", "spawn"):
            self._start_cluster(model, training, world_size, init, ckpt_dir,
                                seed, lr)
        else:
            plan = plan_for_world(model, 1, training=training)
            if training:
                self._trainer = PipelineTrainer(plan, 0, 1, device=device,
                                                init=init, ckpt_dir=ckpt_dir,
                                                seed=seed, lr=lr,
                                                **trainer_kw)
                self._runner = self._trainer.runner
            else:
                self._trainer = None
                self._runner = PipelineRunner(plan, 0, 1, device=device,
                                              init=init, ckpt_dir=ckpt_dir,
                                              seed=seed)
        self.config = self._runner.config

    # ------------------------------------------------------------------
    def _join_torchrun(self, model, training, world, init, ckpt_dir, seed,
                       lr):
        """Under torchrun every rank constructs DistributedModel; rank 0
        is the user-facing instance, other ranks enter the worker loop
        via serve_worker() (call it after construction on rank != 0).

        ``tp > 1`` shards each pipeline stage tensor-parallel across the
        tp x (world//tp) grid (inference; random init — the grid slices
        the same seeded per-stage init a pure-PP run uses)."""
        from tensorlink_amd.parallel.comm import init_distributed
        rank, _ = init_distributed()
        self._rank = rank
        if self.tp > 1:
            assert not training, "tp>1 training: use parallel.tp.TPTrainer"
            from tensorlink_amd.parallel.tp import TPPPRunner
            grid = TPPPRunner(model, rank, world, self.tp,
                              device=self.device, seed=seed)
            self._trainer = None
            self._runner = grid.runner
            self._p2p = self._runner.p2p
            return
        plan = plan_for_world(model, world, training=training)
        if training:
            self._trainer = PipelineTrainer(plan, rank, world,
                                            device=self.device, init=init,
                                            ckpt_dir=ckpt_dir, seed=seed,
                                            lr=lr)
            self._runner = self._trainer.runner
        else:
            self._trainer = None
            self._runner = PipelineRunner(plan, rank, world,
                                          device=self.device, init=init,
                                          ckpt_dir=ckpt_dir, seed=seed)
        self._p2p = self._runner.p2p

    def serve_worker(self):
        """Worker loop for torchrun ranks != 0 (mirrors the spawned-
        cluster workers). tp>1: commands carry the inputs and travel over
        the WORLD group (every pipeline replica's first rank needs the
        prompt — its own pp-group broadcast source is itself)."""
        holder = self._trainer
        runner = self._runner
        if self.tp > 1:
            import torch.distributed as dist
            while True:
                box = [None]
                dist.broadcast_object_list(box, src=0)
                cmd = box[0]
                if cmd[0] == "shutdown":
                    return
                assert cmd[0] == "generate"
                ids = torch.tensor(cmd[1], dtype=torch.int64)
                runner.generate(ids if runner.is_first else None, cmd[2])
        p2p = runner.p2p
        while True:
            cmd = p2p.broadcast_obj(None, src=0)
            op = cmd[0]
            if op == "shutdown":
                return
            elif op == "generate":
                runner.generate(None, None)
            elif op == "fwd_eval":
                runner.forward_logits(None)
            elif op == "forward":
                holder.spmd_forward(None)
            elif op == "backward":
                holder.spmd_backward(None)
            elif op == "train_step":
                holder.train_step(None, None, n_micro=cmd[1])
            elif op == "opt_step":
                holder.optimizer.step()
            elif op == "opt_zero":
                holder.optimizer.zero_grad()
            elif op == "save":
                from tensorlink_amd.models.loader import \
                    save_stage_to_safetensors
                save_stage_to_safetensors(runner.stage, cmd[1],
                                          self._rank)

    def _start_cluster(self, model, training, world, init, ckpt_dir, seed,
                       lr):
        import torch.multiprocessing as mp
        port = 29000 + (os.getpid() % 1000)
        ctx = mp.get_context("spawn")
        for r in range(1, world):
            p = ctx.Process(target=_cluster_worker,
                            args=(r, world, port, model, training, init,
                                  ckpt_dir, seed, lr), daemon=True)
            p.start()
            self._procs.append(p)
        os.environ.update(RANK="0", WORLD_SIZE=str(world),
                          MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        from tensorlink_amd.parallel.comm import init_distributed
        init_distributed()
        plan = plan_for_world(model, world, training=training)
        if training:
            self._trainer = PipelineTrainer(plan, 0, world,
                                            device=self.device, init=init,
                                            ckpt_dir=ckpt_dir, seed=seed,
                                            lr=lr)
            self._runner = self._trainer.runner
        else:
            self._trainer = None
            self._runner = PipelineRunner(plan, 0, world, device=self.device,
                                          init=init, ckpt_dir=ckpt_dir,
                                          seed=seed)
        self._p2p = self._runner.p2p

    def _bcast(self, cmd):
        if self._p2p is not None:
            self._p2p.broadcast_obj(cmd, src=0)

    # ------------------------------------------------------------------
    def forward(self, input_ids: torch.Tensor, labels=None, **kw):
        """Returns logits [B,S,V]. With training=True the returned tensor
        participates in autograd: a user loss computed from it can be
        .backward()-ed (PP>1 routes through the pipeline). If labels are
        given, the shifted CE loss is attached as ``output.loss``."""
        if not self.training_mode:
            self._bcast(("fwd_eval",))
            return self._runner.forward_logits(input_ids)
        if self.world_size == 1:
            B, S = input_ids.shape
            pos = torch.arange(S).unsqueeze(0).expand(B, -1).contiguous()
            logits = self._runner.stage(input_ids.to(self._runner.device),
                                        pos.to(self._runner.device),
                                        training=True)
        else:
            self._bcast(("forward",))
            logits = self._trainer.spmd_forward(input_ids)
            logits = logits.detach().requires_grad_(True)
            logits = _Router.apply(logits, self)
        if labels is not None:
            from tensorlink_amd import ops
            loss = ops.causal_lm_loss(logits, labels.to(logits.device))
            # torch.Tensor supports instance attributes; fail loudly if a
            # subclass ever forbids it rather than silently dropping loss
            logits.loss = loss
        return logits

    def _pipeline_backward(self, grad_logits):
        self._bcast(("backward",))
        self._trainer.spmd_backward(grad_logits)

    def backward(self, loss: torch.Tensor):
        """Reference-parity explicit backward (ml/module.py:414)."""
        loss.backward()

    # ------------------------------------------------------------------
    def train_step(self, input_ids, labels, n_micro: Optional[int] = None
                   ) -> float:
        """1F1B micro-batched training step incl. optimizer step."""
        assert self.training_mode
        n_micro = n_micro or max(1, self.world_size)
        self._bcast(("train_step", n_micro))
        return self._trainer.train_step(input_ids, labels, n_micro=n_micro)

    @torch.no_grad()
    def generate(self, input_ids=None, max_new_tokens: int = 64,
                 temperature: float = 0.0, top_p: float = 1.0,
                 top_k: int = 0, do_sample: bool = False,
                 eos_token_id: Optional[int] = None,
                 seed: Optional[int] = None, **kw) -> torch.Tensor:
        if isinstance(input_ids, dict):
            input_ids = input_ids["input_ids"]
        sp = SamplingParams(
            temperature=temperature if do_sample else 0.0, top_p=top_p,
            top_k=top_k, max_new_tokens=max_new_tokens,
            eos_token_id=eos_token_id, seed=seed)
        if self.tp > 1 and self.world_size > 1:
            import torch.distributed as dist
            box = [("generate", input_ids.tolist(), sp)]
            dist.broadcast_object_list(box, src=0)
            out = self._runner.generate(input_ids, sp)
        else:
            self._bcast(("generate",))
            out = self._runner.generate(input_ids, sp,
                                        micro_batches=self.n_pipelines)
        return torch.cat([input_ids.to(out.device), out], dim=1)

    # ------------------------------------------------------------------
    def create_optimizer(self, **kwargs):
        """Reference parity: model.create_optimizer(lr=...) →
        DistributedOptimizer with step()/zero_grad()
        (ml/module.py:1019, ml/optim.py:81-203)."""
        assert self.training_mode, "create_optimizer requires training=True"
        lr = kwargs.pop("lr", None)
        if lr is not None:
            self._trainer.optimizer.lr = lr
        self.optimizer = _ClusterOptimizer(self)
        return self.optimizer

    def train(self, mode: bool = True):
        self._runner.stage.train(mode)
        return self

    def eval(self):
        return self.train(False)

    def parameters(self, recurse: bool = True):
        return self._runner.stage.parameters(recurse)

    def named_parameters(self, *a, **k):
        return self._runner.stage.named_parameters(*a, **k)

    def state_dict(self, *a, **k):
        """Rank-0 stage state (full model when world_size==1)."""
        return self._runner.stage.state_dict(*a, **k)

    def save_checkpoint(self, out_dir: str):
        """All ranks dump their stage as safetensors (reference parameter
        retrieval → models/<name>/, ml/module.py:577-670)."""
        from tensorlink_amd.models.loader import save_stage_to_safetensors
        self._bcast(("save", out_dir))
        save_stage_to_safetensors(self._runner.stage, out_dir, 0)
        return out_dir

    # ------------------------------------------------------------------
    def _shutdown_grid(self):
        import torch.distributed as dist
        box = [("shutdown",)]
        dist.broadcast_object_list(box, src=0)

    def shutdown(self):
        if self.tp > 1 and self.world_size > 1 and not self._procs \
                and torch.distributed.is_initialized() \
                and getattr(self, "_rank", 0) == 0:
            self._shutdown_grid()
            return
        if self._procs:
            self._bcast(("shutdown",))
            for p in self._procs:
                p.join(30)
                if p.is_alive():
                    p.terminate()
            self._procs = []
            if torch.distributed.is_initialized():
                torch.distributed.destroy_process_group()

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass


class _ClusterOptimizer:
    """step()/zero_grad() fan-out (reference DistributedOptimizer,
    ml/optim.py:85-203 — no ack polling: collectives are ordered)."""

    def __init__(self, model: DistributedModel):
        self._model = model

    def step(self):
        self._model._bcast(("opt_step",))
        self._model._trainer.optimizer.step()

    def zero_grad(self, set_to_none: bool = False):
        self._model._bcast(("opt_zero",))
        self._model._trainer.optimizer.zero_grad(set_to_none)
