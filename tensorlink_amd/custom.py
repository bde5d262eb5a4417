"""Custom-module pipeline sharding (the reference's "trusted mode").

The reference lets a user hand ``DistributedModel`` an arbitrary
``nn.Module`` and ships it to workers as a serialized module file
(``trusted=True`` — ``ml/module.py:259,771``; module transfer
``p2p/torch_node.py:879-924``; docs/examples/EXAMPLES.md "Upload and
distribute your model"). Here the same capability runs on one node: the
module's longest child chain (an ``nn.Sequential``, or its longest
``nn.ModuleList``, matching the reference parser's loop-module discovery
``ml/graphing.py:131``) is split into contiguous slices, one per spawned
worker process; activations and gradients move between ranks with
shape-prefixed sends, and a ``CustomAutogradRouter``-style autograd
Function stitches the remote slice chain into rank 0's graph so
``loss.backward()`` and ``optimizer.step()`` work like the reference's.

Scope: sequential layer chains (the reference's AST injector handles the
same family — models whose forward is "a loop over layers"). Dims may
change between layers; shapes are discovered at runtime.
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

_HDR = 9       # [ndim, d0..d7]


def send_tensor_any(t: torch.Tensor, dst: int):
    """Send a tensor whose shape the receiver does not know."""
    hdr = torch.zeros(_HDR, dtype=torch.int64)
    hdr[0] = t.dim()
    for i, d in enumerate(t.shape):
        hdr[1 + i] = d
    dist.send(hdr, dst)
    dist.send(t.contiguous().cpu(), dst)


def recv_tensor_any(src: int, dtype=torch.float32) -> torch.Tensor:
    hdr = torch.zeros(_HDR, dtype=torch.int64)
    dist.recv(hdr, src)
    shape = [int(hdr[1 + i]) for i in range(int(hdr[0]))]
    buf = torch.empty(shape, dtype=dtype)
    dist.recv(buf, src)
    return buf


def split_layer_chain(module: nn.Module, world: int):
    """Find the sliceable layer chain and split it into `world`
    contiguous stages (reference: `_group_sequential_layers`,
    ml/graphing.py:64)."""
    if isinstance(module, nn.Sequential):
        chain = list(module)
        pre, post = [], []
    else:
        lists = [(name, m) for name, m in module.named_children()
                 if isinstance(m, (nn.ModuleList, nn.Sequential))]
        if not lists:
            raise ValueError(
                "custom module needs an nn.Sequential (or a child "
                "ModuleList layer chain) to shard")
        name, chain_mod = max(lists, key=lambda kv: len(kv[1]))
        chain = list(chain_mod)
        kids = list(module.named_children())
        idx = [k for k, (n, _) in enumerate(kids) if n == name][0]
        pre = [m for _, m in kids[:idx]]
        post = [m for _, m in kids[idx + 1:]]
    n = len(chain)
    assert n >= world, f"{n} layers cannot split {world} ways"
    per = [n // world + (1 if i < n % world else 0) for i in range(world)]
    stages, start = [], 0
    for i, p in enumerate(per):
        mods = chain[start:start + p]
        if i == 0:
            mods = pre + mods
        if i == world - 1:
            mods = mods + post
        stages.append(nn.Sequential(*mods))
        start += p
    return stages


class _RemoteChain(torch.autograd.Function):
    """Stitches ranks 1..N-1 into rank 0's autograd graph (the
    reference's CustomAutogradRouter, ml/module.py:126)."""

    @staticmethod
    def forward(ctx, x, pipe):
        ctx.pipe = pipe
        # grad mode is disabled inside Function.forward — the engine's
        # needs_input_grad tells us whether a backward will follow
        pipe._bcast(("fwd", bool(ctx.needs_input_grad[0])))
        send_tensor_any(x.detach(), 1)
        return recv_tensor_any(pipe.world - 1)

    @staticmethod
    def backward(ctx, dy):
        pipe = ctx.pipe
        pipe._bcast(("bwd",))
        send_tensor_any(dy.contiguous(), pipe.world - 1)
        dx = recv_tensor_any(1)
        return dx, None


def _custom_worker(rank, world, port, model, lr_default):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    stage = split_layer_chain(model, world)[rank]
    opt = None
    stash = []
    while True:
        box = [None]
        dist.broadcast_object_list(box, src=0)
        cmd = box[0]
        op = cmd[0]
        if op == "shutdown":
            dist.destroy_process_group()
            return
        elif op == "fwd":
            x = recv_tensor_any(rank - 1).requires_grad_(cmd[1])
            with torch.enable_grad() if cmd[1] else torch.no_grad():
                y = stage(x)
            if cmd[1]:
                stash.append((x, y))
            send_tensor_any(y.detach(),
                            (rank + 1) if rank + 1 < world else 0)
        elif op == "bwd":
            x, y = stash.pop()
            dy = recv_tensor_any((rank + 1) if rank + 1 < world else 0)
            y.backward(dy)
            send_tensor_any(x.grad, rank - 1 if rank > 1 else 0)
        elif op == "opt_init":
            opt = torch.optim.AdamW(stage.parameters(), lr=cmd[1])
        elif op == "opt_step":
            opt.step()
        elif op == "opt_zero":
            opt.zero_grad()
        elif op == "train":
            stage.train(cmd[1])


class CustomDistributedModel(nn.Module):
    """Pipeline-shard an arbitrary user module across local worker
    processes (reference trusted mode). Rank 0 (this process) holds the
    first slice; `forward` routes through the remote chain and returns
    an autograd-connected output, so ``loss.backward()`` and the
    created optimizer behave exactly like the reference API."""

    def __init__(self, model: nn.Module, world_size: int = 2,
                 trusted: bool = False, lr: float = 1e-3,
                 port: Optional[int] = None):
        super().__init__()
        if not trusted:
            raise ValueError(
                "distributing an arbitrary module executes its code in "
                "worker processes — pass trusted=True to confirm "
                "(reference ml/module.py:259 prompts the same way)")
        assert world_size >= 2, "world_size>=2 (1 needs no sharding)"
        self.world = world_size
        self.stage = split_layer_chain(model, world_size)[0]
        self._opt = None
        port = port or (29600 + os.getpid() % 200)
        ctx = torch.multiprocessing.get_context("spawn")
        self._procs = [
            ctx.Process(target=_custom_worker,
                        args=(r, world_size, port, model, lr), daemon=True)
            for r in range(1, world_size)]
        for p in self._procs:
            p.start()
        os.environ.update(RANK="0", WORLD_SIZE=str(world_size),
                          MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        if not dist.is_initialized():
            dist.init_process_group("gloo", rank=0, world_size=world_size)

    def _bcast(self, cmd):
        dist.broadcast_object_list([cmd], src=0)

    def forward(self, x):
        y0 = self.stage(x)
        return _RemoteChain.apply(y0, self)

    def train(self, mode: bool = True):
        super().train(mode)
        self._bcast(("train", mode))
        return self

    def create_optimizer(self, lr: float = 1e-3, **kw):
        self._bcast(("opt_init", lr))
        self._opt = torch.optim.AdamW(self.stage.parameters(), lr=lr, **kw)
        return _CustomOptimizer(self)

    def shutdown(self):
        if self._procs:
            self._bcast(("shutdown",))
            for p in self._procs:
                p.join(30)
                if p.is_alive():
                    p.terminate()
            self._procs = []
            if dist.is_initialized():
                dist.destroy_process_group()

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass


class _CustomOptimizer:
    def __init__(self, pipe: CustomDistributedModel):
        self.pipe = pipe

    def step(self):
        self.pipe._bcast(("opt_step",))
        self.pipe._opt.step()

    def zero_grad(self, set_to_none: bool = False):
        self.pipe._bcast(("opt_zero",))
        self.pipe._opt.zero_grad(set_to_none)
