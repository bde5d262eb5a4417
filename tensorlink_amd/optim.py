"""Optimizers: fused-AdamW per stage + the user-facing DistributedOptimizer.

Reference parity: ``tensorlink/ml/optim.py`` wraps a torch optimizer class
and fans ``step``/``zero_grad`` out to workers over IPC+TCP with ack polling
(``optim.py:131-203``). Here every rank owns a :class:`FusedAdamW` whose
step is ONE HIP kernel launch over a flat parameter buffer
(``ops/csrc/adamw.hip``), and :class:`DistributedOptimizer` is a thin SPMD
wrapper that keeps the reference's ``step()/zero_grad()`` call signature.
"""

from __future__ import annotations

from typing import Iterable, List

import torch

from tensorlink_amd import ops


class FusedAdamW:
    """AdamW over a flattened parameter set: one kernel launch per step.

    Re-points each parameter's storage into a single contiguous buffer and
    pre-assigns ``.grad`` views into a flat gradient buffer, so autograd
    accumulates directly into the flat buffer.
    """

    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-4,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.01):
        self.params: List[torch.nn.Parameter] = [p for p in params
                                                 if p.requires_grad]
        if not self.params:
            raise ValueError("no trainable parameters")
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0

        dev = self.params[0].device
        dt = self.params[0].dtype
        assert all(p.dtype == dt and p.device == dev for p in self.params), \
            "FusedAdamW requires uniform dtype/device"
        total = sum(p.numel() for p in self.params)
        self.flat_param = torch.empty(total, device=dev, dtype=dt)
        self.flat_grad = torch.zeros(total, device=dev, dtype=dt)
        self.exp_avg = torch.zeros(total, device=dev, dtype=torch.float32)
        self.exp_avg_sq = torch.zeros(total, device=dev, dtype=torch.float32)
        off = 0
        with torch.no_grad():
            for p in self.params:
                n = p.numel()
                self.flat_param[off:off + n].copy_(p.reshape(-1))
                p.data = self.flat_param[off:off + n].view(p.shape)
                p.grad = self.flat_grad[off:off + n].view(p.shape)
                off += n

    @torch.no_grad()
    def step(self):
        self.step_count += 1
        ops.adamw_(self.flat_param, self.flat_grad, self.exp_avg,
                   self.exp_avg_sq, lr=self.lr, beta1=self.beta1,
                   beta2=self.beta2, eps=self.eps,
                   weight_decay=self.weight_decay, step=self.step_count)

    @torch.no_grad()
    def zero_grad(self, set_to_none: bool = False):
        # grads are views into the flat buffer — zero it and restore views
        # (autograd may have replaced .grad if set_to_none was used)
        self.flat_grad.zero_()
        off = 0
        for p in self.params:
            n = p.numel()
            if p.grad is None or p.grad.data_ptr() != \
                    self.flat_grad[off:off + n].data_ptr():
                p.grad = self.flat_grad[off:off + n].view(p.shape)
            off += n

    @torch.no_grad()
    def grad_norm_sq(self) -> torch.Tensor:
        """Squared L2 norm of the flat gradient (one reduction kernel;
        callers all-reduce it across ranks for the global norm)."""
        return self.flat_grad.float().pow(2).sum()

    @torch.no_grad()
    def scale_grads(self, scale: float) -> None:
        self.flat_grad.mul_(scale)

    def clip_grad_norm_(self, max_norm: float,
                        norm_sq: "torch.Tensor | None" = None) -> float:
        """Clip by global norm over the flat buffer (semantics of
        torch.nn.utils.clip_grad_norm_). Pass an all-reduced norm_sq for
        the cross-rank global norm in pipeline training."""
        ns = self.grad_norm_sq() if norm_sq is None else norm_sq
        total = float(ns.sqrt())
        if total > max_norm:
            self.scale_grads(max_norm / (total + 1e-6))
        return total

    def state_dict(self):
        return {"step": self.step_count, "exp_avg": self.exp_avg,
                "exp_avg_sq": self.exp_avg_sq, "lr": self.lr}

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])


class DistributedOptimizer:
    """User-facing optimizer for a DistributedModel (reference
    ``ml/optim.py:85-203`` — same public surface, no IPC/acks: SPMD calls).
    """

    def __init__(self, trainer, lr: float = 1e-4, **kwargs):
        self._trainer = trainer
        self.inner = FusedAdamW(trainer.stage.parameters(), lr=lr, **kwargs)

    def step(self):
        self.inner.step()

    def zero_grad(self, set_to_none: bool = False):
        self.inner.zero_grad(set_to_none)


class WarmupCosineLR:
    """Linear warmup then cosine decay to min_lr — the standard LLM
    schedule (the reference exposes raw optimizer kwargs only; a
    scheduler is this build's addition for the training path)."""

    def __init__(self, optimizer, max_lr: float, warmup_steps: int,
                 total_steps: int, min_lr: float = 0.0):
        self.opt = optimizer
        self.max_lr = max_lr
        self.warmup = max(1, warmup_steps)
        self.total = total_steps
        self.min_lr = min_lr
        self.t = 0

    def step(self) -> float:
        import math
        self.t += 1
        if self.t <= self.warmup:
            lr = self.max_lr * self.t / self.warmup
        else:
            frac = min(1.0, (self.t - self.warmup)
                       / max(1, self.total - self.warmup))
            lr = self.min_lr + 0.5 * (self.max_lr - self.min_lr) * (
                1 + math.cos(math.pi * frac))
        self.opt.lr = lr
        return lr

    def state_dict(self):
        return {"t": self.t}

    def load_state_dict(self, sd):
        self.t = sd["t"]
