"""Computation-verification utilities.

Functional port of the reference's proof-of-learning skeleton
(``tensorlink/ml/proofs.py``: ``calculate_gradient_hash`` 6,
``verify_gradient_continuity`` 23, ``verify_loss_trajectory`` 41) — there
it was meant to check worker honesty; here it doubles as a debugging tool
for validating HIP kernels against reference numerics across steps.
"""

from __future__ import annotations

import hashlib
from typing import Dict, Iterable, List

import torch


def gradient_hash(grads: Iterable[torch.Tensor]) -> str:
    """Deterministic content hash of a set of gradients."""
    h = hashlib.sha256()
    for g in grads:
        if g is None:
            h.update(b"none")
            continue
        h.update(g.detach().to(torch.float32).cpu().numpy().tobytes())
    return h.hexdigest()


def model_gradient_hash(model: torch.nn.Module) -> str:
    return gradient_hash(p.grad for _, p in
                         sorted(model.named_parameters(), key=lambda x: x[0]))


def verify_gradient_continuity(prev: Dict[str, torch.Tensor],
                               cur: Dict[str, torch.Tensor],
                               max_drift: float = 0.9) -> bool:
    """Cosine-similarity drift check between consecutive gradient snapshots
    (reference: drift <= 0.1 on cosine distance, proofs.py:23-39)."""
    sims = []
    for name, g0 in prev.items():
        g1 = cur.get(name)
        if g1 is None or g0.numel() == 0:
            continue
        a = g0.flatten().float()
        b = g1.flatten().float()
        denom = a.norm() * b.norm()
        if denom > 0:
            sims.append(float((a @ b) / denom))
    if not sims:
        return True
    return sum(sims) / len(sims) >= -max_drift  # gradients may rotate, but
    # a persistent exact reversal indicates a sign/permutation bug


def verify_loss_trajectory(losses: List[float],
                           patience: int = 10) -> bool:
    """Sanity check that training loss trends down (proofs.py:41-53)."""
    if len(losses) < 2 * patience:
        return True
    first = sum(losses[:patience]) / patience
    last = sum(losses[-patience:]) / patience
    return last <= first * 1.05
