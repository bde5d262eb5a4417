"""LoRA adapters: low-rank fine-tuning + merge-for-serving.

Beyond the reference's surface (it fine-tunes full weights through its
distributed optimizer only). LoRA trains two small matrices per
projection (W + (alpha/r)·B@A with A [r,in], B [out,r]) while the base
weights stay frozen — the optimizer state shrinks by orders of
magnitude and a trained adapter merges back into the base weight for
zero-overhead serving.

Works on any stage from the zoo: `apply_lora` wraps the projection
Linears in place (TLLinear included, so the skinny-GEMM dispatch still
serves the frozen base weight), `merge_lora` folds adapters into the
weights and restores plain Linears, `lora_state_dict`/`load_lora_state`
round-trip just the adapter tensors.
"""

from __future__ import annotations

import math
from typing import Dict, Iterable, List, Optional

import torch
import torch.nn as nn

# llama/qwen/mixtral projections + the GPT-2 Conv1D-derived names
DEFAULT_TARGETS = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj",
                   "c_attn", "c_proj", "c_fc")


class LoRALinear(nn.Module):
    def __init__(self, base: nn.Linear, r: int = 8, alpha: float = 16.0,
                 dropout: float = 0.0):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad_(False)
        dt = base.weight.dtype
        dev = base.weight.device
        self.lora_A = nn.Parameter(torch.empty(r, base.in_features,
                                               dtype=dt, device=dev))
        self.lora_B = nn.Parameter(torch.zeros(base.out_features, r,
                                               dtype=dt, device=dev))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.scaling = alpha / r
        self.r, self.alpha = r, alpha
        self.dropout = nn.Dropout(dropout) if dropout > 0 else None

    @property
    def weight(self):                     # shape introspection compat
        return self.base.weight

    @property
    def bias(self):
        return self.base.bias

    @property
    def in_features(self):
        return self.base.in_features

    @property
    def out_features(self):
        return self.base.out_features

    def forward(self, x):
        y = self.base(x)
        h = self.dropout(x) if self.dropout is not None else x
        return y + (h @ self.lora_A.t() @ self.lora_B.t()) * self.scaling

    def merge_(self) -> nn.Linear:
        """Fold the adapter into the base weight; returns the base."""
        with torch.no_grad():
            self.base.weight += (self.lora_B @ self.lora_A) * self.scaling
        return self.base


def apply_lora(stage: nn.Module, r: int = 8, alpha: float = 16.0,
               targets: Iterable[str] = DEFAULT_TARGETS,
               dropout: float = 0.0) -> int:
    """Wrap every target projection in place; freezes everything else.
    Returns the number of wrapped layers."""
    for p in stage.parameters():
        p.requires_grad_(False)
    n = 0
    for mod in stage.modules():
        for name in targets:
            lin = getattr(mod, name, None)
            if isinstance(lin, nn.Linear) and not isinstance(lin,
                                                             LoRALinear):
                setattr(mod, name, LoRALinear(lin, r=r, alpha=alpha,
                                              dropout=dropout))
                n += 1
    return n


def merge_lora(stage: nn.Module) -> int:
    """Merge all adapters into their base weights (serving form)."""
    n = 0
    for mod in stage.modules():
        for name, child in list(mod.named_children()):
            if isinstance(child, LoRALinear):
                setattr(mod, name, child.merge_())
                n += 1
    return n


def lora_parameters(stage: nn.Module) -> List[nn.Parameter]:
    return [p for m in stage.modules() if isinstance(m, LoRALinear)
            for p in (m.lora_A, m.lora_B)]


def lora_state_dict(stage: nn.Module) -> Dict[str, torch.Tensor]:
    out = {}
    for name, m in stage.named_modules():
        if isinstance(m, LoRALinear):
            out[f"{name}.lora_A"] = m.lora_A.detach().cpu()
            out[f"{name}.lora_B"] = m.lora_B.detach().cpu()
    return out


def load_lora_state(stage: nn.Module, state: Dict[str, torch.Tensor]):
    mods = dict(stage.named_modules())
    with torch.no_grad():
        for key, t in state.items():
            mod_name, which = key.rsplit(".", 1)
            m = mods[mod_name]
            getattr(m, which).copy_(t.to(getattr(m, which).dtype))
