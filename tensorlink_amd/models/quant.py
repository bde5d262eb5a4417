"""FP8 (OCP e4m3fn) weight quantization for expert GEMMs.

BASELINE config #5: "Mixtral 8x7B MoE with expert groups split across
stages, fp8 weights on CDNA4 MFMA". gfx950's fp8 is OCP e4m3fn (NOT the
MI300X fnuz variant — guide §4). Weights are stored fp8 with per-output-
channel fp32 scales (absmax); the GEMM runs on the matrix cores through
torch._scaled_mm (hipBLASLt fp8 path) with dynamic per-tensor activation
scaling, falling back to dequant+bf16 GEMM where _scaled_mm is
unavailable (CPU tests).

Memory: an 8-expert Mixtral layer's expert weights drop 2x (1.4 GB ->
0.7 GB per layer), which is what lets wider expert groups sit per stage.
"""

from __future__ import annotations

import torch
import torch.nn as nn

FP8_DTYPE = torch.float8_e4m3fn
FP8_MAX = 448.0


def quantize_fp8_per_channel(w: torch.Tensor):
    """w [out, in] -> (w_fp8 [out, in], scale [out] fp32) with absmax
    per-output-channel scaling."""
    absmax = w.abs().amax(dim=1).float().clamp(min=1e-12)
    scale = absmax / FP8_MAX
    w_fp8 = (w.float() / scale[:, None]).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    return w_fp8, scale


class Fp8Linear(nn.Module):
    """Linear with fp8 weight storage + per-channel scales."""

    def __init__(self, weight_fp8: torch.Tensor, scale: torch.Tensor,
                 bias=None):
        super().__init__()
        self.register_buffer("weight_fp8", weight_fp8)
        self.register_buffer("scale", scale)
        self.bias = bias
        self.out_features, self.in_features = weight_fp8.shape

    @classmethod
    def from_linear(cls, linear: nn.Linear) -> "Fp8Linear":
        w_fp8, scale = quantize_fp8_per_channel(linear.weight.detach())
        return cls(w_fp8, scale,
                   linear.bias.detach() if linear.bias is not None else None)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape[:-1]
        x2 = x.reshape(-1, self.in_features)
        if x.is_cuda and hasattr(torch, "_scaled_mm"):
            try:
                # dynamic per-tensor activation scale
                amax = x2.abs().amax().float().clamp(min=1e-12)
                x_scale = (amax / FP8_MAX)
                x_fp8 = (x2.float() / x_scale).clamp(
                    -FP8_MAX, FP8_MAX).to(FP8_DTYPE)
                out = torch._scaled_mm(
                    x_fp8, self.weight_fp8.t(),
                    scale_a=x_scale.reshape(1, 1),
                    scale_b=self.scale.reshape(1, -1),
                    out_dtype=x.dtype)
                if self.bias is not None:
                    out = out + self.bias
                return out.reshape(*shape, self.out_features)
            except Exception:
                pass
        # dequant fallback (CPU tier / missing fp8 GEMM support)
        w = self.weight_fp8.float() * self.scale[:, None]
        out = torch.nn.functional.linear(x2.float(), w, self.bias)
        return out.to(x.dtype).reshape(*shape, self.out_features)


def quantize_experts_fp8(stage: nn.Module) -> int:
    """Convert every MoE expert Linear in a stage to Fp8Linear in place.
    Returns the number of converted layers."""
    from tensorlink_amd.models.dense import MoEMLP
    n = 0
    for mod in stage.modules():
        if isinstance(mod, MoEMLP):
            for expert in mod.experts:
                for name in ("gate_up_proj", "down_proj"):
                    lin = getattr(expert, name)
                    if isinstance(lin, nn.Linear):
                        setattr(expert, name, Fp8Linear.from_linear(lin))
                        n += 1
    return n


def quantize_dense_fp8(stage: nn.Module) -> int:
    """Convert every projection Linear (qkv/o/gate_up/down — attention
    and MLP alike) to weight-only fp8 in place: ~2x weight-memory and
    HBM-stream reduction for the decode-bound serving path (dense-model
    analog of the Mixtral expert quantization in BASELINE config #5;
    embeddings, norms and the LM head stay bf16). Returns the number of
    converted layers. hipGraph capture is disabled by the runner for any
    fp8 mode (torch._scaled_mm is not capture-safe on ROCm 7.2)."""
    n = 0
    targets = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj")
    for mod in stage.modules():
        for name in targets:
            lin = getattr(mod, name, None)
            if isinstance(lin, nn.Linear) and not isinstance(lin,
                                                             Fp8Linear):
                setattr(mod, name, Fp8Linear.from_linear(lin))
                n += 1
    return n
