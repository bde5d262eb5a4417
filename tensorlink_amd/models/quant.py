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

    def _kernel_tables(self):
        if getattr(self, "_ptr_key", None) != self.weight_fp8.data_ptr():
            dev = self.weight_fp8.device
            self._wp = torch.tensor([self.weight_fp8.data_ptr()],
                                    dtype=torch.int64, device=dev)
            self._sp = torch.tensor([self.scale.data_ptr()],
                                    dtype=torch.int64, device=dev)
            self._segs = {}
            self._ptr_key = self.weight_fp8.data_ptr()
        return self._wp, self._sp

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape[:-1]
        x2 = x.reshape(-1, self.in_features)
        from tensorlink_amd import ops
        if (x.is_cuda and x.dtype == torch.bfloat16
                and ops.extension_loaded()
                and x2.shape[0] <= 512
                and self.out_features % 64 == 0
                and self.in_features % 32 == 0):
            # serving-M rows only: the grouped kernel fills the chip by
            # column panels (N/64 blocks), fine for decode batches but
            # pathological at prefill M — big-M stays on _scaled_mm /
            # dequant below
            # weight-only fp8 through the grouped-GEMM kernel as a
            # single-expert group: in-kernel e4m3 dequant + epilogue
            # per-channel scales, hipGraph-capture-safe, bf16 x (more
            # accurate than _scaled_mm's fp8-quantized activations)
            M = x2.shape[0]
            wp, sp = self._kernel_tables()
            seg = self._segs.get(M)
            if seg is None:
                # per-M segment tensors persist (captured graphs read
                # them by address; no device sync on the hot path)
                seg = torch.tensor([0, M], dtype=torch.int32,
                                   device=x.device)
                self._segs[M] = seg
            out = ops._require_ext().moe_gemm(
                x2.contiguous(), None, seg, wp, sp,
                self.out_features, True)
            if self.bias is not None:
                out = out + self.bias
            return out.reshape(*shape, self.out_features)
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
        out = torch.nn.functional.linear(
            x2.float(), w,
            self.bias.float() if self.bias is not None else None)
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


# ---------------------------------------------------------------------------
# MXFP4 (OCP microscaling fp4): the gfx950-native 4-bit format
# ---------------------------------------------------------------------------
_E2M1_VALUES = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])


def quantize_mxfp4(w: torch.Tensor, group: int = 32):
    """Quantize [N, K] weights to MXFP4: groups of 32 along K share one
    power-of-two e8m0 scale; elements are e2m1 (±{0,.5,1,1.5,2,3,4,6}),
    packed two per byte. This is the exact operand format of gfx950's
    scaled-MFMA instructions (CDNA4 guide §MX), so round-2's 4-bit GEMM
    kernel consumes these tensors directly; until then Fp4Linear
    dequantizes on the fly. Returns (packed [N, K//2] uint8,
    exponents [N, K/group] int8)."""
    N, K = w.shape
    assert K % group == 0
    wf = w.float().reshape(N, K // group, group)
    amax = wf.abs().amax(-1).clamp(min=1e-12)
    # shared exponent: amax maps near the top code (6 = 1.5 * 2^2)
    e = torch.floor(torch.log2(amax)) - 2
    scale = torch.pow(2.0, e)
    x = (wf / scale.unsqueeze(-1)).clamp(-6, 6)
    vals = _E2M1_VALUES.to(w.device)
    idx = (x.abs().unsqueeze(-1) - vals).abs().argmin(-1).to(torch.uint8)
    code = idx | ((x < 0).to(torch.uint8) << 3)          # sign bit 3
    code = code.reshape(N, K)
    packed = (code[:, 0::2] | (code[:, 1::2] << 4)).contiguous()
    return packed, e.to(torch.int8)


def dequantize_mxfp4(packed: torch.Tensor, exponents: torch.Tensor,
                     group: int = 32, dtype=torch.float32):
    N = packed.shape[0]
    K = packed.shape[1] * 2
    code = torch.empty(N, K, dtype=torch.uint8, device=packed.device)
    code[:, 0::2] = packed & 0xF
    code[:, 1::2] = packed >> 4
    vals = _E2M1_VALUES.to(packed.device)
    mag = vals[(code & 0x7).long()]
    sign = torch.where((code & 0x8) != 0, -1.0, 1.0)
    scale = torch.pow(2.0, exponents.float()).repeat_interleave(group, -1)
    return (mag * sign * scale).to(dtype)


class Fp4Linear(nn.Module):
    """Weight-only MXFP4 linear: 4x smaller weights than bf16. Forward
    dequantizes per call (correct everywhere; the scaled-MFMA GEMM that
    consumes the packed form directly is round-2 kernel work —
    ROADMAP.md §6)."""

    def __init__(self, packed, exponents, bias, in_features, out_features):
        super().__init__()
        self.register_buffer("packed", packed)
        self.register_buffer("exponents", exponents)
        self.bias = bias
        self.in_features = in_features
        self.out_features = out_features

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "Fp4Linear":
        packed, e = quantize_mxfp4(lin.weight.detach())
        return cls(packed, e, lin.bias, lin.in_features, lin.out_features)

    def forward(self, x):
        w = dequantize_mxfp4(self.packed, self.exponents, dtype=torch.float32)
        y = torch.nn.functional.linear(x.float(), w,
                                       self.bias.float()
                                       if self.bias is not None else None)
        return y.to(x.dtype)


def quantize_dense_fp4(stage: nn.Module) -> int:
    """Convert every projection Linear to weight-only MXFP4 in place
    (4-bit analog of quantize_dense_fp8)."""
    n = 0
    targets = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj")
    for mod in stage.modules():
        for name in targets:
            lin = getattr(mod, name, None)
            if isinstance(lin, nn.Linear) and not isinstance(
                    lin, (Fp8Linear,)):
                setattr(mod, name, Fp4Linear.from_linear(lin))
                n += 1
    return n
