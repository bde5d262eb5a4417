"""Model configurations for the native model zoo.

The reference delegates all model structure to HuggingFace ``AutoConfig`` /
``AutoModelForCausalLM`` (reference ``tensorlink/ml/utils.py:890``,
``tensorlink/ml/worker.py:1122``).  The MI355X build owns its model
definitions, so configs are explicit dataclasses; ``from_hf_config`` maps a
HuggingFace ``config.json`` (dict or transformers object) onto them so the
same checkpoints/skeletons remain loadable.
"""

from __future__ import annotations

import dataclasses
import json
import os
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class ModelConfig:
    name: str = "custom"
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    head_dim: Optional[int] = None          # defaults to hidden/heads
    max_position_embeddings: int = 4096
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-6
    tie_word_embeddings: bool = False
    attention_bias: bool = False            # Qwen2 uses qkv bias
    qkv_bias: bool = False
    hidden_act: str = "silu"
    dtype: str = "bfloat16"
    # MoE; num_local_experts == 0 => dense MLP. expert_intermediate_size
    # defaults to intermediate_size (Mixtral); Qwen3-MoE uses a smaller
    # per-expert width (moe_intermediate_size)
    num_local_experts: int = 0
    num_experts_per_tok: int = 2
    expert_intermediate_size: Optional[int] = None
    # HF rope_scaling dict (Llama-3.1 "llama3" frequency scaling)
    rope_scaling: Optional[dict] = None
    # GPT-NeoX: fraction of each head rotated by RoPE; parallel residual
    rotary_pct: float = 1.0
    use_parallel_residual: bool = True
    # architecture tag for checkpoint key mapping
    architecture: str = "llama"

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads
        if self.expert_intermediate_size is None:
            self.expert_intermediate_size = self.intermediate_size

    @property
    def is_moe(self) -> bool:
        return self.num_local_experts > 0

    @property
    def gated_mlp(self) -> bool:
        """SwiGLU-style 3-matrix MLP (llama/qwen/mixtral) vs the plain
        2-matrix GELU MLP (gpt2, neox)."""
        return self.architecture not in ("gpt2", "neox")

    @property
    def q_size(self) -> int:
        return self.num_attention_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_key_value_heads * self.head_dim

    def param_count(self, include_embeddings: bool = True) -> int:
        """Exact parameter count for this architecture (used by the planner)."""
        h, hd = self.hidden_size, self.head_dim
        q, kv = self.q_size, self.kv_size
        attn = h * q + 2 * h * kv + q * h
        if self.qkv_bias:
            attn += q + 2 * kv
        if self.is_moe:
            mlp = (self.num_local_experts * 3 * h
                   * self.expert_intermediate_size)
            mlp += h * self.num_local_experts  # router
        else:
            mlp = (3 if self.gated_mlp else 2) * h * self.intermediate_size
        norms = 2 * h
        per_layer = attn + mlp + norms
        total = self.num_hidden_layers * per_layer + h  # final norm
        if include_embeddings:
            total += self.vocab_size * h
            if not self.tie_word_embeddings:
                total += self.vocab_size * h
        return total

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self))

    @classmethod
    def from_json(cls, s: str) -> "ModelConfig":
        return cls(**json.loads(s))

    @classmethod
    def from_hf_config(cls, cfg, name: str = "custom") -> "ModelConfig":
        """Build from a HF config object / dict (see reference
        ``ml/utils.py:890-916`` which uses AutoConfig the same way)."""
        if not isinstance(cfg, dict):
            cfg = {k: getattr(cfg, k) for k in dir(cfg) if not k.startswith("_")
                   if not callable(getattr(cfg, k, None))}
        if "architecture" in cfg and "architectures" not in cfg:
            # our own to_json round-trip (save_hf_checkpoint writes it)
            known = {f.name for f in dataclasses.fields(cls)}
            return cls(**{k: v for k, v in cfg.items() if k in known})
        archs = cfg.get("architectures") or []
        # llama-SHAPED architectures map onto the llama family; anything
        # not recognized raises loudly instead of silently mis-mapping
        # (a Gemma/Phi config would otherwise run with wrong structure)
        _known = ("Llama", "Mistral", "Qwen", "Mixtral", "GPTNeoX",
                  "GPT2", "LlamaForCausalLM")
        if archs and not any(k in a for a in archs for k in _known):
            raise KeyError(
                f"unsupported architecture(s) {archs}: the native zoo "
                "covers Llama/Mistral-shaped, Qwen2, Qwen3(-MoE), "
                "Mixtral, GPT-2 and GPT-NeoX/Pythia families")
        arch = "llama"
        if any("Qwen" in a for a in archs):
            arch = "qwen2"
        if any("Qwen3" in a for a in archs):
            arch = "qwen3"
        if any("Mixtral" in a for a in archs):
            arch = "mixtral"
        if any("GPTNeoX" in a for a in archs):
            arch = "neox"
        get = cfg.get
        return cls(
            name=name,
            vocab_size=get("vocab_size", 32000),
            hidden_size=get("hidden_size", 4096),
            intermediate_size=get("intermediate_size", 11008),
            num_hidden_layers=get("num_hidden_layers", 32),
            num_attention_heads=get("num_attention_heads", 32),
            num_key_value_heads=get("num_key_value_heads",
                                    get("num_attention_heads", 32)),
            head_dim=get("head_dim", None),
            max_position_embeddings=get("max_position_embeddings", 4096),
            rope_theta=get("rope_theta", 10000.0),
            rms_norm_eps=get("rms_norm_eps",
                             get("layer_norm_eps", 1e-6)),
            tie_word_embeddings=get("tie_word_embeddings", False),
            qkv_bias=(arch == "qwen2" and "Qwen3" not in str(archs)),
            num_local_experts=(get("num_local_experts", 0)
                               or get("num_experts", 0) or 0),
            num_experts_per_tok=get("num_experts_per_tok", 2) or 2,
            expert_intermediate_size=get("moe_intermediate_size", None),
            rope_scaling=get("rope_scaling", None),
            rotary_pct=get("rotary_pct", 1.0) or 1.0,
            use_parallel_residual=get("use_parallel_residual", True),
            architecture=arch,
        )


def _qwen25_7b() -> ModelConfig:
    return ModelConfig(
        name="Qwen/Qwen2.5-7B-Instruct", vocab_size=152064, hidden_size=3584,
        intermediate_size=18944, num_hidden_layers=28, num_attention_heads=28,
        num_key_value_heads=4, max_position_embeddings=32768,
        rope_theta=1000000.0, rms_norm_eps=1e-6, tie_word_embeddings=False,
        qkv_bias=True, architecture="qwen2")


def _qwen3_8b() -> ModelConfig:
    return ModelConfig(
        name="Qwen/Qwen3-8B", vocab_size=151936, hidden_size=4096,
        intermediate_size=12288, num_hidden_layers=36, num_attention_heads=32,
        num_key_value_heads=8, head_dim=128, max_position_embeddings=40960,
        rope_theta=1000000.0, rms_norm_eps=1e-6, tie_word_embeddings=False,
        qkv_bias=False, architecture="qwen3")


def _llama3_70b() -> ModelConfig:
    return ModelConfig(
        name="meta-llama/Llama-3-70B", vocab_size=128256, hidden_size=8192,
        intermediate_size=28672, num_hidden_layers=80, num_attention_heads=64,
        num_key_value_heads=8, max_position_embeddings=8192,
        rope_theta=500000.0, rms_norm_eps=1e-5, tie_word_embeddings=False,
        architecture="llama")


def _llama3_8b() -> ModelConfig:
    return ModelConfig(
        name="meta-llama/Llama-3-8B", vocab_size=128256, hidden_size=4096,
        intermediate_size=14336, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, max_position_embeddings=8192,
        rope_theta=500000.0, rms_norm_eps=1e-5, architecture="llama")


def _mixtral_8x7b() -> ModelConfig:
    return ModelConfig(
        name="mistralai/Mixtral-8x7B-v0.1", vocab_size=32000, hidden_size=4096,
        intermediate_size=14336, num_hidden_layers=32, num_attention_heads=32,
        num_key_value_heads=8, max_position_embeddings=32768,
        rope_theta=1000000.0, rms_norm_eps=1e-5, num_local_experts=8,
        num_experts_per_tok=2, architecture="mixtral")


def _tiny() -> ModelConfig:
    # CI-scale stand-in for sshleifer/tiny-gpt2-class smoke models
    # (reference tests/test_distributed_model.py:24).
    return ModelConfig(
        name="tiny", vocab_size=1024, hidden_size=256, intermediate_size=512,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=512, rope_theta=10000.0, architecture="llama")


def _tiny_qwen3() -> ModelConfig:
    # exercises the qk-norm (Qwen3) attention path at kernel-supported D=64
    return ModelConfig(
        name="tiny-qwen3", vocab_size=1024, hidden_size=256,
        intermediate_size=512, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, head_dim=64, max_position_embeddings=512,
        rope_theta=10000.0, architecture="qwen3")


def _tiny_moe() -> ModelConfig:
    return ModelConfig(
        name="tiny-moe", vocab_size=1024, hidden_size=256,
        intermediate_size=384, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, max_position_embeddings=512,
        num_local_experts=4, num_experts_per_tok=2, architecture="mixtral")


def _gpt2(name):
    def make():
        from tensorlink_amd.models.gpt2 import gpt2_config
        return gpt2_config(name)
    return make


def _tiny_qwen3_moe() -> ModelConfig:
    return ModelConfig(
        name="tiny-qwen3-moe", vocab_size=1024, hidden_size=256,
        intermediate_size=768, num_hidden_layers=4,
        num_attention_heads=4, num_key_value_heads=2, head_dim=64,
        max_position_embeddings=4096, num_local_experts=4,
        num_experts_per_tok=2, expert_intermediate_size=192,
        architecture="qwen3")


def _qwen3_30b_a3b() -> ModelConfig:
    # Qwen3-30B-A3B: 128 experts, 8 active, per-expert width 768
    return ModelConfig(
        name="Qwen/Qwen3-30B-A3B", vocab_size=151936, hidden_size=2048,
        intermediate_size=6144, num_hidden_layers=48,
        num_attention_heads=32, num_key_value_heads=4, head_dim=128,
        max_position_embeddings=40960, rope_theta=1000000.0,
        num_local_experts=128, num_experts_per_tok=8,
        expert_intermediate_size=768, architecture="qwen3")


def _tiny_bigvocab() -> ModelConfig:
    # tiny body with a >=32k vocabulary: exercises the chunked-CE loss
    # path the real models take (vocab gate in PipelineTrainer._loss)
    return ModelConfig(
        name="tiny-bigvocab", vocab_size=38400, hidden_size=256,
        intermediate_size=512, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, head_dim=64,
        max_position_embeddings=512)


def _tiny_neox() -> ModelConfig:
    # LayerNorm + parallel residual + partial rotary + ungated GELU MLP
    return ModelConfig(
        name="tiny-neox", vocab_size=1024, hidden_size=256,
        intermediate_size=1024, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=4, head_dim=64, max_position_embeddings=512,
        rotary_pct=0.25, rms_norm_eps=1e-5, architecture="neox")


def _pythia_28b() -> ModelConfig:
    return ModelConfig(
        name="EleutherAI/pythia-2.8b", vocab_size=50304, hidden_size=2560,
        intermediate_size=10240, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=32, head_dim=80,
        max_position_embeddings=2048, rotary_pct=0.25, rms_norm_eps=1e-5,
        architecture="neox")


def _pythia_69b() -> ModelConfig:
    return ModelConfig(
        name="EleutherAI/pythia-6.9b", vocab_size=50432, hidden_size=4096,
        intermediate_size=16384, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=32, head_dim=128,
        max_position_embeddings=2048, rotary_pct=0.25, rms_norm_eps=1e-5,
        architecture="neox")


def _llama31_8b() -> ModelConfig:
    return ModelConfig(
        name="meta-llama/Llama-3.1-8B", vocab_size=128256,
        hidden_size=4096, intermediate_size=14336, num_hidden_layers=32,
        num_attention_heads=32, num_key_value_heads=8, head_dim=128,
        max_position_embeddings=131072, rope_theta=500000.0,
        rms_norm_eps=1e-5, rope_scaling={
            "rope_type": "llama3", "factor": 8.0,
            "low_freq_factor": 1.0, "high_freq_factor": 4.0,
            "original_max_position_embeddings": 8192})


PRESETS = {
    "gpt2-small": _gpt2("gpt2-small"),
    "gpt2": _gpt2("gpt2-small"),
    "gpt2-medium": _gpt2("gpt2-medium"),
    "tiny-gpt2": _gpt2("tiny-gpt2"),
    "sshleifer/tiny-gpt2": _gpt2("tiny-gpt2"),
    "Qwen/Qwen2.5-7B-Instruct": _qwen25_7b,
    "Qwen/Qwen2.5-7B": _qwen25_7b,
    "Qwen/Qwen3-8B": _qwen3_8b,
    "meta-llama/Llama-3-70B": _llama3_70b,
    "meta-llama/Llama-3-8B": _llama3_8b,
    "mistralai/Mixtral-8x7B-v0.1": _mixtral_8x7b,
    "tiny": _tiny,
    "tiny-qwen3": _tiny_qwen3,
    "tiny-moe": _tiny_moe,
    "tiny-qwen3-moe": _tiny_qwen3_moe,
    "tiny-bigvocab": _tiny_bigvocab,
    "tiny-neox": _tiny_neox,
    "EleutherAI/pythia-2.8b": _pythia_28b,
    "EleutherAI/pythia-6.9b": _pythia_69b,
    "meta-llama/Llama-3.1-8B": _llama31_8b,
    "Qwen/Qwen3-30B-A3B": _qwen3_30b_a3b,
}


def get_config(name: str) -> ModelConfig:
    """Resolve a model name to a config: preset table, then a local HF
    checkpoint directory with a config.json (no hub access on this node)."""
    if name in PRESETS:
        return PRESETS[name]()
    if os.path.isdir(name) and os.path.exists(os.path.join(name, "config.json")):
        with open(os.path.join(name, "config.json")) as f:
            return ModelConfig.from_hf_config(json.load(f), name=name)
    raise KeyError(
        f"unknown model {name!r}: not a preset and no local checkpoint dir "
        f"(no network access for hub downloads)")
