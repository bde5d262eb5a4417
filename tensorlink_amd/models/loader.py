"""Per-stage checkpoint loading + random init.

Keeps the reference's checkpoint story (HF layout: config.json +
safetensors shards) and its per-stage shard scan with prefix remap
``model.layers.N.* -> layers.i.*`` (reference ``ml/worker.py:542-638,
589-616``) — but loads from a *local* directory (no hub on this node) and
maps onto the native :class:`StageModel` parameter names.

Random init (`init_random_stage`) builds the synthetic-weight models used by
bench.py / smoke (BASELINE.json: "synthetic data / random-init weights").
"""

from __future__ import annotations

import json
import math
import os
from typing import Dict, Optional

import torch

from tensorlink_amd.models.configs import ModelConfig
from tensorlink_amd.models.dense import StageModel, build_stage


# HF parameter name -> (native name, row offset key), relative to one
# decoder layer. Offsets are symbolic ("q"/"kv"/"i") because the fused
# qkv/gate_up parameters pack several HF tensors as row slices.
_HF_LAYER_MAP = {
    "input_layernorm.weight": ("input_layernorm", None),
    "post_attention_layernorm.weight": ("post_attention_layernorm", None),
    "self_attn.q_proj.weight": ("self_attn.qkv_proj.weight", 0),
    "self_attn.q_proj.bias": ("self_attn.qkv_proj.bias", 0),
    "self_attn.k_proj.weight": ("self_attn.qkv_proj.weight", "q"),
    "self_attn.k_proj.bias": ("self_attn.qkv_proj.bias", "q"),
    "self_attn.v_proj.weight": ("self_attn.qkv_proj.weight", "q+kv"),
    "self_attn.v_proj.bias": ("self_attn.qkv_proj.bias", "q+kv"),
    "self_attn.o_proj.weight": ("self_attn.o_proj.weight", None),
    "self_attn.q_norm.weight": ("self_attn.q_norm", None),
    "self_attn.k_norm.weight": ("self_attn.k_norm", None),
    "mlp.gate_proj.weight": ("mlp.gate_up_proj.weight", 0),
    "mlp.up_proj.weight": ("mlp.gate_up_proj.weight", "i"),
    "mlp.down_proj.weight": ("mlp.down_proj.weight", None),
    # MoE routers: Mixtral / Qwen3-MoE
    "block_sparse_moe.gate.weight": ("mlp.gate.weight", None),
    "mlp.gate.weight": ("mlp.gate.weight", None),
}


def _resolve_offset(sym, config) -> int:
    if sym in (None, 0):
        return 0
    return {"q": config.q_size, "q+kv": config.q_size + config.kv_size,
            "i": config.intermediate_size,
            "ei": config.expert_intermediate_size}[sym]


def _map_hf_key(key: str, layer_start: int, layer_end: int, stage):
    """Map an HF checkpoint key to (native stage parameter name,
    row offset) or None if the key belongs to another stage."""
    if key.startswith("model.embed_tokens.weight"):
        if stage.has_embedding:
            return "embed_tokens.weight", 0
        if stage.has_head and stage.config.tie_word_embeddings:
            return "lm_head.weight", 0
        return None
    if key == "model.norm.weight":
        return ("norm", 0) if stage.has_head else None
    if key == "lm_head.weight":
        return ("lm_head.weight", 0) if (stage.has_head and
                                         hasattr(stage, "lm_head")) else None
    if key.startswith("model.layers."):
        rest = key[len("model.layers."):]
        idx_str, _, sub = rest.partition(".")
        idx = int(idx_str)
        if not (layer_start <= idx < layer_end):
            return None
        local = idx - layer_start
        if sub in _HF_LAYER_MAP:
            native, off = _HF_LAYER_MAP[sub]
            return (f"layers.{local}.{native}",
                    _resolve_offset(off, stage.config))
        # Mixtral experts: block_sparse_moe.experts.E.w1/w3/w2
        if sub.startswith("block_sparse_moe.experts."):
            parts = sub.split(".")
            e = parts[2]
            wname = parts[3]
            native, off = {
                "w1": ("gate_up_proj.weight", 0),
                "w3": ("gate_up_proj.weight", "ei"),
                "w2": ("down_proj.weight", None),
            }[wname]
            return (f"layers.{local}.mlp.experts.{e}.{native}",
                    _resolve_offset(off, stage.config))
        # Qwen3-MoE experts: mlp.experts.E.gate_proj/up_proj/down_proj
        if sub.startswith("mlp.experts."):
            parts = sub.split(".")
            e = parts[2]
            wname = parts[3]
            native, off = {
                "gate_proj": ("gate_up_proj.weight", 0),
                "up_proj": ("gate_up_proj.weight", "ei"),
                "down_proj": ("down_proj.weight", None),
            }[wname]
            return (f"layers.{local}.mlp.experts.{e}.{native}",
                    _resolve_offset(off, stage.config))
    return None


def load_stage_from_checkpoint(stage: StageModel, ckpt_dir: str,
                               device="cpu", dtype=torch.bfloat16) -> int:
    """Load only this stage's parameters from a local HF checkpoint dir.
    Returns the number of tensors loaded."""
    arch = stage.config.architecture
    if arch == "gpt2":
        from tensorlink_amd.models.gpt2 import load_gpt2_hf_weights
        n = load_gpt2_hf_weights(stage, ckpt_dir, dtype=dtype)
        stage.to(device=device, dtype=dtype)
        return n
    if arch == "neox":
        from tensorlink_amd.models.neox import load_neox_hf_weights
        n = load_neox_hf_weights(stage, ckpt_dir, dtype=dtype)
        stage.to(device=device, dtype=dtype)
        return n
    from safetensors import safe_open

    index_path = os.path.join(ckpt_dir, "model.safetensors.index.json")
    shards: Dict[str, list] = {}
    if os.path.exists(index_path):
        with open(index_path) as f:
            weight_map = json.load(f)["weight_map"]
        for key, shard in weight_map.items():
            shards.setdefault(shard, []).append(key)
    else:
        single = os.path.join(ckpt_dir, "model.safetensors")
        if not os.path.exists(single):
            raise FileNotFoundError(f"no safetensors in {ckpt_dir}")
        shards = {"model.safetensors": None}

    params = dict(stage.named_parameters())
    loaded = 0
    for shard, keys in shards.items():
        path = os.path.join(ckpt_dir, shard)
        with safe_open(path, framework="pt", device="cpu") as f:
            shard_keys = keys if keys is not None else f.keys()
            for key in shard_keys:
                mapped = _map_hf_key(key, stage.layer_start, stage.layer_end,
                                     stage)
                if mapped is None or mapped[0] not in params:
                    continue
                native, off = mapped
                t = f.get_tensor(key).to(dtype)
                with torch.no_grad():
                    params[native][off:off + t.shape[0]].copy_(t)
                loaded += 1
    stage.to(device=device, dtype=dtype)
    return loaded


@torch.no_grad()
def init_random_stage(stage: StageModel, device="cpu", dtype=torch.bfloat16,
                      seed: int = 0) -> StageModel:
    """Deterministic scaled-normal init directly on the target device."""
    gen_dev = device if str(device).startswith("cuda") else "cpu"
    g = torch.Generator(device=gen_dev)
    g.manual_seed(seed)
    stage.to(device=device, dtype=dtype)
    std = 1.0 / math.sqrt(stage.config.hidden_size)
    for name, p in stage.named_parameters():
        if ("layernorm" in name or name.endswith(("norm", "q_norm", "k_norm"))
                or (".ln_" in name and name.endswith(".weight"))
                or (name.startswith("ln_") and name.endswith(".weight"))):
            p.fill_(1.0)
        elif name.endswith(".bias"):
            p.zero_()
        else:
            tmp = torch.empty(p.shape, device=gen_dev, dtype=torch.float32)
            tmp.normal_(0.0, std, generator=g)
            p.copy_(tmp.to(dtype))
    return stage


def save_stage_to_safetensors(stage: StageModel, out_dir: str,
                              rank: int) -> str:
    """Checkpoint dump (reference parity: parameter retrieval →
    ``models/<name>/`` safetensors, ``ml/module.py:577-670``)."""
    from safetensors.torch import save_file
    os.makedirs(out_dir, exist_ok=True)
    path = os.path.join(out_dir, f"stage_{rank}.safetensors")
    state = {k: v.detach().cpu().contiguous()
             for k, v in stage.state_dict().items()}
    save_file(state, path)
    with open(os.path.join(out_dir, f"stage_{rank}.json"), "w") as f:
        json.dump({"layer_start": stage.layer_start,
                   "layer_end": stage.layer_end,
                   "has_embedding": stage.has_embedding,
                   "has_head": stage.has_head,
                   "config": stage.config.to_json()}, f)
    return path


@torch.no_grad()
def save_hf_checkpoint(stage: StageModel, out_dir: str) -> str:
    """Export a FULL model (single stage, PP=1) to the HuggingFace
    safetensors layout — the exact inverse of :func:`_map_hf_key`: fused
    qkv / gate_up rows are split back into q/k/v and gate/up, names get
    the ``model.layers.N.`` prefix, and a ``model.safetensors.index.json``
    is written so :func:`load_stage_from_checkpoint` (and HF tooling)
    can read the result. Completes the round trip the reference only
    half-has (parameter retrieval dumps raw per-module state dicts,
    ``ml/worker.py:1395-1413`` — not a loadable HF layout)."""
    from safetensors.torch import save_file
    assert stage.has_embedding and stage.has_head, \
        "export needs the full model (use the PP=1 stage or gather first)"
    assert type(stage).__name__ == "StageModel", \
        "HF export covers the llama/qwen/mixtral families (GPT-2 uses "\
        "the Conv1D layout; export for it is not implemented)"
    cfg = stage.config
    q, kv, i = cfg.q_size, cfg.kv_size, cfg.intermediate_size
    out = {}

    def put(key, t):
        out[key] = t.detach().cpu().contiguous()

    sd = stage.state_dict()
    put("model.embed_tokens.weight", sd["embed_tokens.weight"])
    put("model.norm.weight", sd["norm"])
    if not cfg.tie_word_embeddings and "lm_head.weight" in sd:
        put("lm_head.weight", sd["lm_head.weight"])
    for li in range(stage.layer_start, stage.layer_end):
        loc = li - stage.layer_start
        p = f"model.layers.{li}."
        n = f"layers.{loc}."
        put(p + "input_layernorm.weight", sd[n + "input_layernorm"])
        put(p + "post_attention_layernorm.weight",
            sd[n + "post_attention_layernorm"])
        w = sd[n + "self_attn.qkv_proj.weight"]
        put(p + "self_attn.q_proj.weight", w[:q])
        put(p + "self_attn.k_proj.weight", w[q:q + kv])
        put(p + "self_attn.v_proj.weight", w[q + kv:])
        bkey = n + "self_attn.qkv_proj.bias"
        if bkey in sd:
            b = sd[bkey]
            put(p + "self_attn.q_proj.bias", b[:q])
            put(p + "self_attn.k_proj.bias", b[q:q + kv])
            put(p + "self_attn.v_proj.bias", b[q + kv:])
        put(p + "self_attn.o_proj.weight", sd[n + "self_attn.o_proj.weight"])
        for extra in ("q_norm", "k_norm"):
            k2 = n + f"self_attn.{extra}"
            if k2 in sd:
                put(p + f"self_attn.{extra}.weight", sd[k2])
        if n + "mlp.gate_up_proj.weight" in sd:          # dense MLP
            gu = sd[n + "mlp.gate_up_proj.weight"]
            put(p + "mlp.gate_proj.weight", gu[:i])
            put(p + "mlp.up_proj.weight", gu[i:])
            put(p + "mlp.down_proj.weight", sd[n + "mlp.down_proj.weight"])
        else:                                            # MoE block
            ei = cfg.expert_intermediate_size
            mixtral = cfg.architecture == "mixtral"
            put(p + ("block_sparse_moe.gate.weight" if mixtral
                     else "mlp.gate.weight"), sd[n + "mlp.gate.weight"])
            e = 0
            while n + f"mlp.experts.{e}.gate_up_proj.weight" in sd:
                gu = sd[n + f"mlp.experts.{e}.gate_up_proj.weight"]
                dn = sd[n + f"mlp.experts.{e}.down_proj.weight"]
                if mixtral:
                    ep = p + f"block_sparse_moe.experts.{e}."
                    put(ep + "w1.weight", gu[:ei])
                    put(ep + "w3.weight", gu[ei:])
                    put(ep + "w2.weight", dn)
                else:
                    ep = p + f"mlp.experts.{e}."
                    put(ep + "gate_proj.weight", gu[:ei])
                    put(ep + "up_proj.weight", gu[ei:])
                    put(ep + "down_proj.weight", dn)
                e += 1

    os.makedirs(out_dir, exist_ok=True)
    shard = "model.safetensors"
    save_file(out, os.path.join(out_dir, shard))
    index = {"metadata": {"total_size": sum(t.numel() * t.element_size()
                                            for t in out.values())},
             "weight_map": {k: shard for k in out}}
    with open(os.path.join(out_dir, "model.safetensors.index.json"),
              "w") as f:
        json.dump(index, f)
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        f.write(stage.config.to_json())      # JSON string
    return out_dir


@torch.no_grad()
def load_stage_from_stage_ckpt(stage: StageModel, out_dir: str) -> int:
    """Re-partition-aware checkpoint load: assemble THIS stage's weights
    from a directory of per-stage files saved under a possibly DIFFERENT
    pipeline partitioning (elastic recovery after a world-size change —
    parallel/elastic.py). The sidecar ``stage_N.json`` gives each saved
    shard's global layer range; local layer indices are remapped through
    it. Returns the number of tensors loaded."""
    import glob

    from safetensors.torch import load_file
    target = dict(stage.state_dict())
    loaded = 0
    for meta_path in sorted(glob.glob(os.path.join(out_dir,
                                                   "stage_*.json"))):
        r = int(os.path.basename(meta_path)[6:-5])
        with open(meta_path) as f:
            meta = json.load(f)
        state = load_file(os.path.join(out_dir, f"stage_{r}.safetensors"))
        ls = meta["layer_start"]
        for k, v in state.items():
            if k.startswith("layers."):
                _, loc, rest = k.split(".", 2)
                gl = ls + int(loc)
                if not (stage.layer_start <= gl < stage.layer_end):
                    continue
                nk = f"layers.{gl - stage.layer_start}.{rest}"
            else:
                nk = k                  # embed / final norm / lm head
            if nk in target:
                target[nk].copy_(v.to(target[nk].dtype))
                loaded += 1
    return loaded


def load_stage_from_safetensors(stage: StageModel, out_dir: str,
                                rank: int) -> None:
    from safetensors.torch import load_file
    state = load_file(os.path.join(out_dir, f"stage_{rank}.safetensors"))
    stage.load_state_dict(state)
