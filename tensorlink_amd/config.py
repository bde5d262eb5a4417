"""Layered configuration (reference ``bin/config.json`` +
``bin/run_node.py:25-96``): JSON file -> environment -> defaults, with the
same node/ml section shapes minus the chain/crypto sections (no blockchain
in this build)."""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import List, Optional

DEFAULT_MODELS = ["Qwen/Qwen2.5-7B-Instruct"]   # reference config/models.json
                                                # defaults to Qwen3-8B


@dataclass
class NodeConfig:
    endpoint: bool = True
    endpoint_host: str = "127.0.0.1"
    endpoint_port: int = 8000
    logging: str = "info"


@dataclass
class MLConfig:
    max_memory_gb: Optional[float] = None     # cap per-rank memory (tests)
    default_models: List[str] = field(default_factory=lambda: list(DEFAULT_MODELS))
    init: str = "random"                      # random | checkpoint
    checkpoint_root: Optional[str] = None
    dtype: str = "bfloat16"
    # reference ml.trusted (bin/config.json): allow distributing
    # user-supplied nn.Modules (module.py trusted mode)
    trusted: bool = False
    # serving defaults applied by `tlctl serve` when flags are absent
    continuous: bool = False
    prefix_caching: bool = False
    prefill_chunk: Optional[int] = None
    job_ttl_s: Optional[float] = None         # idle-job eviction


@dataclass
class EngineConfig:
    node: NodeConfig = field(default_factory=NodeConfig)
    ml: MLConfig = field(default_factory=MLConfig)

    @classmethod
    def load(cls, path: Optional[str] = None) -> "EngineConfig":
        cfg = cls()
        path = path or os.environ.get("TL_CONFIG", "config.json")
        if path and os.path.exists(path):
            with open(path) as f:
                data = json.load(f)
            for k, v in data.get("node", {}).items():
                if hasattr(cfg.node, k):
                    setattr(cfg.node, k, v)
            for k, v in data.get("ml", {}).items():
                if hasattr(cfg.ml, k):
                    setattr(cfg.ml, k, v)
        if os.environ.get("TL_ENDPOINT_PORT"):
            cfg.node.endpoint_port = int(os.environ["TL_ENDPOINT_PORT"])
        return cfg
