from tensorlink_amd.models.configs import ModelConfig, PRESETS, get_config  # noqa: F401
from tensorlink_amd.models.dense import (  # noqa: F401
    KVCache, StageModel, build_full_model, build_stage)
