"""Model registry: arch name -> constructor."""
from __future__ import annotations

from dynamo_amd.engine.config import ModelConfig, PRESETS


def build_model(cfg: ModelConfig, device, dtype, tp=None, seed: int = 0):
    if cfg.arch == "llama" or cfg.arch == "qwen2":
        from .llama import LlamaForCausalLM
        return LlamaForCausalLM(cfg, device, dtype, tp, seed)
    if cfg.arch == "opt":
        from .opt import OPTForCausalLM
        return OPTForCausalLM(cfg, device, dtype, tp, seed)
    if cfg.arch == "mixtral":
        from .mixtral import MixtralForCausalLM
        return MixtralForCausalLM(cfg, device, dtype, tp, seed)
    raise ValueError(f"unknown arch {cfg.arch}")


def resolve_model_config(name_or_cfg) -> ModelConfig:
    if isinstance(name_or_cfg, ModelConfig):
        return name_or_cfg
    if isinstance(name_or_cfg, dict):
        return ModelConfig.from_dict(name_or_cfg)
    if name_or_cfg in PRESETS:
        return PRESETS[name_or_cfg]
    import os
    if os.path.isdir(name_or_cfg) and os.path.exists(
            os.path.join(name_or_cfg, "config.json")):
        # local HF checkpoint dir: architecture from config.json, weights
        # loaded after model build (models/loader.py)
        from dynamo_amd.models.loader import config_from_hf
        import dataclasses
        cfg = config_from_hf(name_or_cfg)
        return dataclasses.replace(cfg, weights_path=name_or_cfg)
    raise ValueError(f"unknown model preset {name_or_cfg!r}; "
                     f"known: {sorted(PRESETS)} (or a local HF checkpoint "
                     f"directory)")
