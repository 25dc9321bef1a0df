"""Model registry: preset name -> (config, model). The trainer builds
any registered family through this single entry point; jobs select a
model by preset name in TOK_TRAINER_CONFIG."""
from __future__ import annotations

from torch_on_k8s_amd.models import gpt2, llama


def get_model_config(name: str, **overrides):
    if name in llama.PRESETS:
        return llama.get_config(name, **overrides)
    if name in gpt2.GPT2_PRESETS:
        cfg = gpt2.GPT2_PRESETS[name]
        if overrides:
            d = cfg.to_dict()
            d.update(overrides)
            cfg = gpt2.GPT2Config(**d)
        return cfg
    raise KeyError(f"unknown model preset: {name} "
                   f"(known: {sorted(list(llama.PRESETS) + list(gpt2.GPT2_PRESETS))})")


def build_model(cfg, activation_checkpointing: bool = False):
    if isinstance(cfg, llama.LlamaConfig):
        return llama.LlamaModel(
            cfg, activation_checkpointing=activation_checkpointing)
    if isinstance(cfg, gpt2.GPT2Config):
        return gpt2.GPT2Model(
            cfg, activation_checkpointing=activation_checkpointing)
    raise TypeError(f"unknown model config type: {type(cfg)}")
