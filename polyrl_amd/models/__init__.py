"""Model factory."""
from __future__ import annotations

import torch

from .registry import MODEL_CONFIGS, DecoderConfig, get_model_config


def create_model(name_or_cfg, kind: str = "actor", dtype: str = "bfloat16",
                 device: str = "cpu"):
    """kind: actor | critic.  Random-init weights (no network access)."""
    cfg = (get_model_config(name_or_cfg) if isinstance(name_or_cfg, str)
           else name_or_cfg)
    td = getattr(torch, dtype) if isinstance(dtype, str) else dtype
    # construct directly on the target device: CPU-side random init of an
    # 8B model costs ~30 s per model; on-device init is seconds
    with torch.device(device):
        if cfg.arch in ("llama", "qwen2"):
            from .llama import CausalLM, CausalLMWithValueHead
            model = CausalLM(cfg) if kind == "actor" \
                else CausalLMWithValueHead(cfg)
        elif cfg.arch == "gpt2":
            from .gpt2 import GPT2LMModel, GPT2WithValueHead
            model = GPT2LMModel(cfg) if kind == "actor" \
                else GPT2WithValueHead(cfg)
        else:
            raise ValueError(f"unknown arch {cfg.arch!r}")
    return model.to(device=device, dtype=td)


__all__ = ["MODEL_CONFIGS", "DecoderConfig", "get_model_config", "create_model"]
