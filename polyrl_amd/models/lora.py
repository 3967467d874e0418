"""Minimal LoRA for the trainer models (reference capability: verl's LoRA
path — peft adapters on the FSDP policy, collected/merged for the rollout
weight plane, stream_fsdp_workers.py:221-238).

Design: LoRALinear wraps an existing nn.Linear (base frozen); publication
merges W + (alpha/r) B A into the dense HF-named weight so the rollout
engine never needs adapter-awareness.
"""
from __future__ import annotations

import math
from typing import Iterable, List

import torch
import torch.nn as nn

DEFAULT_TARGETS = ("q_proj", "k_proj", "v_proj", "o_proj",
                   "gate_proj", "up_proj", "down_proj")


class LoRALinear(nn.Module):
    def __init__(self, base: nn.Linear, r: int, alpha: float = 16.0,
                 dropout: float = 0.0):
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad_(False)
        self.r = r
        self.scaling = alpha / r
        dt = base.weight.dtype
        dev = base.weight.device
        self.lora_A = nn.Parameter(
            torch.empty(r, base.in_features, dtype=dt, device=dev))
        self.lora_B = nn.Parameter(
            torch.zeros(base.out_features, r, dtype=dt, device=dev))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()

    def forward(self, x):
        y = self.base(x)
        return y + (self.dropout(x) @ self.lora_A.t()) @ self.lora_B.t() \
            * self.scaling

    @torch.no_grad()
    def merged_weight(self) -> torch.Tensor:
        """Dense W + scale * B A (full tensors gathered if FSDP-sharded)."""
        def full(t):
            return t.full_tensor() if hasattr(t, "full_tensor") else t
        w = full(self.base.weight).float()
        a = full(self.lora_A).float()
        b = full(self.lora_B).float()
        return (w + b @ a * self.scaling).to(self.base.weight.dtype)


def apply_lora(model: nn.Module, r: int, alpha: float = 16.0,
               targets: Iterable[str] = DEFAULT_TARGETS,
               dropout: float = 0.0) -> List[str]:
    """Wrap matching Linears in place; freezes everything except adapters
    (and keeps value/lm heads trainable if present).  Returns wrapped paths.
    Call BEFORE FSDP sharding."""
    wrapped = []
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if isinstance(child, nn.Linear) and child_name in targets:
                setattr(module, child_name,
                        LoRALinear(child, r, alpha, dropout))
                wrapped.append(f"{name}.{child_name}" if name else child_name)
    # freeze non-adapter, non-head parameters
    for pname, p in model.named_parameters():
        if "lora_A" in pname or "lora_B" in pname:
            p.requires_grad_(True)
        elif "value_head" in pname:
            p.requires_grad_(True)
        else:
            p.requires_grad_(False)
    return wrapped


def merged_state_dict(model: nn.Module) -> dict:
    """HF-named dense state dict with LoRA deltas merged — what the weight
    plane publishes to the rollout engine (the reference's LoRA
    collect-and-convert capability)."""
    out = {}
    lora_mods = {}
    for name, module in model.named_modules():
        if isinstance(module, LoRALinear):
            lora_mods[name] = module
    sd = model.state_dict()
    for k, v in sd.items():
        if ".lora_A" in k or ".lora_B" in k:
            continue
        # base weights of adapted modules live under '<mod>.base.weight'
        if k.endswith(".base.weight"):
            mod = k[: -len(".base.weight")]
            if mod in lora_mods:
                out[mod + ".weight"] = lora_mods[mod].merged_weight()
                continue
        if k.endswith(".base.bias"):
            out[k.replace(".base.bias", ".bias")] = v
            continue
        out[k] = v
    return out
