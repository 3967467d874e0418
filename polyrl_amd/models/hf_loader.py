"""HF-checkpoint loading: our trainer models use HF-style parameter names
(model.layers.N.self_attn.q_proj.weight, ...), so Llama/Qwen safetensors
load with an identity name map (the reference's convert_weight_keys
HF-alignment capability, stream_fsdp_workers.py:236-238 — here the
alignment is by construction).

Call BEFORE FSDP sharding (plain nn.Module)."""
from __future__ import annotations

import glob
import os
from typing import List, Tuple

import torch


def load_hf_checkpoint(model: torch.nn.Module, path: str,
                       strict: bool = False) -> Tuple[List[str], List[str]]:
    """Load safetensors weights from a file or an HF model dir (sharded
    ``*.safetensors``) into ``model`` by name.  Returns (missing,
    unexpected).  Tied lm_head falls back to embed_tokens."""
    from safetensors.torch import load_file
    if os.path.isdir(path):
        files = sorted(glob.glob(os.path.join(path, "*.safetensors")))
        assert files, f"no *.safetensors under {path}"
    else:
        files = [path]
    target = model.state_dict()
    missing = set(target.keys())
    unexpected: List[str] = []
    with torch.no_grad():
        for f in files:
            for k, v in load_file(f).items():
                if k in target:
                    target[k].copy_(v.to(target[k].dtype))
                    missing.discard(k)
                else:
                    unexpected.append(k)
        if "lm_head.weight" in missing and \
                "model.embed_tokens.weight" in target and \
                "model.embed_tokens.weight" not in missing:
            target["lm_head.weight"].copy_(target["model.embed_tokens.weight"])
            missing.discard("lm_head.weight")
    if strict and (missing or unexpected):
        raise RuntimeError(f"HF load mismatch: missing={sorted(missing)} "
                           f"unexpected={unexpected}")
    return sorted(missing), unexpected
