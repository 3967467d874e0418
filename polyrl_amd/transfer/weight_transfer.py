"""Trainer -> rollout weight publication.

Reference capability: the whole rlboost/weight_transfer plane (sender/receiver
agents, shm staging, TCP streams, TP re-broadcast — SURVEY.md §3.3).  On a
single MI355X node none of that machinery is needed: trainer rank r and its
co-located engine share one device, so publication is per-parameter
FSDP all-gather (RCCL over xGMI) -> direct device copy into the engine's
fused weight buffers.  No host staging, no serialization.

Version gating mirrors handlers.rs:566-600: bump version, publish, instances
(engines) adopt the version atomically between generation steps — the engine
is never mid-forward during a publish because the SPMD trainer calls this
between engine steps (the in-process equivalent of SGLang's
model_update_lock.writer_lock contract, patches.py:482).
"""
from __future__ import annotations

import time
from typing import Dict, Optional

import torch
import torch.distributed as dist


class WeightPublisher:
    """Streams full params from an FSDP2-sharded trainer model into one or
    more engine-side consumers (InferenceModel.update_named)."""

    def __init__(self, model: torch.nn.Module, consumers: list,
                 tie_word_embeddings: bool = False):
        self.model = model
        self.consumers = list(consumers)
        self.version = 0
        self.tie = tie_word_embeddings
        self.last_publish_s = 0.0

    def add_consumer(self, c):
        self.consumers.append(c)

    @torch.no_grad()
    def publish(self, cpu_cache: Optional[Dict[str, torch.Tensor]] = None
                ) -> int:
        """All-gather each FSDP shard to a full tensor and copy into every
        consumer.  Per-parameter streaming keeps peak memory at one full
        param (cf. fsdp_interface.py:186-207 which materializes the whole
        state dict into a host buffer)."""
        t0 = time.perf_counter()
        self.version += 1
        from ..models.lora import LoRALinear, merged_state_dict
        if any(isinstance(m, LoRALinear) for m in self.model.modules()):
            sd = merged_state_dict(self.model)   # LoRA deltas folded in
        else:
            sd = self.model.state_dict()
        for name, param in sd.items():
            if hasattr(param, "full_tensor"):      # DTensor -> all-gather
                full = param.full_tensor()
            else:
                full = param
            for c in self.consumers:
                c.update_named(name, full)
            if cpu_cache is not None:
                # keep a host copy for async TCP pushes to elastic remotes
                cpu_cache[name] = full.detach().to("cpu", copy=True)
        if self.tie:
            for c in self.consumers:
                emb = sd.get("model.embed_tokens.weight")
                if emb is not None:
                    full = emb.full_tensor() if hasattr(emb, "full_tensor") else emb
                    c.update_named("lm_head.weight", full)
        for c in self.consumers:
            eng = getattr(c, "_engine_owner", None)
            if eng is not None:
                eng.flush_radix()   # cached KV prefixes are stale now
        if dist.is_available() and dist.is_initialized():
            dist.barrier()
        self.last_publish_s = time.perf_counter() - t0
        return self.version
