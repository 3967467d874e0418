"""Timing spans, metric reduction and the Tracking logger.

Parity targets: marked_timer spans, per-step metric
families (timing, throughput, losses), Tracking multi-backend logger
(console / tensorboard / jsonl).
"""
from __future__ import annotations

import json
import os
import time
from contextlib import contextmanager
from typing import Any, Dict, List

import torch


@contextmanager
def marked_timer(name: str, timing_raw: Dict[str, float]):
    """Accumulating wall-clock span: timing_raw[name] += elapsed.
    Also emits a roctx range so rocprofv3 marker traces group kernels by
    trainer phase (SURVEY.md §5.1 capability)."""
    import torch
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        torch.cuda.nvtx.range_push(f"polyrl/{name}")
    start = time.perf_counter()
    try:
        yield
    finally:
        timing_raw[name] = timing_raw.get(name, 0.0) + (time.perf_counter() - start)
        if on_gpu:
            torch.cuda.nvtx.range_pop()


def reduce_metrics(metrics: Dict[str, List[float]]) -> Dict[str, float]:
    out = {}
    for k, v in metrics.items():
        if isinstance(v, list) and v:
            out[k] = sum(float(x) for x in v) / len(v)
        elif isinstance(v, (int, float)):
            out[k] = float(v)
    return out


def compute_data_metrics(batch, use_critic: bool = False) -> Dict[str, float]:
    """Reward / advantage / length statistics for one global batch."""
    m: Dict[str, float] = {}
    t = batch.tensors
    if "token_level_scores" in t:
        seq_score = t["token_level_scores"].sum(-1)
        m["critic/score/mean"] = seq_score.mean().item()
        m["critic/score/max"] = seq_score.max().item()
        m["critic/score/min"] = seq_score.min().item()
    if "token_level_rewards" in t:
        seq_rew = t["token_level_rewards"].sum(-1)
        m["critic/rewards/mean"] = seq_rew.mean().item()
    if "advantages" in t and "response_mask" in t:
        mask = t["response_mask"].bool()
        adv = t["advantages"][mask]
        if adv.numel():
            m["critic/advantages/mean"] = adv.mean().item()
            m["critic/advantages/max"] = adv.max().item()
            m["critic/advantages/min"] = adv.min().item()
    if "returns" in t and "response_mask" in t:
        mask = t["response_mask"].bool()
        ret = t["returns"][mask]
        if ret.numel():
            m["critic/returns/mean"] = ret.mean().item()
    if "response_mask" in t:
        resp_len = t["response_mask"].sum(-1).float()
        m["response_length/mean"] = resp_len.mean().item()
        m["response_length/max"] = resp_len.max().item()
        m["response_length/min"] = resp_len.min().item()
    if "attention_mask" in t and "response_mask" in t:
        prompt_len = (t["attention_mask"].sum(-1) - t["response_mask"].sum(-1)).float()
        m["prompt_length/mean"] = prompt_len.mean().item()
    return m


def compute_timing_metrics(batch, timing_raw: Dict[str, float]) -> Dict[str, float]:
    m = {f"timing_s/{k}": v for k, v in timing_raw.items()}
    if "response_mask" in batch.tensors and "step" in timing_raw and timing_raw["step"] > 0:
        n_resp_tokens = batch.tensors["response_mask"].sum().item()
        m["perf/response_tokens_per_s"] = n_resp_tokens / timing_raw["step"]
        m["perf/samples_per_s"] = len(batch) / timing_raw["step"]
    return m


class FlopsCounter:
    """Dense-decoder FLOP estimates (the reference's verl FlopsCounter
    capability): attainable-vs-achieved MFU from model geometry."""

    # MI355X dense bf16 MFMA peak (AMD spec without 2:1 sparsity)
    MI355X_BF16_DENSE_PEAK = 2.5e15

    def __init__(self, model_cfg):
        c = model_cfg
        h = c.hidden_size
        self.per_token_params_flops = 2 * (
            c.num_hidden_layers * (
                h * (c.num_attention_heads + 2 * c.num_key_value_heads)
                * c.head_dim                      # qkv
                + c.num_attention_heads * c.head_dim * h   # o
                + 3 * h * c.intermediate_size)    # gate/up/down
            + h * c.vocab_size)                   # lm head
        self.cfg = c

    def forward_flops(self, total_tokens: int, avg_seqlen: float) -> float:
        attn = (2 * 2 * self.cfg.num_hidden_layers
                * self.cfg.num_attention_heads * self.cfg.head_dim
                * total_tokens * avg_seqlen)      # QK^T + PV (causal ~ /2)
        return self.per_token_params_flops * total_tokens + attn / 2

    def train_step_flops(self, total_tokens: int, avg_seqlen: float,
                         grad_ckpt: bool = False) -> float:
        mult = 4.0 if grad_ckpt else 3.0          # fwd + (recompute) + bwd
        return mult * self.forward_flops(total_tokens, avg_seqlen)

    def mfu(self, flops: float, seconds: float, n_gpus: int = 1,
            peak: float = MI355X_BF16_DENSE_PEAK) -> float:
        if seconds <= 0:
            return 0.0
        return flops / seconds / (peak * max(n_gpus, 1))


def compute_throughput_metrics(batch, timing_raw: Dict[str, float],
                               n_gpus: int,
                               model_cfg=None,
                               use_critic: bool = False) -> Dict[str, float]:
    m = {}
    if "attention_mask" in batch.tensors and timing_raw.get("step", 0) > 0:
        total_tokens = batch.tensors["attention_mask"].sum().item()
        m["perf/total_tokens"] = total_tokens
        m["perf/throughput_tokens_per_s_all_gpus"] = total_tokens / timing_raw["step"]
        m["perf/throughput_tokens_per_s_per_gpu"] = total_tokens / timing_raw["step"] / max(n_gpus, 1)
        if model_cfg is not None and timing_raw.get("update", 0) > 0:
            fc = FlopsCounter(model_cfg)
            L = batch.tensors["attention_mask"].shape[1]
            models = 2 if use_critic else 1
            flops = models * fc.train_step_flops(total_tokens, L)
            m["perf/update_tflops_per_gpu"] = \
                flops / timing_raw["update"] / max(n_gpus, 1) / 1e12
            m["perf/update_mfu"] = fc.mfu(flops, timing_raw["update"], n_gpus)
    return m


class Tracking:
    """Multi-backend metric logger: console, jsonl, tensorboard (if available)."""

    def __init__(self, project_name: str, experiment_name: str,
                 backends: List[str], default_local_dir: str = "."):
        self.backends = list(backends)
        self.project = project_name
        self.experiment = experiment_name
        self._tb = None
        self._jsonl = None
        log_dir = os.path.join(default_local_dir, "logs", experiment_name)
        if "tensorboard" in self.backends:
            try:
                from torch.utils.tensorboard import SummaryWriter
                os.makedirs(log_dir, exist_ok=True)
                self._tb = SummaryWriter(log_dir=log_dir)
            except Exception:
                self._tb = None
        if "jsonl" in self.backends:
            os.makedirs(log_dir, exist_ok=True)
            self._jsonl = open(os.path.join(log_dir, "metrics.jsonl"), "a")

    def log(self, data: Dict[str, Any], step: int):
        if "console" in self.backends:
            parts = " ".join(f"{k}:{v:.4g}" if isinstance(v, float) else f"{k}:{v}"
                             for k, v in sorted(data.items()))
            print(f"[step {step}] {parts}", flush=True)
        if self._tb is not None:
            for k, v in data.items():
                if isinstance(v, (int, float)):
                    self._tb.add_scalar(k, v, step)
        if self._jsonl is not None:
            self._jsonl.write(json.dumps({"step": step, **data}) + "\n")
            self._jsonl.flush()

    def close(self):
        if self._tb is not None:
            self._tb.close()
        if self._jsonl is not None:
            self._jsonl.close()

