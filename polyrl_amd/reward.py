"""Reward managers (reference capability: verl_stream/trainer/ppo/reward.py +
utils/reward_score dispatch, SURVEY.md §2.1 rows 'Reward loader/scores').

A reward manager maps a finished TensorBatch to token_level_scores
(B, resp_len) with the outcome score on the LAST valid response token.
Registry: constant | random | length | custom python fn from file.
"""
from __future__ import annotations

import hashlib
import importlib.util
from typing import Callable, Dict, Optional

import torch

from .protocol import TensorBatch


def _place_scores(batch: TensorBatch, scores: torch.Tensor) -> torch.Tensor:
    """scatter sequence-level scores onto last valid response token."""
    resp_mask = batch["response_mask"]
    B, Lr = resp_mask.shape
    out = torch.zeros(B, Lr, dtype=torch.float32)
    lens = resp_mask.sum(-1).long()
    for i in range(B):
        if lens[i] > 0:
            out[i, lens[i] - 1] = scores[i]
    return out


class ConstantReward:
    """BASELINE config #1: constant reward (plumbing tier)."""

    def __init__(self, value: float = 1.0, **_):
        self.value = value

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        return _place_scores(batch, torch.full((len(batch),), self.value))


class RandomReward:
    """Deterministic pseudo-random reward keyed by response content — gives
    GRPO non-degenerate group variance on synthetic data (bench realism)."""

    def __init__(self, seed: int = 0, **_):
        self.seed = seed

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        resp = batch["responses"]
        mask = batch["response_mask"]
        scores = torch.empty(len(batch))
        for i in range(len(batch)):
            ids = resp[i][mask[i].bool()].tolist()
            h = hashlib.md5(f"{self.seed}:{ids}".encode()).digest()
            scores[i] = (h[0] % 2)  # {0, 1}
        return _place_scores(batch, scores)


class LengthReward:
    """Reward proportional to response length (smoke-testing shaping)."""

    def __init__(self, target: Optional[int] = None, **_):
        self.target = target

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        lens = batch["response_mask"].sum(-1).float()
        if self.target:
            scores = 1.0 - (lens - self.target).abs() / self.target
        else:
            scores = lens / batch["response_mask"].shape[1]
        return _place_scores(batch, scores)


class FunctionReward:
    """Custom scoring function loaded from a python file
    (reference: custom_reward_function config)."""

    def __init__(self, path: str, name: str = "compute_score", **kw):
        spec = importlib.util.spec_from_file_location("custom_reward", path)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        self.fn: Callable = getattr(mod, name)
        self.kw = kw

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        scores = torch.tensor([
            float(self.fn(batch.slice(i), **self.kw)) for i in range(len(batch))
        ])
        return _place_scores(batch, scores)


class NaiveRewardManager:
    """Detokenize each response and score it against its ground truth with
    the data_source-dispatched rule scorer (reference: verl naive reward
    manager + default_compute_score, SURVEY.md §2.1 'Reward loader/scores').

    Needs ``tokenizer`` (anything with .decode(list[int]) -> str) and the
    batch's non_tensors to carry 'data_source' and 'ground_truth'."""

    def __init__(self, tokenizer=None, compute_score=None, **_):
        from .reward_score import default_compute_score
        self.tokenizer = tokenizer
        self.compute_score = compute_score or default_compute_score

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        assert self.tokenizer is not None, "naive reward needs a tokenizer"
        resp = batch["responses"]
        mask = batch["response_mask"]
        ds = batch.non_tensors.get("data_source")
        gt = batch.non_tensors.get("ground_truth")
        assert ds is not None and gt is not None, \
            "naive reward needs data_source + ground_truth in non_tensors"
        scores = torch.zeros(len(batch))
        for i in range(len(batch)):
            ids = resp[i][mask[i].bool()].tolist()
            text = self.tokenizer.decode(ids)
            try:
                scores[i] = float(self.compute_score(str(ds[i]), text,
                                                     str(gt[i])))
            except (KeyError, NotImplementedError):
                scores[i] = 0.0
        return _place_scores(batch, scores)


class PrimeRewardManager(NaiveRewardManager):
    """Parallel-verification manager (reference: verl's PrimeRewardManager,
    selected by reward_model.reward_manager='prime', reward.py:112-120):
    scores every sample concurrently in a thread pool — the scorers that
    matter here (code execution, sandboxed HTTP) block on I/O, so threads
    give real overlap; a scorer exception marks that sample 0."""

    def __init__(self, tokenizer=None, compute_score=None,
                 max_workers: int = 16, **_):
        super().__init__(tokenizer=tokenizer, compute_score=compute_score)
        self.max_workers = max_workers

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        from concurrent.futures import ThreadPoolExecutor
        assert self.tokenizer is not None, "prime reward needs a tokenizer"
        resp = batch["responses"]
        mask = batch["response_mask"]
        ds = batch.non_tensors.get("data_source")
        gt = batch.non_tensors.get("ground_truth")
        assert ds is not None and gt is not None

        def score_one(i: int) -> float:
            ids = resp[i][mask[i].bool()].tolist()
            text = self.tokenizer.decode(ids)
            try:
                return float(self.compute_score(str(ds[i]), text,
                                                str(gt[i])))
            except Exception:                      # noqa: BLE001
                return 0.0

        with ThreadPoolExecutor(max_workers=self.max_workers) as ex:
            scores = torch.tensor(list(ex.map(score_one,
                                              range(len(batch)))))
        return _place_scores(batch, scores)


class BatchRewardManager(NaiveRewardManager):
    """Batched-API manager (reference: verl's BatchRewardManager): calls
    compute_score ONCE with lists (data_sources, solution_strs,
    ground_truths) — for scorers that amortize over the batch (an RM
    forward, a batched sandbox call)."""

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        assert self.tokenizer is not None, "batch reward needs a tokenizer"
        resp = batch["responses"]
        mask = batch["response_mask"]
        ds = batch.non_tensors.get("data_source")
        gt = batch.non_tensors.get("ground_truth")
        assert ds is not None and gt is not None
        texts = [self.tokenizer.decode(resp[i][mask[i].bool()].tolist())
                 for i in range(len(batch))]
        try:
            res = self.compute_score([str(d) for d in ds], texts,
                                     [str(g) for g in gt])
            scores = torch.as_tensor([float(r) for r in res],
                                     dtype=torch.float32)
        except TypeError:
            # scorer has the scalar signature: fall back to a loop
            scores = torch.tensor([
                float(self.compute_score(str(ds[i]), texts[i], str(gt[i])))
                for i in range(len(batch))])
        return _place_scores(batch, scores)


class DAPORewardManager(NaiveRewardManager):
    """DAPO manager (reference: verl's DAPORewardManager, the style the
    math_dapo/aime recipes train with): rule score plus a soft overlong
    penalty — responses longer than (max_response_length -
    overlong_buffer_len) lose up to penalty_factor linearly with the
    overflow, discouraging truncation-length collapse."""

    def __init__(self, tokenizer=None, compute_score=None,
                 overlong_buffer_len: int = 0,
                 overlong_penalty_factor: float = 1.0,
                 max_response_length: Optional[int] = None, **_):
        super().__init__(tokenizer=tokenizer, compute_score=compute_score)
        self.overlong_buffer_len = overlong_buffer_len
        self.overlong_penalty_factor = overlong_penalty_factor
        self.max_response_length = max_response_length

    def __call__(self, batch: TensorBatch) -> torch.Tensor:
        out = super().__call__(batch)
        if not self.overlong_buffer_len:
            return out
        resp_mask = batch["response_mask"]
        max_len = self.max_response_length or resp_mask.shape[1]
        lens = resp_mask.sum(-1).long()
        expected = max_len - self.overlong_buffer_len
        for i in range(len(batch)):
            over = int(lens[i]) - expected
            if over > 0 and lens[i] > 0:
                pen = min(over / self.overlong_buffer_len, 1.0) \
                    * self.overlong_penalty_factor
                out[i, lens[i] - 1] -= pen
        return out


_REGISTRY: Dict[str, type] = {
    "constant": ConstantReward,
    "random": RandomReward,
    "length": LengthReward,
    "naive": NaiveRewardManager,
    "prime": PrimeRewardManager,
    "batch": BatchRewardManager,
    "dapo": DAPORewardManager,
}


def load_reward_manager(name: str = "constant", *,
                        sandbox_fusion_url: Optional[str] = None,
                        sandbox_max_concurrent: int = 64,
                        sandbox_memory_limit_mb: int = 1024,
                        **kwargs):
    """Build a reward manager by name (reference surface:
    trainer/ppo/reward.py:95-150 — naive/prime/batch/dapo registry +
    custom fn from file + sandbox-fusion url gated by a concurrency
    semaphore)."""
    if name == "custom":
        return FunctionReward(**kwargs)
    if name not in _REGISTRY:
        raise KeyError(f"unknown reward manager {name!r}: {sorted(_REGISTRY)}")
    if sandbox_fusion_url and "compute_score" not in kwargs:
        import functools
        import threading

        from .reward_score import default_compute_score
        sem = threading.Semaphore(sandbox_max_concurrent)
        kwargs["compute_score"] = functools.partial(
            default_compute_score, sandbox_fusion_url=sandbox_fusion_url,
            concurrent_semaphore=sem,
            memory_limit_mb=sandbox_memory_limit_mb)
    return _REGISTRY[name](**kwargs)


def load_reward_manager_from_config(cfg, tokenizer=None):
    """Reference precedence (trainer/ppo/reward.py:95-150): a custom
    reward function file wins; else the data_source dispatch, wrapped with
    the sandbox-fusion url + semaphore when configured; manager style from
    reward_model.reward_manager."""
    crf = getattr(cfg, "custom_reward_function", None)
    if crf is not None and crf.path:
        return FunctionReward(path=crf.path, name=crf.name)
    rm = cfg.reward_model
    sb = rm.sandbox_fusion
    kwargs = {}
    if rm.reward_manager == "dapo":
        kwargs.update(
            overlong_buffer_len=rm.overlong_buffer_len,
            overlong_penalty_factor=rm.overlong_penalty_factor,
            max_response_length=cfg.data.max_response_length)
    return load_reward_manager(
        rm.reward_manager, tokenizer=tokenizer,
        sandbox_fusion_url=sb.url,
        sandbox_max_concurrent=sb.max_concurrent,
        sandbox_memory_limit_mb=sb.memory_limit_mb, **kwargs)


def compute_reward(batch: TensorBatch, reward_fn) -> torch.Tensor:
    return reward_fn(batch)
