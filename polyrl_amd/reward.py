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


_REGISTRY: Dict[str, type] = {
    "constant": ConstantReward,
    "random": RandomReward,
    "length": LengthReward,
    "naive": NaiveRewardManager,
}


def load_reward_manager(name: str = "constant", **kwargs):
    if name == "custom":
        return FunctionReward(**kwargs)
    if name not in _REGISTRY:
        raise KeyError(f"unknown reward manager {name!r}: {sorted(_REGISTRY)}")
    return _REGISTRY[name](**kwargs)


def compute_reward(batch: TensorBatch, reward_fn) -> torch.Tensor:
    return reward_fn(batch)
