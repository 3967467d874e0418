"""Datasets: synthetic prompts (bench/tests, no network) and parquet RLHF data.

Synthetic prompts are deterministic across ranks (seeded), left-padded to
max_prompt_length — the layout the rollout coordinator expects.
"""
from __future__ import annotations

import hashlib
from typing import Iterator, List, Optional

import numpy as np
import torch

from .protocol import TensorBatch


class SyntheticPromptDataset:
    """Random-token prompts of varying length (BASELINE: synthetic data)."""

    def __init__(self, num_prompts: int, vocab_size: int,
                 max_prompt_length: int, min_prompt_length: Optional[int] = None,
                 seed: int = 0, pad_token_id: int = 0):
        self.num_prompts = num_prompts
        self.vocab_size = vocab_size
        self.Lp = max_prompt_length
        self.Lmin = min_prompt_length or max(max_prompt_length // 2, 1)
        self.pad = pad_token_id
        g = torch.Generator().manual_seed(seed)
        self._lens = torch.randint(self.Lmin, self.Lp + 1, (num_prompts,),
                                   generator=g)
        # one flat pool of tokens, sliced per prompt (cheap, deterministic)
        self._tokens = torch.randint(1, vocab_size, (num_prompts, self.Lp),
                                     generator=g)

    def __len__(self):
        return self.num_prompts

    def __getitem__(self, i: int) -> dict:
        L = int(self._lens[i])
        ids = torch.full((self.Lp,), self.pad, dtype=torch.long)
        mask = torch.zeros(self.Lp, dtype=torch.long)
        ids[self.Lp - L:] = self._tokens[i, :L]       # left-padded
        mask[self.Lp - L:] = 1
        return {"input_ids": ids, "attention_mask": mask, "uid": f"p{i}"}

    def batch(self, indices: List[int]) -> TensorBatch:
        rows = [self[i] for i in indices]
        return TensorBatch.from_dict(
            tensors={
                "input_ids": torch.stack([r["input_ids"] for r in rows]),
                "attention_mask": torch.stack([r["attention_mask"] for r in rows]),
            },
            non_tensors={"uid": np.array([r["uid"] for r in rows], dtype=object)},
        )


def epoch_batches(dataset, batch_size: int, shuffle: bool = True,
                  seed: int = 0, drop_last: bool = True
                  ) -> Iterator[TensorBatch]:
    """Deterministic batch iterator (identical on every rank for a seed)."""
    n = len(dataset)
    order = torch.randperm(n, generator=torch.Generator().manual_seed(seed)) \
        if shuffle else torch.arange(n)
    end = n - n % batch_size if drop_last else n
    for i in range(0, end, batch_size):
        yield dataset.batch(order[i:i + batch_size].tolist())


class ParquetRLHFDataset:
    """Tokenized prompts from parquet files (reference: RLHFDataset capability).

    Expects a column of token-id lists (`input_ids_key`) OR raw text plus a
    supplied tokenizer callable.
    """

    def __init__(self, files: List[str], max_prompt_length: int,
                 prompt_key: str = "prompt", tokenizer=None,
                 input_ids_key: Optional[str] = None, pad_token_id: int = 0,
                 data_source_key: str = "data_source",
                 filter_overlong: bool = True):
        import pandas as pd
        frames = [pd.read_parquet(f) for f in files]
        self.df = pd.concat(frames, ignore_index=True)
        self.Lp = max_prompt_length
        self.pad = pad_token_id
        self.prompt_key = prompt_key
        self.input_ids_key = input_ids_key
        self.tokenizer = tokenizer
        self.data_source_key = data_source_key
        self._ids: List[List[int]] = []
        keep = []
        for i, row in self.df.iterrows():
            if input_ids_key and input_ids_key in row:
                ids = list(row[input_ids_key])
            else:
                assert tokenizer is not None, "need tokenizer for text prompts"
                ids = tokenizer(str(row[prompt_key]))
            if len(ids) > self.Lp:
                if filter_overlong:
                    continue
                ids = ids[-self.Lp:]
            self._ids.append(ids)
            keep.append(i)
        self.df = self.df.iloc[keep].reset_index(drop=True)

    def __len__(self):
        return len(self._ids)

    def __getitem__(self, i: int) -> dict:
        ids_raw = self._ids[i]
        L = len(ids_raw)
        ids = torch.full((self.Lp,), self.pad, dtype=torch.long)
        mask = torch.zeros(self.Lp, dtype=torch.long)
        ids[self.Lp - L:] = torch.tensor(ids_raw, dtype=torch.long)
        mask[self.Lp - L:] = 1
        uid = hashlib.md5(str(ids_raw).encode()).hexdigest()[:12]
        return {"input_ids": ids, "attention_mask": mask, "uid": f"d{i}-{uid}"}

    def batch(self, indices: List[int]) -> TensorBatch:
        rows = [self[i] for i in indices]
        extra = {}
        if self.data_source_key in self.df.columns:
            extra["data_source"] = np.array(
                [self.df.iloc[i][self.data_source_key] for i in indices],
                dtype=object)
        # ground truth for the rule-based scorers (reward_score/)
        for col in ("ground_truth", "answer", "reward_model"):
            if col in self.df.columns:
                vals = [self.df.iloc[i][col] for i in indices]
                if col == "reward_model":  # verl style: dict with ground_truth
                    vals = [v.get("ground_truth") if isinstance(v, dict)
                            else v for v in vals]
                extra["ground_truth"] = np.array(vals, dtype=object)
                break
        return TensorBatch.from_dict(
            tensors={
                "input_ids": torch.stack([r["input_ids"] for r in rows]),
                "attention_mask": torch.stack([r["attention_mask"] for r in rows]),
            },
            non_tensors={
                "uid": np.array([r["uid"] for r in rows], dtype=object),
                **extra,
            },
        )
