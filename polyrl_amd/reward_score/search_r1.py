"""SearchR1-style QA exact-match scorer (searchR1_* data sources).

Reference capability: verl_stream/utils/reward_score/__init__.py:100-114
routes the seven searchR1_* datasets to verl's search_r1_like_qa_em:
extract the model's final <answer>...</answer> span and score normalized
exact-match against the ground truth (which may be one string or a list of
acceptable answers).  Re-derived from the dispatch contract.
"""
from __future__ import annotations

import re
import string
from typing import Iterable, Union


def normalize_answer(s: str) -> str:
    """SQuAD-style normalization: lowercase, strip punctuation/articles,
    collapse whitespace."""
    s = s.lower()
    s = "".join(ch for ch in s if ch not in set(string.punctuation))
    s = re.sub(r"\b(a|an|the)\b", " ", s)
    return " ".join(s.split())


def extract_answer(solution_str: str):
    """Last <answer>...</answer> span (the searchR1 rollout format)."""
    matches = re.findall(r"<answer>(.*?)</answer>", solution_str,
                         flags=re.DOTALL)
    if matches:
        return matches[-1].strip()
    return None


def compute_score(solution_str: str,
                  ground_truth: Union[str, Iterable[str], dict],
                  format_score: float = 0.0) -> float:
    pred = extract_answer(solution_str)
    if pred is None:
        return 0.0
    if isinstance(ground_truth, dict):     # {"target": [...]} style
        ground_truth = ground_truth.get("target", [])
    if isinstance(ground_truth, str):
        golds = [ground_truth]
    else:
        golds = [str(g) for g in ground_truth]
    npred = normalize_answer(pred)
    for g in golds:
        if npred == normalize_answer(g):
            return 1.0
    return format_score
