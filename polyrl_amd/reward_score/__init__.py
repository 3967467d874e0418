"""Rule-based outcome scorers dispatched by data_source.

Reference capability: verl_stream/utils/reward_score/__init__.py:19-117
(default_compute_score routing gsm8k / MATH variants / math_dapo / numina /
code / geo3k by the sample's data_source tag).  Network-dependent scorers
(sandbox-fusion code execution) are represented by the ``code_stub`` entry —
the routing contract is what matters here.
"""
from __future__ import annotations

from typing import Optional

from . import gsm8k, math_score


def default_compute_score(data_source: str, solution_str: str,
                          ground_truth: str, extra_info=None) -> float:
    """Route to the scorer for this data source; returns a float score
    (1.0 correct / 0.0 wrong for the exact-match scorers)."""
    if data_source in ("openai/gsm8k", "gsm8k"):
        return gsm8k.compute_score(solution_str, ground_truth)
    if data_source in ("lighteval/MATH", "math", "math_dapo",
                       "HuggingFaceH4/aime_2024", "aime",
                       "open-r1/OpenR1-Math-220k", "openr1"):
        return math_score.compute_score(solution_str, ground_truth)
    if data_source.startswith("numina"):
        return math_score.compute_score(solution_str, ground_truth)
    if data_source in ("code", "sandbox"):
        raise NotImplementedError(
            "code execution scoring needs a sandbox service (reference: "
            "sandbox-fusion URL config) — not available offline")
    raise KeyError(f"no scorer for data_source {data_source!r}")
