"""Geometry3K scorer (hiyouga/geometry3k data source).

Reference capability: verl_stream/utils/reward_score/__init__.py:96-99
routes geometry3k to verl's geo3k scorer: extract the last \\boxed{...}
answer and compare numerically with a relative tolerance (geo answers are
numeric — lengths/angles/areas).  Re-derived from the dispatch contract.
"""
from __future__ import annotations

from .math_score import last_boxed, normalize


def _to_float(s: str):
    try:
        return float(normalize(s).replace(",", ""))
    except (ValueError, AttributeError):
        return None


def compute_score(solution_str: str, ground_truth: str,
                  rel_tol: float = 1e-4) -> float:
    pred = last_boxed(solution_str)
    if pred is None:
        return 0.0
    fp, fg = _to_float(pred), _to_float(str(ground_truth))
    if fp is not None and fg is not None:
        if fg == 0:
            return 1.0 if abs(fp) <= rel_tol else 0.0
        return 1.0 if abs(fp - fg) / abs(fg) <= rel_tol else 0.0
    return 1.0 if normalize(pred) == normalize(str(ground_truth)) else 0.0
