"""DAPO-style math scorer (math_dapo / aime data sources).

Reference capability: verl_stream/utils/reward_score/__init__.py:62-65
routes ``math_dapo`` and ``aime*`` to verl's math_dapo scorer.  That scorer
family (DAPO paper convention) extracts the LAST \\boxed{...} answer (or a
final "Answer: ..." line), compares after aggressive normalization, and
returns +1 for a correct answer and -1 for an incorrect one (the asymmetric
reward DAPO trains with), plus an ``acc`` flag.  Re-derived here from the
dispatch contract; no reference code available (verl submodule is empty).
"""
from __future__ import annotations

import re
from typing import Dict, Optional, Union

from .math_score import last_boxed, normalize


def extract_solution(solution_str: str) -> Optional[str]:
    """Last boxed answer; else the text after a final 'Answer:' marker."""
    boxed = last_boxed(solution_str)
    if boxed is not None:
        return boxed
    m = None
    for m in re.finditer(r"(?:final answer|answer)\s*(?:is|:)\s*([^\n.]+)",
                         solution_str, flags=re.IGNORECASE):
        pass
    if m:
        return m.group(1).strip()
    return None


def compute_score(solution_str: str, ground_truth: str,
                  strict_box_verify: bool = False,
                  ) -> Dict[str, Union[float, bool]]:
    """+1 correct / -1 wrong (DAPO convention), with ``acc`` and the
    extracted prediction for logging.  Returns a dict — callers that need a
    float take ["score"] (default_compute_score handles both)."""
    pred = extract_solution(solution_str)
    if strict_box_verify and last_boxed(solution_str) is None:
        pred = None
    acc = (pred is not None
           and normalize(pred) == normalize(str(ground_truth)))
    return {"score": 1.0 if acc else -1.0, "acc": acc,
            "pred": pred if pred is not None else ""}
