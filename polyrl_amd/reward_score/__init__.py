"""Rule-based outcome scorers dispatched by data_source.

Reference capability: verl_stream/utils/reward_score/__init__.py:19-117
(default_compute_score routing gsm8k / MATH variants / math_dapo+aime /
numina / code (sandbox-fusion or local execution) / geo3k / searchR1 by the
sample's data_source tag, with the sandbox URL + concurrency semaphore +
memory limit threaded through from the reward loader,
trainer/ppo/reward.py:128-141).
"""
from __future__ import annotations

from typing import Optional

from . import code_exec, geo3k, gsm8k, math_dapo, math_score, search_r1

_MATH_SOURCES = frozenset([
    "lighteval/MATH", "DigitalLearningGmbH/MATH-lighteval",
    "HuggingFaceH4/MATH-500", "agentica-org/DeepScaleR-Preview-Dataset",
    "open-r1/OpenR1-Math-220k", "math", "openr1",
])
_NUMINA_SOURCES = frozenset([
    "numina_aops_forum", "numina_synthetic_math", "numina_amc_aime",
    "numina_synthetic_amc", "numina_cn_k12", "numina_olympiads",
])
_CODE_SOURCES = frozenset(["codecontests", "apps", "codeforces", "taco",
                           "code", "sandbox"])
_SEARCH_SOURCES = frozenset([
    "searchR1_nq", "searchR1_triviaqa", "searchR1_popqa",
    "searchR1_hotpotqa", "searchR1_2wikimultihopqa", "searchR1_musique",
    "searchR1_bamboogle",
])


def default_compute_score(data_source: str, solution_str: str,
                          ground_truth, extra_info=None,
                          sandbox_fusion_url: Optional[str] = None,
                          concurrent_semaphore=None,
                          memory_limit_mb: Optional[int] = None) -> float:
    """Route to the scorer for this data source.  Returns a float score;
    dict-returning scorers (math_dapo) are collapsed to their 'score'."""
    if data_source in ("openai/gsm8k", "gsm8k"):
        res = gsm8k.compute_score(solution_str, ground_truth)
    elif data_source in _MATH_SOURCES:
        res = math_score.compute_score(solution_str, ground_truth)
    elif data_source == "math_dapo" or data_source.startswith("aime") \
            or data_source in ("HuggingFaceH4/aime_2024",):
        res = math_dapo.compute_score(solution_str, ground_truth)
    elif data_source in _NUMINA_SOURCES or data_source.startswith("numina"):
        res = math_score.compute_score(solution_str, ground_truth)
    elif data_source in _CODE_SOURCES:
        if sandbox_fusion_url:
            res = code_exec.compute_score_sandbox(
                sandbox_fusion_url, concurrent_semaphore, memory_limit_mb,
                solution_str, ground_truth, continuous=True)
        else:
            res = code_exec.compute_score(
                solution_str, ground_truth, continuous=True,
                memory_limit_mb=memory_limit_mb or 1024)
    elif data_source in ("hiyouga/geometry3k", "geo3k"):
        res = geo3k.compute_score(solution_str, ground_truth)
    elif data_source in _SEARCH_SOURCES or data_source.startswith("searchR1"):
        res = search_r1.compute_score(solution_str, ground_truth)
    else:
        raise KeyError(f"no scorer for data_source {data_source!r}")
    if isinstance(res, dict):
        return float(res["score"])
    if isinstance(res, (int, float, bool)):
        return float(res)
    return float(res[0])
