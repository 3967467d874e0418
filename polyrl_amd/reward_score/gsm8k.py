"""GSM8K exact-match scorer (reference capability: verl's
utils/reward_score/gsm8k.py surface — extract the final numeric answer in
'#### <answer>' format, fall back to the last number in the solution)."""
from __future__ import annotations

import re

_ANS = re.compile(r"####\s*([\-\$0-9\.,]+)")
_NUM = re.compile(r"[\-]?[0-9][0-9,\.]*")


def _norm(num: str) -> str:
    num = num.strip().replace(",", "").replace("$", "")
    num = num.rstrip(".")
    try:
        f = float(num)
        if f == int(f):
            return str(int(f))
        return str(f)
    except ValueError:
        return num


def extract_answer(solution: str, method: str = "strict"):
    m = _ANS.findall(solution)
    if m:
        return _norm(m[-1])
    if method == "flexible":
        nums = _NUM.findall(solution)
        if nums:
            return _norm(nums[-1])
    return None


def compute_score(solution_str: str, ground_truth: str,
                  method: str = "flexible", format_score: float = 0.0,
                  score: float = 1.0) -> float:
    ans = extract_answer(solution_str, method)
    if ans is None:
        return 0.0
    return score if ans == _norm(str(ground_truth)) else format_score
