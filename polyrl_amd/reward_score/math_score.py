"""MATH-style boxed-answer scorer (reference capability: verl's
utils/reward_score math family — extract \\boxed{...}, normalize LaTeX,
exact-match).  Covers MATH / math_dapo / AIME / OpenR1-style outputs."""
from __future__ import annotations

import re


def last_boxed(text: str):
    """Return the content of the last \\boxed{...} (brace-balanced)."""
    idx = text.rfind("\\boxed")
    if idx < 0:
        fidx = text.rfind("\\fbox")
        if fidx < 0:
            return None
        idx = fidx
    i = text.find("{", idx)
    if i < 0:
        return None
    depth = 0
    for j in range(i, len(text)):
        if text[j] == "{":
            depth += 1
        elif text[j] == "}":
            depth -= 1
            if depth == 0:
                return text[i + 1:j]
    return None


_SUBS = [
    ("an ", ""), ("a ", ""), (".$", "$"), ("\\$", ""), (r"\ ", ""),
    (" ", ""), ("mbox", "text"), (",\\text{and}", ","),
    ("\\text{and}", ","), ("\\text{m}", "\\text{}"),
]
_REMOVE = [
    "square", "ways", "integers", "dollars", "mph", "inches", "ft", "hours",
    "km", "units", "\\ldots", "sue", "points", "feet", "minutes", "digits",
    "cents", "degrees", "cm", "gm", "pounds", "meters", "meals", "edges",
    "students", "childrentickets", "multiples", "\\text{s}", "\\text{.}",
    "\\text{\ns}", "\\text{}^2", "\\text{}^3", "\\text{\n}", "\\text{}",
    r"\mathrm{th}", r"^\circ", r"^{\circ}", r"\;", r",\!", "{,}", '"',
    "\\dots",
]


def normalize(ans: str) -> str:
    if ans is None:
        return ""
    s = ans.strip()
    for a, b in _SUBS:
        s = s.replace(a, b)
    for r in _REMOVE:
        s = s.replace(r, "")
    s = s.replace("tfrac", "frac").replace("dfrac", "frac")
    s = s.replace("\\left", "").replace("\\right", "")
    s = re.sub(r"\\text\{(.*?)\}", r"\1", s)
    s = re.sub(r"(frac)([^{])([^{])", r"frac{\2}{\3}", s)
    s = re.sub(r"(sqrt)([^{])", r"sqrt{\2}", s)
    s = s.replace("$", "").replace("%", "").replace("percent", "")
    if s.startswith("."):
        s = "0" + s
    if "=" in s:
        s = s.split("=")[-1]
    # 0.5 == 1/2 style: canonicalize plain decimals that equal ints
    try:
        f = float(s)
        if f == f and abs(f) != float("inf") and f == int(f):
            return str(int(f))
    except (ValueError, OverflowError):
        pass
    return s


def compute_score(solution_str: str, ground_truth: str) -> float:
    ans = last_boxed(solution_str)
    if ans is None:
        # fall back: answer may be stated bare at the end
        ans = solution_str.strip().split("\n")[-1]
    return 1.0 if normalize(ans) == normalize(str(ground_truth)) else 0.0
