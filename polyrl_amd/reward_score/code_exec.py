"""Code-execution scorers for RL-for-code training rewards.

Reference capability: verl_stream/utils/reward_score/__init__.py:79-95 —
the codecontests/apps/codeforces/taco data sources score by RUNNING the
model's program against the ground-truth test cases, either through a
remote **sandbox-fusion** service (HTTP POST /run_code, gated by a
concurrency semaphore and a memory limit — trainer/ppo/reward.py:128-141)
or the local **prime_code** fallback (subprocess execution).

Ground truth format (prime/codecontests convention): a JSON string
``{"inputs": [...], "outputs": [...]}`` (stdin/stdout pairs), or a list of
``assert``-style snippets.  ``continuous=True`` returns the pass fraction;
otherwise 1.0 only when every test passes.

The local runner executes model-generated code in a subprocess with a
wall-clock timeout and an address-space rlimit — the standard RL code
reward harness (authorized by construction: scoring our own model's
training rollouts).
"""
from __future__ import annotations

import json
import subprocess
import sys
from typing import List, Optional, Tuple

_RUN_TIMEOUT_S = 10.0


def _parse_ground_truth(ground_truth) -> Tuple[List[str], List[str]]:
    """-> (inputs, outputs) stdin/stdout pairs."""
    if isinstance(ground_truth, (dict,)):
        gt = ground_truth
    else:
        try:
            gt = json.loads(ground_truth)
        except (TypeError, ValueError):
            # a bare expected-stdout string: single test, empty stdin
            return [""], [str(ground_truth)]
    if isinstance(gt, dict):
        return [str(x) for x in gt.get("inputs", [""])], \
               [str(x) for x in gt.get("outputs", [])]
    if isinstance(gt, list):                       # assert snippets
        return list(gt), []
    return [""], [str(gt)]


def extract_code(solution_str: str) -> Optional[str]:
    """Last fenced code block; else the raw text if it looks like code."""
    import re
    blocks = re.findall(r"```(?:python|py|cpp|c\+\+)?\n(.*?)```",
                        solution_str, flags=re.DOTALL)
    if blocks:
        return blocks[-1]
    if "def " in solution_str or "print(" in solution_str \
            or "input()" in solution_str:
        return solution_str
    return None


def _run_local(code: str, stdin: str, memory_limit_mb: int) -> Tuple[bool, str]:
    """Run one test in a subprocess; returns (completed_ok, stdout)."""
    preamble = (
        "import resource, sys\n"
        f"resource.setrlimit(resource.RLIMIT_AS, "
        f"({memory_limit_mb} * 1024 * 1024,) * 2)\n"
        "sys.setrecursionlimit(10000)\n")
    try:
        proc = subprocess.run(
            [sys.executable, "-c", preamble + code],
            input=stdin.encode(), capture_output=True,
            timeout=_RUN_TIMEOUT_S)
        return proc.returncode == 0, proc.stdout.decode(errors="replace")
    except subprocess.TimeoutExpired:
        return False, ""
    except OSError:
        return False, ""


def _outputs_match(got: str, want: str) -> bool:
    g = [ln.rstrip() for ln in got.strip().splitlines()]
    w = [ln.rstrip() for ln in want.strip().splitlines()]
    return g == w


def compute_score(solution_str: str, ground_truth,
                  continuous: bool = True,
                  memory_limit_mb: int = 1024) -> float:
    """prime_code-style local scorer: run the extracted program on every
    (stdin, stdout) test; pass fraction (continuous) or all-or-nothing."""
    code = extract_code(solution_str)
    if code is None:
        return 0.0
    inputs, outputs = _parse_ground_truth(ground_truth)
    if not outputs:
        # assert-snippet style: append each snippet and check exit code
        passed = 0
        for snippet in inputs:
            ok, _ = _run_local(code + "\n" + snippet, "", memory_limit_mb)
            passed += ok
        n = max(len(inputs), 1)
        return passed / n if continuous else float(passed == n)
    passed = 0
    for stdin, want in zip(inputs, outputs):
        ok, got = _run_local(code, stdin, memory_limit_mb)
        passed += ok and _outputs_match(got, want)
    n = max(len(outputs), 1)
    return passed / n if continuous else float(passed == n)


def compute_score_sandbox(sandbox_fusion_url: str, concurrent_semaphore,
                          memory_limit_mb, solution_str: str, ground_truth,
                          continuous: bool = True,
                          timeout_s: float = 30.0) -> float:
    """sandbox-fusion remote scorer (reference: sandbox_fusion.compute_score
    called with (url, semaphore, memory_limit_mb, solution, gt, continuous),
    trainer/ppo/reward.py:131-134).  POSTs each test to {url}/run_code and
    compares stdout; the semaphore bounds concurrent sandbox calls."""
    import urllib.request

    code = extract_code(solution_str)
    if code is None:
        return 0.0
    inputs, outputs = _parse_ground_truth(ground_truth)
    tests = list(zip(inputs, outputs)) if outputs else \
        [(code + "\n" + s, None) for s in inputs]
    passed = 0
    for stdin, want in tests:
        payload = {
            "code": code if want is not None else stdin,
            "stdin": stdin if want is not None else "",
            "language": "python",
            "run_timeout": timeout_s,
        }
        if memory_limit_mb:
            payload["memory_limit_MB"] = memory_limit_mb
        body = json.dumps(payload).encode()
        req = urllib.request.Request(
            sandbox_fusion_url.rstrip("/") + "/run_code", data=body,
            headers={"Content-Type": "application/json"})
        ctx = concurrent_semaphore
        try:
            if ctx is not None:
                ctx.acquire()
            try:
                with urllib.request.urlopen(req, timeout=timeout_s + 5) as r:
                    res = json.loads(r.read().decode())
            finally:
                if ctx is not None:
                    ctx.release()
        except Exception:                          # noqa: BLE001
            continue                               # sandbox error = failed test
        status_ok = res.get("status") in ("Success", "success", "ok") or \
            res.get("run_result", {}).get("return_code", 1) == 0
        got = res.get("run_result", {}).get("stdout", res.get("stdout", ""))
        if want is None:
            passed += status_ok
        else:
            passed += status_ok and _outputs_match(got, str(want))
    n = max(len(tests), 1)
    return passed / n if continuous else float(passed == n)
