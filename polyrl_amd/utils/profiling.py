"""Tracing / profiling utilities (reference capability SURVEY.md §5.1:
nsys + DistProfiler.annotate + GPUMemoryLogger + log_gpu_memory_usage).

MI355X-native mapping: torch.cuda.nvtx lowers to roctx on ROCm, so ranges
show up in rocprofv3 --marker-trace / --sys-trace sessions; per-step
torch.profiler windows replace the reference's per-step nsys start/stop
hooks (stream_ray_trainer.py:356-361,629-641)."""
from __future__ import annotations

import contextlib
import functools
import logging
import os
from typing import Optional

import torch

logger = logging.getLogger("polyrl_amd")


@contextlib.contextmanager
def roctx_range(name: str):
    """roctx marker range (visible in rocprofv3 marker traces)."""
    if torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def annotate(name: Optional[str] = None, color: str = ""):
    """Decorator: wrap a worker method in a roctx range (the reference's
    @DistProfiler.annotate role/color surface)."""
    def deco(fn):
        rng = name or fn.__qualname__

        @functools.wraps(fn)
        def wrapper(*a, **kw):
            with roctx_range(rng):
                return fn(*a, **kw)
        return wrapper
    return deco


def log_gpu_memory(tag: str, rank: int = 0):
    """Breadcrumb like the reference's log_gpu_memory_usage
    (stream_fsdp_workers.py:216-259)."""
    if not torch.cuda.is_available():
        return
    alloc = torch.cuda.memory_allocated() / (1 << 30)
    reserved = torch.cuda.memory_reserved() / (1 << 30)
    logger.info("[mem][rank %d] %s: allocated=%.1fGiB reserved=%.1fGiB",
                rank, tag, alloc, reserved)


class GPUMemoryLogger:
    """Context manager OR decorator measuring peak memory of a region
    (the reference uses the decorator form — stream_dp_actor.py:84)."""

    def __init__(self, tag: str):
        self.tag = tag

    def __call__(self, fn):
        @functools.wraps(fn)
        def wrapper(*a, **kw):
            with GPUMemoryLogger(self.tag):
                return fn(*a, **kw)
        return wrapper

    def __enter__(self):
        if torch.cuda.is_available():
            torch.cuda.reset_peak_memory_stats()
        return self

    def __exit__(self, *exc):
        if torch.cuda.is_available():
            peak = torch.cuda.max_memory_allocated() / (1 << 30)
            logger.info("[mem] %s: peak=%.1fGiB", self.tag, peak)
        return False


@contextlib.contextmanager
def step_profiler(enabled: bool, trace_dir: str = "profiles/torch",
                  step: int = 0):
    """Per-step torch.profiler window -> chrome trace under trace_dir
    (the reference's per-step nsys start/stop capability)."""
    if not enabled:
        yield None
        return
    os.makedirs(trace_dir, exist_ok=True)
    acts = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        acts.append(torch.profiler.ProfilerActivity.CUDA)
    import warnings
    with warnings.catch_warnings():
        # torch's informational "Profiler clears events each cycle" notice
        warnings.filterwarnings("ignore", message=".*Profiler clears events.*")
        with torch.profiler.profile(activities=acts) as prof:
            yield prof
    prof.export_chrome_trace(os.path.join(trace_dir, f"step_{step}.json"))
