"""Request/response types + continuation helpers for the rollout scheduler.

Mirrors the serde models and merge utilities of the reference
(rollout-manager/src/models.rs, utils.rs:45-65 merge_responses,
utils.rs:140-182 extend_input_ids_with_response_tokens,
utils.rs:256-291 adjust_sampling_params_for_used_tokens).
"""
from __future__ import annotations

from dataclasses import dataclass, field, replace
from typing import List


@dataclass
class SamplingSpec:
    temperature: float = 1.0
    top_k: int = -1
    top_p: float = 1.0
    max_new_tokens: int = 128
    stop_token_ids: tuple = ()


@dataclass
class GroupRequest:
    """One prompt group: n samples of the same prompt (the scheduler's relay
    unit, like BatchGenerationRequest entries in models.rs)."""
    gid: int
    input_ids: List[int]
    n: int
    sampling: SamplingSpec
    return_logprob: bool = True


@dataclass
class SampleResult:
    output_ids: List[int] = field(default_factory=list)
    output_logprobs: List[float] = field(default_factory=list)
    finish_reason: str = ""            # stop | length | abort | error
    completion_tokens: int = 0
    num_migrations: int = 0            # continuation hops (fault tolerance)


@dataclass
class GroupResult:
    gid: int
    samples: List[SampleResult] = field(default_factory=list)
    instance_ids: List[str] = field(default_factory=list)  # who served it


def merge_sample(prev: SampleResult, cont: SampleResult) -> SampleResult:
    """Concatenate a continuation's output onto the partial result
    (utils.rs:45-65: concat output_token_logprobs, sum completion_tokens)."""
    return SampleResult(
        output_ids=prev.output_ids + cont.output_ids,
        output_logprobs=prev.output_logprobs + cont.output_logprobs,
        finish_reason=cont.finish_reason,
        completion_tokens=prev.completion_tokens + cont.completion_tokens,
        num_migrations=prev.num_migrations + 1,
    )


def continuation_request(req: GroupRequest, sample_idx: int,
                         partial: SampleResult) -> GroupRequest:
    """Build the n=1 continuation request for one sample: already-generated
    tokens are appended to the prompt and max_new_tokens shrinks by the used
    amount (utils.rs:140-182 + :256-291 capability, done per sample so the
    continuation is token-exact for every sample, not just the shortest)."""
    used = len(partial.output_ids)
    remaining = max(req.sampling.max_new_tokens - used, 0)
    return GroupRequest(
        gid=req.gid,
        input_ids=list(req.input_ids) + list(partial.output_ids),
        n=1,
        sampling=replace(req.sampling, max_new_tokens=remaining),
        return_logprob=req.return_logprob,
    )


@dataclass
class InstanceStats:
    """Snapshot filled by the 1 Hz stats poll (instance_manager.rs:39-62)."""
    num_running: int = 0
    num_queued: int = 0
    gen_throughput: float = 0.0


@dataclass
class MetricsUpdate:
    """Trainer feedback for the load balancer (handlers.rs:867-901)."""
    step_time_s: float = 0.0
    trainer_bubble_time_s: float = 0.0
    step_throughput: float = 0.0
