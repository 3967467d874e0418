"""Rollout instance abstraction for the scheduler.

The reference talks HTTP to SGLang servers (handlers.rs:152-328 relays
``POST /generate`` SSE streams); here an instance is an object with an async
contract, so co-located engines take no network hop.  Implementations:

  * InProcessInstance — wraps rollout.engine.Engine on this process's GPU;
    a pump thread advances engine.step() while work is pending.
  * FakeInstance — deterministic token echo + fault injection for the
    CPU-tier scheduler tests (SURVEY.md §4 test strategy).
  * (server.HttpInstance — remote elastic instances over the HTTP facade,
    polyrl_amd/server/.)
"""
from __future__ import annotations

import asyncio
from typing import List

from .types import GroupRequest, InstanceStats, SampleResult


class RolloutInstance:
    """Abstract instance the scheduler schedules onto."""

    instance_id: str = "?"
    is_local: bool = True

    async def generate_group(self, req: GroupRequest) -> List[SampleResult]:
        """Run all n samples of one prompt group.  On abort, returns partial
        outputs with finish_reason='abort'.  Raises on instance failure."""
        raise NotImplementedError

    def get_stats(self) -> InstanceStats:
        raise NotImplementedError

    async def health(self) -> bool:
        return True

    async def update_weights(self, version: int, bootstrap: bool = False
                             ) -> bool:
        """Install the pending weight version (transfer plane delivers the
        bytes; this is the activation call, patches.py:169-241 capability)."""
        raise NotImplementedError

    def abort_all(self):
        """Abort in-flight generation (engine /abort_request{abort_all})."""
        raise NotImplementedError

    async def shutdown(self):
        pass


class InProcessInstance(RolloutInstance):
    """Co-located engine on this rank's GPU, pumped by rollout.runner.
    Weight updates run under the step lock so they exclude in-flight
    generation (patches.py:482 writer-lock contract)."""

    def __init__(self, engine, instance_id: str = "local-0",
                 weight_source=None):
        from ..rollout.runner import EngineRunner
        self.engine = engine
        self.runner = EngineRunner(engine)
        self.instance_id = instance_id
        self.is_local = True
        self.weight_source = weight_source  # callable version -> state_dict

    @property
    def weight_version(self) -> int:
        return self.runner.weight_version

    async def generate_group(self, req: GroupRequest) -> List[SampleResult]:
        from ..rollout.engine import SamplingParams
        sp = SamplingParams(
            temperature=req.sampling.temperature, top_k=req.sampling.top_k,
            top_p=req.sampling.top_p,
            max_new_tokens=req.sampling.max_new_tokens,
            stop_token_ids=tuple(req.sampling.stop_token_ids))
        outs = await self.runner.generate(req.input_ids, sp, n=req.n)  # group-shared prefill
        return [SampleResult(output_ids=list(o.output_ids),
                             output_logprobs=list(o.output_logprobs),
                             finish_reason=o.finish_reason,
                             completion_tokens=len(o.output_ids))
                for o in outs]

    def get_stats(self) -> InstanceStats:
        s = self.runner.stats()
        return InstanceStats(num_running=s["#running_req"],
                             num_queued=s["#queue_req"],
                             gen_throughput=s["last_gen_throughput"])

    async def update_weights(self, version: int, bootstrap: bool = False
                             ) -> bool:
        if self.weight_source is None:
            self.runner.weight_version = version
            return True
        sd = self.weight_source(version)
        self.runner.update_weights(sd, version, strict=False)
        return True

    def abort_all(self):
        self.runner.abort(abort_all=True)

    async def shutdown(self):
        self.runner.stop()


class FakeInstance(RolloutInstance):
    """Deterministic fake engine for scheduler tests.

    Echoes ``prompt[-1]+1+i`` as output token i, logprob = -0.5 per token,
    one token per ``token_time_s``.  Fault injection: ``fail_after_tokens``
    raises mid-generation (leaving partials via the exception payload);
    ``aborted`` event forces abort finish.
    """

    class Failure(Exception):
        def __init__(self, partials: List[SampleResult]):
            self.partials = partials
            super().__init__("injected instance failure")

    def __init__(self, instance_id: str, is_local: bool = True,
                 token_time_s: float = 0.0, fail_after_tokens: int = -1):
        self.instance_id = instance_id
        self.is_local = is_local
        self.token_time_s = token_time_s
        self.fail_after_tokens = fail_after_tokens
        self.abort_event = asyncio.Event()
        self.weight_version = 0
        self.healthy = True
        self.update_calls: List[int] = []
        self.served_gids: List[int] = []
        self._running = 0

    async def generate_group(self, req: GroupRequest) -> List[SampleResult]:
        self.served_gids.append(req.gid)
        self._running += req.n
        try:
            outs = [SampleResult() for _ in range(req.n)]
            base = req.input_ids[-1]
            for t in range(req.sampling.max_new_tokens):
                if self.token_time_s:
                    try:
                        await asyncio.wait_for(self.abort_event.wait(),
                                               timeout=self.token_time_s)
                    except asyncio.TimeoutError:
                        pass
                if self.abort_event.is_set():
                    for o in outs:
                        o.finish_reason = "abort"
                    return outs
                if self.fail_after_tokens >= 0 and t >= self.fail_after_tokens:
                    raise FakeInstance.Failure(outs)
                for o in outs:
                    o.output_ids.append((base + 1 + t) % 50000)
                    o.output_logprobs.append(-0.5)
                    o.completion_tokens += 1
            for o in outs:
                o.finish_reason = "length"
            return outs
        finally:
            self._running -= req.n

    def get_stats(self) -> InstanceStats:
        return InstanceStats(num_running=self._running, num_queued=0,
                             gen_throughput=0.0)

    async def health(self) -> bool:
        return self.healthy

    async def update_weights(self, version: int, bootstrap: bool = False
                             ) -> bool:
        self.update_calls.append(version)
        if not self.healthy:
            return False
        self.weight_version = version
        return True

    def abort_all(self):
        self.abort_event.set()
