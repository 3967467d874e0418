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
import threading
import time
from typing import Dict, List, Optional

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
    """Co-located engine on this rank's GPU.

    A dedicated pump thread advances ``engine.step()`` whenever requests are
    pending, and resolves per-request futures back onto the asyncio loop.
    The weight-update path takes the pump lock so updates exclude in-flight
    generation (the reference's model_update_lock.writer_lock contract,
    patches.py:482).
    """

    def __init__(self, engine, instance_id: str = "local-0",
                 weight_source=None):
        self.engine = engine
        self.instance_id = instance_id
        self.is_local = True
        self.weight_source = weight_source  # callable version -> state_dict
        self._lock = threading.Lock()       # engine-step / weight-update mutex
        self._futures: Dict[str, asyncio.Future] = {}
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._pump: Optional[threading.Thread] = None
        self._stop = threading.Event()
        self._wake = threading.Event()
        self._gen_tokens = 0
        self._gen_window_t = time.monotonic()
        self._throughput = 0.0
        self.weight_version = 0

    # ------------------------------------------------------------- pump
    def _ensure_pump(self):
        if self._pump is None or not self._pump.is_alive():
            self._stop.clear()
            self._pump = threading.Thread(target=self._pump_loop, daemon=True)
            self._pump.start()

    def _pump_loop(self):
        while not self._stop.is_set():
            if not self.engine.has_work():
                self._wake.wait(timeout=0.05)
                self._wake.clear()
                continue
            with self._lock:
                outs = self.engine.step()
            if outs:
                self._gen_tokens += sum(len(o.output_ids) for o in outs)
                now = time.monotonic()
                dt = now - self._gen_window_t
                if dt > 0.5:
                    self._throughput = self._gen_tokens / dt
                    self._gen_tokens = 0
                    self._gen_window_t = now
                for o in outs:
                    fut = self._futures.pop(o.rid, None)
                    if fut is not None and self._loop is not None:
                        self._loop.call_soon_threadsafe(
                            fut.set_result, SampleResult(
                                output_ids=list(o.output_ids),
                                output_logprobs=list(o.output_logprobs),
                                finish_reason=o.finish_reason,
                                completion_tokens=len(o.output_ids)))

    # ---------------------------------------------------------- contract
    async def generate_group(self, req: GroupRequest) -> List[SampleResult]:
        from ..rollout.engine import SamplingParams
        self._loop = asyncio.get_running_loop()
        self._ensure_pump()
        sp = SamplingParams(
            temperature=req.sampling.temperature, top_k=req.sampling.top_k,
            top_p=req.sampling.top_p,
            max_new_tokens=req.sampling.max_new_tokens,
            stop_token_ids=tuple(req.sampling.stop_token_ids))
        futs = []
        for s in range(req.n):
            rid = f"g{req.gid}-s{s}-{id(req)}"
            fut = self._loop.create_future()
            self._futures[rid] = fut
            with self._lock:
                self.engine.add_request(rid, req.input_ids, sp)
            futs.append(fut)
        self._wake.set()
        return list(await asyncio.gather(*futs))

    def get_stats(self) -> InstanceStats:
        return InstanceStats(num_running=self.engine.num_running(),
                             num_queued=self.engine.num_queued(),
                             gen_throughput=self._throughput)

    async def update_weights(self, version: int, bootstrap: bool = False
                             ) -> bool:
        if self.weight_source is None:
            self.weight_version = version
            return True
        sd = self.weight_source(version)
        with self._lock:  # excludes generation while swapping
            self.engine.model.load_state_dict(sd, strict=False)
        self.weight_version = version
        return True

    def abort_all(self):
        with self._lock:
            self.engine.abort_request(abort_all=True)
        self._wake.set()

    async def shutdown(self):
        self._stop.set()
        self._wake.set()
        if self._pump is not None:
            self._pump.join(timeout=2.0)


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
