"""EngineRunner — thread-pumped async facade over the synchronous Engine.

One background thread advances ``engine.step()`` whenever work is pending;
per-request futures resolve on the caller's asyncio loop.  Weight updates
take the step lock, so they exclude in-flight generation (the reference's
model_update_lock.writer_lock contract, rlboost/sglang/patches.py:482).

Used by scheduler.InProcessInstance (co-located) and server.engine_server
(remote elastic instance HTTP facade).
"""
from __future__ import annotations

import asyncio
import threading
import time
from typing import Dict, List, Optional

from .engine import Engine, RequestOutput, SamplingParams


class EngineRunner:
    def __init__(self, engine: Engine):
        self.engine = engine
        self.lock = threading.Lock()        # step / weight-update mutex
        # Python locks are not FIFO: the pump re-acquires within
        # microseconds of releasing, so an abort / weight-update / submit
        # caller can starve for an ENTIRE generation (measured: time-box
        # aborts landing after the final token).  Callers raise this count
        # before acquiring; the pump yields between steps while it is set.
        self._lock_waiters = 0
        self._waiters_mu = threading.Lock()
        self._futures: Dict[str, tuple] = {}  # rid -> (loop, future)
        self._pump: Optional[threading.Thread] = None
        self._stop = threading.Event()
        self._wake = threading.Event()
        self._gen_tokens = 0
        self._gen_window_t = time.monotonic()
        self.throughput = 0.0
        self.weight_version = 0
        self._rid_counter = 0

    # ------------------------------------------------------------- lifecycle
    def start(self):
        if self._pump is None or not self._pump.is_alive():
            self._stop.clear()
            self._pump = threading.Thread(target=self._pump_loop, daemon=True)
            self._pump.start()

    def stop(self):
        self._stop.set()
        self._wake.set()
        if self._pump is not None:
            self._pump.join(timeout=2.0)
            self._pump = None

    def _pump_loop(self):
        while not self._stop.is_set():
            if not self.engine.has_work():
                self._wake.wait(timeout=0.02)
                self._wake.clear()
                continue
            while self._lock_waiters > 0:       # yield to admin callers
                time.sleep(0.0005)
            with self.lock:
                outs = self.engine.step()
            if outs:
                self._account(outs)
                for o in outs:
                    entry = self._futures.pop(o.rid, None)
                    if entry is not None:
                        loop, fut = entry
                        loop.call_soon_threadsafe(
                            self._set_result_safe, fut, o)

    @staticmethod
    def _set_result_safe(fut: asyncio.Future, value):
        if not fut.done():
            fut.set_result(value)

    def _account(self, outs: List[RequestOutput]):
        self._gen_tokens += sum(len(o.output_ids) for o in outs)
        now = time.monotonic()
        dt = now - self._gen_window_t
        if dt > 0.5:
            self.throughput = self._gen_tokens / dt
            self._gen_tokens = 0
            self._gen_window_t = now

    # --------------------------------------------------------------- submit
    def submit(self, input_ids: List[int], sp: SamplingParams,
               rid: Optional[str] = None) -> "asyncio.Future[RequestOutput]":
        """Submit one request from an asyncio context; returns its future."""
        loop = asyncio.get_running_loop()
        if rid is None:
            self._rid_counter += 1
            rid = f"r{self._rid_counter}"
        fut = loop.create_future()
        self._futures[rid] = (loop, fut)
        with self._admin_lock():
            self.engine.add_request(rid, input_ids, sp)
        self.start()
        self._wake.set()
        return fut

    def submit_group(self, input_ids: List[int], sp: SamplingParams,
                     n: int) -> List["asyncio.Future[RequestOutput]"]:
        """n samples of one prompt with a shared prompt prefill (prefix
        sharing); returns one future per sample."""
        if n == 1 or not getattr(self.engine, "enable_prefix_sharing", False):
            return [self.submit(input_ids, sp) for _ in range(n)]
        loop = asyncio.get_running_loop()
        self._rid_counter += 1
        prefix = f"g{self._rid_counter}"
        futs = []
        for s_ in range(n):
            fut = loop.create_future()
            self._futures[f"{prefix}-s{s_}"] = (loop, fut)
            futs.append(fut)
        with self._admin_lock():
            self.engine.add_request_group(prefix, input_ids, sp, n)
        self.start()
        self._wake.set()
        return futs

    async def generate(self, input_ids: List[int], sp: SamplingParams,
                       n: int = 1) -> List[RequestOutput]:
        return list(await asyncio.gather(*self.submit_group(input_ids, sp,
                                                            n)))

    # ---------------------------------------------------------------- admin
    def _admin_lock(self):
        """Priority acquisition for non-pump callers (see _lock_waiters)."""
        import contextlib

        @contextlib.contextmanager
        def cm():
            with self._waiters_mu:
                self._lock_waiters += 1
            try:
                with self.lock:
                    yield
            finally:
                with self._waiters_mu:
                    self._lock_waiters -= 1
        return cm()

    def abort(self, rid: Optional[str] = None, abort_all: bool = False):
        with self._admin_lock():
            self.engine.abort_request(rid=rid, abort_all=abort_all)
        self._wake.set()

    def stats(self) -> dict:
        return {"#running_req": self.engine.num_running(),
                "#queue_req": self.engine.num_queued(),
                "last_gen_throughput": self.throughput,
                "weight_version": self.weight_version}

    def update_weights(self, state_dict, version: int, strict: bool = False,
                       abort_in_flight: bool = True):
        """Swap engine weights under the step lock (excludes generation).

        ``abort_in_flight`` mirrors the reference's KV flush after an update
        (patches.py:374-380): requests mid-generation are aborted rather
        than resumed on different weights; the scheduler continues them
        token-exactly on the new version.  Co-located publication happens
        between steps (no in-flight), so this is a no-op there."""
        with self._admin_lock():
            self.engine.model.load_state_dict(state_dict, strict=strict)
            if abort_in_flight and self.engine.has_work():
                self.engine.abort_request(abort_all=True)
            self.engine.flush_radix()   # cached KV is stale on new weights
            self.weight_version = version
        self._wake.set()
