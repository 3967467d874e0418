"""RolloutScheduler — the in-process scheduler core.

Behavioral port of the reference Rust manager (rollout-manager/src/):
  * zero-queue round-robin dispatch with per-stats-window admission caps
    (state.rs:84-147 next_instance_with_type),
  * a task per prompt group, results streamed as they finish; the local
    engines are time-boxed: after ``max_local_gen_s`` they are removed from
    the active pool and their in-flight requests aborted, the tail continues
    on remote instances (handlers.rs:442-564 timed_batch_generate_requests),
  * token-level continuation on instance failure/abort: already-generated
    tokens are appended to the prompt, max_new_tokens shrinks, retry on
    another instance, cap 5 attempts (handlers.rs:330-418),
  * weight-version gating: a version bump clears the active pool and re-adds
    local (already-updated) instances; remote instances rejoin only after
    their update succeeds (handlers.rs:566-795),
  * adaptive time-box via LoadBalanceState (balance.rs; /update_metrics
    feedback loop, handlers.rs:867-901).

Deviation from the reference (deliberate): the time-box only fires when at
least one REMOTE instance is active — with nothing to hand the tail to,
aborting local engines would strand the batch (the reference assumes remote
capacity exists).
"""
from __future__ import annotations

import asyncio
import threading
import time
from dataclasses import dataclass
from typing import AsyncIterator, Dict, List, Optional

from .balance import LoadBalanceState
from .instances import RolloutInstance
from .types import (GroupRequest, GroupResult, MetricsUpdate, SampleResult,
                    continuation_request, merge_sample)


@dataclass
class SchedulerConfig:
    max_assigned_batches_per_stats_check: int = 4   # config.toml default
    # pluggable dispatch policy over the eligible set (the reference's
    # 'interface of algorithm-driven request scheduling' roadmap line):
    # "zero_queue_rr" (state.rs default) | "least_loaded" | a callable
    # (eligible_ids, states, rr_counter) -> instance_id
    scheduling_policy: object = "zero_queue_rr"
    stats_interval_s: float = 1.0                   # instance_manager.rs:39
    max_retries: int = 5                            # handlers.rs retry cap
    initial_max_local_gen_s: float = 150.0          # state.rs:79
    health_check_timeout_s: float = 300.0           # instance_manager.rs:5-37
    health_check_interval_s: float = 2.0


class _InstState:
    """Mutable per-instance scheduling state (state.rs:9-25 atomics)."""

    def __init__(self, inst: RolloutInstance):
        self.inst = inst
        self.assigned_batches = 0      # admission window counter
        self.stats = inst.get_stats()
        self.updating_weight = False
        self.weight_version = 0
        self.weight_sender_endpoint: Optional[str] = None
        self.evicted = False


class RolloutScheduler:
    def __init__(self, cfg: SchedulerConfig = None):
        self.cfg = cfg or SchedulerConfig()
        self._states: Dict[str, _InstState] = {}
        self._active: List[str] = []            # active pool (dispatchable)
        self._rr = 0                            # round-robin cursor
        self._weight_senders: List[str] = []    # sender endpoints (PUT route)
        self._sender_rr = 0                     # sender round-robin cursor
        self._cond: Optional[asyncio.Condition] = None
        self.latest_weight_version = 0
        self.balance = LoadBalanceState(
            initial_gen_s=self.cfg.initial_max_local_gen_s)
        self.max_local_gen_s = self.cfg.initial_max_local_gen_s
        self._stats_task: Optional[asyncio.Task] = None
        self._closed = False
        # timing stats fed back to the trainer
        self.last_batch_gen_s = 0.0

    # ------------------------------------------------------------ registry
    async def register_instance(self, inst: RolloutInstance,
                                skip_health_check: bool = False):
        """Health-gate then activate (handlers.rs:40-86)."""
        if not skip_health_check:
            deadline = time.monotonic() + self.cfg.health_check_timeout_s
            while not await inst.health():
                if time.monotonic() > deadline:
                    raise TimeoutError(
                        f"instance {inst.instance_id} failed health check")
                await asyncio.sleep(self.cfg.health_check_interval_s)
        st = _InstState(inst)
        st.weight_version = getattr(inst, "weight_version", 0)
        st.weight_sender_endpoint = self._assign_weight_sender()
        self._states[inst.instance_id] = st
        # local instances join the active pool immediately; remote instances
        # join once their weights reach the latest version (§3.4 lifecycle)
        if inst.is_local or st.weight_version >= self.latest_weight_version:
            if inst.instance_id not in self._active:
                self._active.append(inst.instance_id)
        await self._notify()

    def instances(self, active_only: bool = False) -> List[RolloutInstance]:
        if active_only:
            return [self._states[i].inst for i in self._active]
        return [s.inst for s in self._states.values()]

    def num_active(self) -> int:
        return len(self._active)

    def num_remote_active(self) -> int:
        return sum(1 for i in self._active
                   if not self._states[i].inst.is_local)

    async def evict_instance(self, instance_id: str, shutdown: bool = True):
        """Remove a failed instance (handlers.rs:363-414 evict path)."""
        st = self._states.pop(instance_id, None)
        if instance_id in self._active:
            self._active.remove(instance_id)
        if st is not None:
            st.evicted = True
            if shutdown:
                try:
                    await st.inst.shutdown()
                except Exception:
                    pass
        await self._notify()

    # ------------------------------------------------------- stats sampling
    def _get_cond(self) -> asyncio.Condition:
        if self._cond is None:
            self._cond = asyncio.Condition()
        return self._cond

    async def _notify(self):
        cond = self._get_cond()
        async with cond:
            cond.notify_all()

    async def start(self):
        if self._stats_task is None:
            self._stats_task = asyncio.ensure_future(self._stats_loop())

    async def close(self):
        self._closed = True
        if self._stats_task is not None:
            self._stats_task.cancel()
            try:
                await self._stats_task
            except (asyncio.CancelledError, Exception):
                pass
            self._stats_task = None

    async def _stats_loop(self):
        """1 Hz: refresh stats, reset admission counters, wake waiters
        (instance_manager.rs:39-62)."""
        while not self._closed:
            await asyncio.sleep(self.cfg.stats_interval_s)
            self.refresh_stats()
            await self._notify()

    def refresh_stats(self):
        for st in self._states.values():
            try:
                st.stats = st.inst.get_stats()
            except Exception:
                pass
            st.assigned_batches = 0

    # ------------------------------------------------------------- dispatch
    async def next_instance(self) -> RolloutInstance:
        """Zero-queue round-robin with admission throttle (state.rs:84-147):
        among active instances with no queued work and assignment headroom,
        pick round-robin; otherwise wait for the stats tick."""
        cond = self._get_cond()
        while True:
            eligible = []
            for iid in self._active:
                st = self._states.get(iid)
                if st is None or st.updating_weight:
                    continue
                if st.assigned_batches >= \
                        self.cfg.max_assigned_batches_per_stats_check:
                    continue
                if st.stats.num_queued > 0:
                    continue
                eligible.append(iid)
            if eligible:
                iid = self._pick(eligible)
                self._states[iid].assigned_batches += 1
                return self._states[iid].inst
            async with cond:
                try:
                    await asyncio.wait_for(cond.wait(),
                                           timeout=self.cfg.stats_interval_s)
                except asyncio.TimeoutError:
                    pass
                self.refresh_stats()

    def _pick(self, eligible: List[str]) -> str:
        """Dispatch policy over the eligible set (algorithm-driven
        scheduling interface)."""
        pol = self.cfg.scheduling_policy
        if callable(pol):
            return pol(eligible, self._states, self._rr)
        if pol == "least_loaded":
            return min(eligible,
                       key=lambda i: (self._states[i].stats.num_running
                                      + self._states[i].stats.num_queued,
                                      self._states[i].assigned_batches, i))
        # zero_queue_rr (default)
        self._rr = (self._rr + 1) % len(eligible)
        return eligible[self._rr]

    # ------------------------------------------------- single group w/ retry
    async def process_group(self, req: GroupRequest) -> GroupResult:
        """Generate one prompt group with token-level fault tolerance
        (handlers.rs:330-418)."""
        result = GroupResult(gid=req.gid,
                             samples=[SampleResult() for _ in range(req.n)])
        first_chunk = [True] * req.n

        def absorb(i: int, out: SampleResult):
            if first_chunk[i]:
                result.samples[i] = out
                first_chunk[i] = False
            else:
                result.samples[i] = merge_sample(result.samples[i], out)

        def enqueue_continuation(i: int, pending: List[tuple]):
            """Continue sample i token-exactly from its accumulated output:
            prompt + generated-so-far, remaining token budget
            (utils.rs:140-182, :256-291 capability, per sample)."""
            cont = continuation_request(req, i, result.samples[i])
            if cont.sampling.max_new_tokens > 0:
                pending.append((cont, [i]))
            else:
                result.samples[i].finish_reason = "length"

        pending: List[tuple] = [(req, list(range(req.n)))]
        attempts = 0
        while pending:
            sub, idxs = pending.pop(0)
            attempts += 1
            if attempts > self.cfg.max_retries * max(req.n, 1):
                for i in idxs:
                    result.samples[i].finish_reason = "error"
                continue
            inst = await self.next_instance()
            result.instance_ids.append(inst.instance_id)
            try:
                outs = await inst.generate_group(sub)
            except Exception as e:
                # instance failure: evict, absorb partial output, continue
                # each sample on another instance
                partials = getattr(e, "partials", None)
                if partials is None:
                    partials = [SampleResult() for _ in idxs]
                await self.evict_instance(inst.instance_id)
                for i, p in zip(idxs, partials):
                    absorb(i, SampleResult(
                        output_ids=list(p.output_ids),
                        output_logprobs=list(p.output_logprobs),
                        finish_reason="abort",
                        completion_tokens=len(p.output_ids)))
                    enqueue_continuation(i, pending)
                continue
            for i, out in zip(idxs, outs):
                absorb(i, out)
                if out.finish_reason == "abort":
                    enqueue_continuation(i, pending)
        return result

    # --------------------------------------------------------- batch stream
    async def submit_batch(self, groups: List[GroupRequest],
                           max_local_gen_s: Optional[float] = None
                           ) -> AsyncIterator:
        """Async stream: first item is the submit notifier (yielded once the
        local time-box expires or all work finishes), then each GroupResult
        as it completes (handlers.rs:442-564 NDJSON contract)."""
        await self.start()
        window = max_local_gen_s if max_local_gen_s is not None \
            else self.max_local_gen_s
        t0 = time.monotonic()
        tasks = [asyncio.ensure_future(self.process_group(g)) for g in groups]

        timebox_task = asyncio.ensure_future(
            self._local_timebox(window)) if window > 0 else None

        done_q: asyncio.Queue = asyncio.Queue()
        for t in tasks:
            t.add_done_callback(lambda fut: done_q.put_nowait(fut))

        notified = False
        if timebox_task is None:
            notified = True
            yield {"type": "notifier", "status": "success"}
        finished = 0
        try:
            while finished < len(tasks):
                get = asyncio.ensure_future(done_q.get())
                wait_for = [get] + ([timebox_task] if timebox_task and
                                    not notified else [])
                await asyncio.wait(wait_for,
                                   return_when=asyncio.FIRST_COMPLETED)
                if timebox_task and timebox_task.done() and not notified:
                    notified = True
                    yield {"type": "notifier", "status": "success"}
                if get.done():
                    fut = get.result()
                    finished += 1
                    if not notified:
                        # all local work may finish before the window
                        if finished == len(tasks):
                            if timebox_task:
                                timebox_task.cancel()
                            notified = True
                            yield {"type": "notifier", "status": "success"}
                    yield fut.result()
                else:
                    get.cancel()
        finally:
            if timebox_task and not timebox_task.done():
                timebox_task.cancel()
            self.last_batch_gen_s = time.monotonic() - t0

    async def _local_timebox(self, window: float):
        """After the window, deactivate local instances and abort their
        in-flight requests so trainer GPUs return to training
        (handlers.rs:500-513).  Only fires with remote capacity active."""
        await asyncio.sleep(window)
        if self.num_remote_active() == 0:
            return
        for iid in list(self._active):
            st = self._states.get(iid)
            if st is not None and st.inst.is_local:
                self._active.remove(iid)
                try:
                    st.inst.abort_all()
                except Exception:
                    pass
        await self._notify()

    def reactivate_local(self):
        for iid, st in self._states.items():
            if st.inst.is_local and iid not in self._active:
                self._active.append(iid)

    # --------------------------------------------------------- weight plane
    async def update_weight_version(self, version: int):
        """Bump latest version; clear active pool; re-add local instances
        (their weights are updated in place by the trainer before generation
        resumes) — handlers.rs:566-600."""
        assert version > self.latest_weight_version, \
            f"version must be monotonic ({version} <= {self.latest_weight_version})"
        self.latest_weight_version = version
        self._active.clear()
        for iid, st in self._states.items():
            if st.inst.is_local:
                st.weight_version = version
                self._active.append(iid)
        await self._notify()

    def _assign_weight_sender(self) -> Optional[str]:
        """Round-robin the registered sender endpoints over instances
        (reference state.rs:149-162 senders×groups rotation; our TCP
        engine carries the per-sender stream fan-out internally)."""
        if not self._weight_senders:
            return None
        ep = self._weight_senders[self._sender_rr % len(self._weight_senders)]
        self._sender_rr += 1
        return ep

    def update_weight_senders(self, senders: List[str]):
        """Replace the weight-sender endpoint registry and re-assign
        endpoints to every known instance (PUT /update_weight_senders,
        handlers.rs /update_weight_senders route)."""
        self._weight_senders = list(senders)
        self._sender_rr = 0
        for st in self._states.values():
            st.weight_sender_endpoint = self._assign_weight_sender()

    def weight_sender_for(self, instance_id: str) -> Optional[str]:
        st = self._states.get(instance_id)
        return st.weight_sender_endpoint if st is not None else None

    def get_receive_instances(self) -> List[RolloutInstance]:
        """Instances needing the latest weights; CAS-marks them updating
        (handlers.rs:602-649)."""
        out = []
        for st in self._states.values():
            if st.inst.is_local:
                continue
            if st.weight_version < self.latest_weight_version \
                    and not st.updating_weight:
                st.updating_weight = True
                out.append(st.inst)
        return out

    async def finish_weight_update(self, instance_id: str, version: int,
                                   success: bool):
        """Activation after transfer: on success set version and re-activate;
        on failure the instance stays out of the pool (handlers.rs:681-795)."""
        st = self._states.get(instance_id)
        if st is None:
            return
        if success:
            try:
                ok = await st.inst.update_weights(version)
            except Exception:
                ok = False
        else:
            ok = False
        st.updating_weight = False
        if ok and version >= self.latest_weight_version:
            st.weight_version = version
            if instance_id not in self._active:
                self._active.append(instance_id)
            await self._notify()

    # ------------------------------------------------------------- metrics
    def update_metrics(self, m: MetricsUpdate) -> dict:
        """Trainer feedback -> new local-gen window (handlers.rs:867-901)."""
        n_remote = sum(1 for s in self._states.values()
                       if not s.inst.is_local)
        self.max_local_gen_s = self.balance.update(
            m.step_time_s, m.trainer_bubble_time_s, m.step_throughput,
            n_remote)
        return {"new_max_gen_s": self.max_local_gen_s,
                "num_instances": len(self._states)}


class StreamingBatchIterator:
    """Sync facade over submit_batch for the SPMD trainer (reference:
    stream_batch_iter.py capability): runs the scheduler's asyncio loop on a
    background thread; ``next()`` blocks for the next finished group.  The
    first item is the submit notifier."""

    def __init__(self, scheduler: RolloutScheduler,
                 groups: List[GroupRequest],
                 max_local_gen_s: Optional[float] = None,
                 loop: Optional[asyncio.AbstractEventLoop] = None):
        self.scheduler = scheduler
        self._q: "asyncio.Queue" = None
        self._sentinel = object()
        self._items: "list" = []
        import queue as _queue
        self._out = _queue.Queue()
        self._thread_loop = loop
        if loop is None:
            self._thread = threading.Thread(
                target=self._run, args=(groups, max_local_gen_s), daemon=True)
            self._thread.start()
        else:
            asyncio.run_coroutine_threadsafe(
                self._pump(groups, max_local_gen_s), loop)

    def _run(self, groups, window):
        asyncio.run(self._pump(groups, window))

    async def _pump(self, groups, window):
        try:
            async for item in self.scheduler.submit_batch(groups, window):
                self._out.put(item)
        except Exception as e:  # surface errors to the consumer
            self._out.put(e)
        finally:
            self._out.put(self._sentinel)

    def __iter__(self):
        return self

    def __next__(self):
        item = self._out.get()
        if item is self._sentinel:
            raise StopIteration
        if isinstance(item, Exception):
            raise item
        return item
