"""Concurrency-stress tier for the in-process scheduler and the engine's
update-vs-generate writer lock (the reference's safety is
by construction — single-writer active pool, version CAS, writer lock —
and the new build should add a race/stress test tier for them).

These tests hammer interleavings rather than assert on timing: every
outcome checked is an invariant that must hold under ANY schedule.
"""
from __future__ import annotations

import asyncio
import random
import threading

import pytest
import torch

from polyrl_amd.scheduler import (FakeInstance, GroupRequest,
                                  RolloutScheduler, SchedulerConfig)
from polyrl_amd.scheduler.types import SamplingSpec


def mk_req(gid, n=2, max_new=6, prompt=None):
    return GroupRequest(gid=gid, input_ids=prompt or [5, 6, 7], n=n,
                        sampling=SamplingSpec(max_new_tokens=max_new))


def fake_expected(prompt, max_new):
    return [(prompt[-1] + 1 + t) % 50000 for t in range(max_new)]


@pytest.mark.timeout(120)
def test_stream_completes_under_instance_churn():
    """30 groups streaming while instances are evicted and (re)registered
    concurrently: every group must come back complete and correct —
    deadlock-freedom of the notify/admission loop under churn
    (SURVEY.md §6 'deadlock-freedom ... under instance churn')."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(max_assigned_batches_per_stats_check=4))
        for i in range(3):
            await sched.register_instance(
                FakeInstance(f"base-{i}", token_time_s=0.001),
                skip_health_check=True)

        stop = asyncio.Event()

        async def churn():
            k = 0
            rng = random.Random(0)
            while not stop.is_set():
                k += 1
                iid = f"churn-{k}"
                await sched.register_instance(
                    FakeInstance(iid, token_time_s=0.001),
                    skip_health_check=True)
                await asyncio.sleep(rng.uniform(0, 0.004))
                await sched.evict_instance(iid, shutdown=False)
                await asyncio.sleep(0)

        churner = asyncio.ensure_future(churn())
        groups = [mk_req(g, n=2, max_new=6) for g in range(30)]
        got = {}
        async for item in sched.submit_batch(groups):
            if isinstance(item, dict):      # notifier
                continue
            got[item.gid] = item
        stop.set()
        churner.cancel()
        assert len(got) == 30
        exp = fake_expected([5, 6, 7], 6)
        for g, res in got.items():
            assert len(res.samples) == 2
            for s in res.samples:
                assert s.output_ids == exp, (g, s.output_ids)
        await sched.close()
    asyncio.run(go())


@pytest.mark.timeout(120)
def test_continuation_correct_under_churn_with_failures():
    """Failing instances mid-stream + churn: token-level continuation must
    still assemble exact outputs (no token lost or duplicated)."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(max_assigned_batches_per_stats_check=4))
        # two flaky instances that die after 2 tokens, two good ones
        for i in range(2):
            await sched.register_instance(
                FakeInstance(f"flaky-{i}", token_time_s=0.001,
                             fail_after_tokens=2),
                skip_health_check=True)
        for i in range(2):
            await sched.register_instance(
                FakeInstance(f"good-{i}", token_time_s=0.001),
                skip_health_check=True)
        groups = [mk_req(g, n=2, max_new=6) for g in range(12)]
        got = {}
        async for item in sched.submit_batch(groups):
            if isinstance(item, dict):
                continue
            got[item.gid] = item
        assert len(got) == 12
        exp = fake_expected([5, 6, 7], 6)
        for res in got.values():
            for s in res.samples:
                assert s.output_ids == exp, s.output_ids
        await sched.close()
    asyncio.run(go())


@pytest.mark.timeout(120)
def test_two_batches_interleaved_no_crosstalk():
    """Two submit_batch streams over the same pool concurrently: results
    must partition exactly by batch (single-writer pool invariant)."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(max_assigned_batches_per_stats_check=3))
        for i in range(3):
            await sched.register_instance(
                FakeInstance(f"i{i}", token_time_s=0.001),
                skip_health_check=True)

        async def drain(groups):
            out = {}
            async for item in sched.submit_batch(groups):
                if isinstance(item, dict):
                    continue
                out[item.gid] = item
            return out

        a = asyncio.ensure_future(
            drain([mk_req(g, prompt=[11], max_new=5) for g in range(10)]))
        b = asyncio.ensure_future(
            drain([mk_req(100 + g, prompt=[31], max_new=4)
                   for g in range(10)]))
        ra, rb = await asyncio.gather(a, b)
        assert sorted(ra) == list(range(10))
        assert sorted(rb) == [100 + g for g in range(10)]
        for res in ra.values():
            for s in res.samples:
                assert s.output_ids == fake_expected([11], 5)
        for res in rb.values():
            for s in res.samples:
                assert s.output_ids == fake_expected([31], 4)
        await sched.close()
    asyncio.run(go())


@pytest.mark.timeout(120)
def test_receive_instances_cas_single_winner():
    """get_receive_instances is a CAS hand-out: under many concurrent
    claimants each stale remote is handed to EXACTLY one
    (handlers.rs:602-649 updating_weight CAS contract)."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig())
        remotes = [FakeInstance(f"r{i}", is_local=False) for i in range(8)]
        for r in remotes:
            await sched.register_instance(r, skip_health_check=True)
        await sched.update_weight_version(1)

        claimed: list = []

        async def claimant():
            # interleave with other claimants at await points
            await asyncio.sleep(0)
            got = sched.get_receive_instances()
            claimed.extend(got)

        await asyncio.gather(*[claimant() for _ in range(16)])
        ids = [i.instance_id for i in claimed]
        assert sorted(ids) == sorted(f"r{i}" for i in range(8)), ids
        assert len(set(ids)) == len(ids), f"double hand-out: {ids}"
        # finishing re-activates each exactly once
        for r in remotes:
            await sched.finish_weight_update(r.instance_id, 1, success=True)
        assert sched.num_remote_active() == 8
        await sched.close()
    asyncio.run(go())


@pytest.mark.timeout(180)
def test_engine_update_weights_vs_generation_writer_lock():
    """EngineRunner: weight swaps from another thread while the pump thread
    generates.  The writer-lock contract (patches.py:482 capability) means
    every completed sample was produced by exactly one weight version —
    greedy outputs must match one of the per-version reference outputs,
    never a mixture."""
    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.rollout.engine import Engine, SamplingParams
    from polyrl_amd.rollout.runner import EngineRunner

    cfg = get_model_config("llama-debug-cpu")
    torch.manual_seed(0)
    m0 = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    sd0 = {k: v.clone() for k, v in m0.state_dict().items()}
    sd1 = {k: (v + 0.05 * torch.randn_like(v)) for k, v in sd0.items()}

    prompt = [5, 9, 2, 7]
    sp = SamplingParams(temperature=0.0, max_new_tokens=6)

    # per-version greedy references
    refs = []
    for sd in (sd0, sd1):
        e = Engine(cfg, device="cpu", dtype=torch.float32,
                   kv_bytes_budget=8 << 20)
        e.model.load_state_dict(sd)
        refs.append(e.generate([prompt], sp, "ref")[0].output_ids)

    eng = Engine(cfg, device="cpu", dtype=torch.float32,
                 kv_bytes_budget=16 << 20)
    eng.model.load_state_dict(sd0)
    runner = EngineRunner(eng)
    runner.start()

    async def drive():
        futs = [runner.submit(prompt, sp, rid=f"q{i}") for i in range(24)]

        def swapper():     # real cross-thread race against the pump thread
            for v in range(1, 9):
                runner.update_weights(sd1 if v % 2 else sd0, version=v,
                                      abort_in_flight=False)
        th = threading.Thread(target=swapper)
        th.start()
        outs = await asyncio.gather(*futs)
        th.join()
        return outs

    try:
        outs = asyncio.run(drive())
    finally:
        runner.stop()
    for o in outs:
        assert o.output_ids in refs, \
            f"torn generation (mixed weight versions): {o.output_ids}"
    assert runner.weight_version == 8


@pytest.mark.timeout(180)
def test_runner_admin_fuzz_no_deadlock_no_lost_futures():
    """Seeded randomized interleaving of the runner's whole admin surface
    (submit / abort(rid) / abort_all / update_weights with and without
    in-flight aborts / stats) from several OS threads while the pump
    thread generates.  Invariants under ANY schedule: every submitted
    future resolves with a legal finish_reason and bounded length, nothing
    deadlocks, and the runner still serves a fresh request afterwards.
    (The round-2 lock-starvation bug lived exactly here: admin callers
    were starved by the non-FIFO pump lock for whole generations.)"""
    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.rollout.engine import Engine, SamplingParams
    from polyrl_amd.rollout.runner import EngineRunner

    cfg = get_model_config("llama-debug-cpu")
    torch.manual_seed(1)
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    sd0 = {k: v.clone() for k, v in m.state_dict().items()}
    sd1 = {k: (v + 0.03 * torch.randn_like(v)) for k, v in sd0.items()}

    eng = Engine(cfg, device="cpu", dtype=torch.float32,
                 kv_bytes_budget=16 << 20)
    eng.model.load_state_dict(sd0)
    runner = EngineRunner(eng)
    runner.start()

    N = 48
    sp = SamplingParams(temperature=0.0, max_new_tokens=8)
    stop_admins = threading.Event()

    def admin(seed):
        rng = random.Random(seed)
        v = 100 * seed
        while not stop_admins.is_set():
            op = rng.random()
            if op < 0.35:
                runner.abort(rid=f"f{rng.randrange(N)}")
            elif op < 0.45:
                runner.abort(abort_all=True)
            elif op < 0.70:
                v += 1
                runner.update_weights(sd1 if v % 2 else sd0, version=v,
                                      abort_in_flight=bool(rng.random() < .5))
            else:
                runner.stats()
            # tiny jitter so ops land at varied pump phases
            if rng.random() < 0.5:
                import time as _t
                _t.sleep(rng.random() * 0.002)

    async def drive():
        futs = [runner.submit([3 + i % 7, 11, 4], sp, rid=f"f{i}")
                for i in range(N)]
        threads = [threading.Thread(target=admin, args=(s,), daemon=True)
                   for s in (1, 2, 3)]
        for t in threads:
            t.start()
        try:
            outs = await asyncio.wait_for(asyncio.gather(*futs), timeout=90)
        finally:
            stop_admins.set()
            for t in threads:
                t.join(timeout=10)
        assert not any(t.is_alive() for t in threads), "admin thread hung"
        return outs

    try:
        outs = asyncio.run(drive())
        for o in outs:
            assert o.finish_reason in ("stop", "length", "abort"), \
                o.finish_reason
            assert len(o.output_ids) <= sp.max_new_tokens
        # runner still serviceable after the storm
        async def one_more():
            return await asyncio.wait_for(
                asyncio.gather(*[runner.submit([9, 9, 9], sp, rid="post")]),
                timeout=30)
        post = asyncio.run(one_more())[0]
        assert post.finish_reason in ("stop", "length")
        assert len(post.output_ids) >= 1
    finally:
        runner.stop()
