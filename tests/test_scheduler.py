"""Scheduler contract tests (CPU tier): zero-queue RR dispatch, token-level
continuation on instance failure, local time-box abort + remote tail,
weight-version gating, adaptive balance — the behavioral contract of the
reference Rust rollout-manager."""
import asyncio

import pytest

from polyrl_amd.scheduler import (FakeInstance, GroupRequest, LoadBalanceState,
                                  RolloutScheduler, SchedulerConfig)
from polyrl_amd.scheduler.manager import StreamingBatchIterator
from polyrl_amd.scheduler.types import MetricsUpdate, SamplingSpec


def run(coro):
    return asyncio.run(coro)


def mk_req(gid=0, n=2, max_new=8, prompt=None):
    return GroupRequest(gid=gid, input_ids=prompt or [5, 6, 7], n=n,
                        sampling=SamplingSpec(max_new_tokens=max_new))


def fake_expected(prompt, max_new):
    """FakeInstance echo: token t = (prompt[-1] + 1 + t) % 50000."""
    return [(prompt[-1] + 1 + t) % 50000 for t in range(max_new)]


def test_basic_group_generation():
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        inst = FakeInstance("i0")
        await sched.register_instance(inst, skip_health_check=True)
        res = await sched.process_group(mk_req(n=3, max_new=6))
        await sched.close()
        assert len(res.samples) == 3
        for s in res.samples:
            assert s.output_ids == fake_expected([5, 6, 7], 6)
            assert s.finish_reason == "length"
            assert len(s.output_logprobs) == 6
            assert s.num_migrations == 0
    run(go())


def test_token_level_continuation_on_failure():
    """Kill the serving instance mid-generation; the sample must continue
    token-exactly on another instance (handlers.rs:330-418)."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        bad = FakeInstance("bad", fail_after_tokens=3)
        good = FakeInstance("good")
        await sched.register_instance(bad, skip_health_check=True)
        await sched.register_instance(good, skip_health_check=True)
        # force the bad instance to serve first
        sched._rr = len(sched._active) - 1
        prompt = [10, 11]
        res = await sched.process_group(mk_req(n=1, max_new=8, prompt=prompt))
        await sched.close()
        s = res.samples[0]
        assert len(s.output_ids) == 8, s
        # first 3 tokens from bad, then continuation: prompt' = prompt + 3 toks
        first = fake_expected(prompt, 3)
        cont = fake_expected(prompt + first, 5)
        assert s.output_ids == first + cont
        assert len(s.output_logprobs) == 8
        assert s.num_migrations == 1
        assert s.finish_reason == "length"
        # bad instance evicted from the registry
        assert "bad" not in [i.instance_id for i in sched.instances()]
    run(go())


def test_retry_cap_gives_error():
    async def go():
        cfg = SchedulerConfig(stats_interval_s=0.02, max_retries=2)
        sched = RolloutScheduler(cfg)
        # every instance dies immediately; with instant re-registration the
        # retry cap must stop the loop
        for k in range(6):
            await sched.register_instance(
                FakeInstance(f"f{k}", fail_after_tokens=0),
                skip_health_check=True)
        res = await sched.process_group(mk_req(n=1, max_new=4))
        await sched.close()
        assert res.samples[0].finish_reason in ("error", "abort")
    run(go())


def test_local_timebox_aborts_and_remote_continues():
    """Submit with a tiny window: the slow local instance is aborted and
    deactivated; the tail continues on the remote instance; the notifier is
    the first streamed item (handlers.rs:442-564, :500-513)."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        local = FakeInstance("local", is_local=True, token_time_s=0.05)
        remote = FakeInstance("remote", is_local=False)
        await sched.register_instance(local, skip_health_check=True)
        await sched.register_instance(remote, skip_health_check=True)
        # force dispatch to the local instance first
        sched._states["remote"].assigned_batches = 10**9
        items = []
        async for it in sched.submit_batch([mk_req(n=1, max_new=20)],
                                           max_local_gen_s=0.12):
            items.append(it)
        await sched.close()
        assert items[0] == {"type": "notifier", "status": "success"}
        res = items[1]
        s = res.samples[0]
        assert len(s.output_ids) == 20          # completed despite the abort
        assert s.num_migrations >= 1            # local -> remote continuation
        assert "remote" in res.instance_ids
        # local removed from the active pool by the time-box
        assert "local" not in sched._active
        sched.reactivate_local()
        assert "local" in sched._active
    run(go())


def test_timebox_noop_without_remote():
    """With only local capacity the time-box must not strand the batch."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        local = FakeInstance("l0", is_local=True, token_time_s=0.01)
        await sched.register_instance(local, skip_health_check=True)
        items = []
        async for it in sched.submit_batch([mk_req(n=1, max_new=10)],
                                           max_local_gen_s=0.03):
            items.append(it)
        await sched.close()
        res = [i for i in items if not isinstance(i, dict)][0]
        assert len(res.samples[0].output_ids) == 10
        assert res.samples[0].finish_reason == "length"
    run(go())


def test_round_robin_spread():
    async def go():
        sched = RolloutScheduler(SchedulerConfig(
            stats_interval_s=0.02, max_assigned_batches_per_stats_check=100))
        insts = [FakeInstance(f"i{k}") for k in range(4)]
        for i in insts:
            await sched.register_instance(i, skip_health_check=True)
        for g in range(8):
            await sched.process_group(mk_req(gid=g, n=1, max_new=2))
        await sched.close()
        served = [len(i.served_gids) for i in insts]
        assert sum(served) == 8
        assert max(served) - min(served) <= 1, served  # even spread
    run(go())


def test_weight_version_gating():
    """Version bump clears the pool and re-adds local; remote instances
    rejoin only after a successful update (handlers.rs:566-795)."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        local = FakeInstance("loc", is_local=True)
        r_ok = FakeInstance("r_ok", is_local=False)
        r_bad = FakeInstance("r_bad", is_local=False)
        for i in (local, r_ok, r_bad):
            await sched.register_instance(i, skip_health_check=True)
        assert sched.num_active() == 3

        await sched.update_weight_version(1)
        assert sched._active == ["loc"]       # only local re-added

        recv = sched.get_receive_instances()
        assert {i.instance_id for i in recv} == {"r_ok", "r_bad"}
        # CAS: a second call returns nothing while updating
        assert sched.get_receive_instances() == []

        r_bad.healthy = False                  # its update will fail
        await sched.finish_weight_update("r_ok", 1, success=True)
        await sched.finish_weight_update("r_bad", 1, success=True)
        assert set(sched._active) == {"loc", "r_ok"}
        assert r_ok.weight_version == 1
        # stale instance never rejoins until a later successful update
        assert "r_bad" not in sched._active
        # monotonicity enforced
        with pytest.raises(AssertionError):
            await sched.update_weight_version(1)
        await sched.close()
    run(go())


def test_streaming_iterator_sync_facade():
    sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))

    async def setup():
        await sched.register_instance(FakeInstance("i0"),
                                      skip_health_check=True)
    asyncio.run(setup())
    groups = [mk_req(gid=g, n=2, max_new=4) for g in range(3)]
    it = StreamingBatchIterator(sched, groups, max_local_gen_s=0)
    items = list(it)
    assert items[0]["type"] == "notifier"
    results = items[1:]
    assert sorted(r.gid for r in results) == [0, 1, 2]
    for r in results:
        assert all(len(s.output_ids) == 4 for s in r.samples)


def test_balance_adapts_window():
    b = LoadBalanceState(initial_gen_s=100.0)
    # trainer bubble large vs remote bubble -> grow local window
    w1 = b.update(step_time_s=200.0, trainer_bubble_s=80.0,
                  step_throughput=10.0, num_instances=2)
    w2 = b.update(step_time_s=200.0, trainer_bubble_s=80.0,
                  step_throughput=10.0, num_instances=2)
    assert w2 > 100.0
    # trainer bubble tiny vs remote bubble -> shrink, floored at 5 s
    b2 = LoadBalanceState(initial_gen_s=30.0)
    b2.update(1000.0, 0.0, 5.0, 3)
    for _ in range(50):
        w = b2.update(1000.0, 0.0, 5.0, 3)
    assert w == pytest.approx(5.0)


def test_scheduler_metrics_feedback():
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        await sched.register_instance(FakeInstance("r0", is_local=False),
                                      skip_health_check=True)
        out = sched.update_metrics(MetricsUpdate(
            step_time_s=100.0, trainer_bubble_time_s=10.0,
            step_throughput=5.0))
        assert "new_max_gen_s" in out and out["num_instances"] == 1
        await sched.close()
    run(go())


def test_http_stream_failure_continuation():
    """HTTP-layer fault injection: an instance whose SSE stream dies
    mid-generation surfaces partials; the scheduler evicts it and the
    sample continues token-exactly on a healthy instance
    (handlers.rs:152-328 partial capture + :363-414 continuation)."""
    import httpx

    class CutTransport(httpx.AsyncBaseTransport):
        """ASGI transport that breaks every /generate response stream."""

        def __init__(self, inner):
            self.inner = inner

        async def handle_async_request(self, request):
            resp = await self.inner.handle_async_request(request)
            if request.url.path == "/generate":
                class CutStream(httpx.AsyncByteStream):
                    async def __aiter__(self):
                        raise httpx.ReadError("injected stream cut")
                        yield b""  # pragma: no cover (makes it a generator)
                return httpx.Response(resp.status_code,
                                      headers=resp.headers,
                                      stream=CutStream(),
                                      request=request)
            return resp

    async def go():
        import torch

        from polyrl_amd.models import create_model, get_model_config
        from polyrl_amd.rollout.engine import Engine
        from polyrl_amd.server import HttpInstance, create_app
        cfg = get_model_config("llama-debug-cpu")
        torch.manual_seed(3)
        model = create_model(cfg, kind="actor", dtype="float32", device="cpu")
        eng = Engine(cfg, device="cpu", dtype=torch.float32,
                     kv_bytes_budget=8 << 20)
        eng.model.load_state_dict(model.state_dict())
        app = create_app(eng)
        flaky = HttpInstance(
            "http://flaky", transport=CutTransport(httpx.ASGITransport(app=app)))
        good = FakeInstance("good", is_local=False)
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        await sched.register_instance(flaky, skip_health_check=True)
        await sched.register_instance(good, skip_health_check=True)
        sched._states["good"].assigned_batches = 10**9  # flaky serves first
        res = await sched.process_group(mk_req(n=1, max_new=6,
                                               prompt=[4, 5]))
        await sched.close()
        s = res.samples[0]
        assert len(s.output_ids) == 6          # completed despite the cut
        assert s.finish_reason == "length"
        assert s.num_migrations == 1
        assert "good" in res.instance_ids      # continuation landed there
        assert "http://flaky" not in [i.instance_id
                                      for i in sched.instances()]  # evicted
    asyncio.run(go())


def test_scheduling_policy_interface():
    """Pluggable dispatch policy (the reference's 'algorithm-driven request
    scheduling' interface): least_loaded prefers the idle instance; a
    custom callable is honored verbatim."""
    async def go():
        # least_loaded: instance with fewer running samples wins
        sched = RolloutScheduler(SchedulerConfig(
            scheduling_policy="least_loaded"))
        a = FakeInstance("a")
        b = FakeInstance("b")
        await sched.register_instance(a, skip_health_check=True)
        await sched.register_instance(b, skip_health_check=True)
        sched._states["a"].stats.num_running = 5
        sched._states["b"].stats.num_running = 0
        picked = await sched.next_instance()
        assert picked.instance_id == "b"

        # custom callable
        sched2 = RolloutScheduler(SchedulerConfig(
            scheduling_policy=lambda elig, states, rr: sorted(elig)[-1]))
        await sched2.register_instance(FakeInstance("x"),
                                       skip_health_check=True)
        await sched2.register_instance(FakeInstance("z"),
                                       skip_health_check=True)
        assert (await sched2.next_instance()).instance_id == "z"
        await sched.close()
        await sched2.close()
    run(go())


def test_health_gate_rejects_dead_instance():
    """register_instance health-gates joins: an instance that never
    responds healthy raises after the (short) timeout and never enters
    the pool (instance_manager.rs:5-37)."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(
            health_check_timeout_s=0.2, health_check_interval_s=0.05))
        dead = FakeInstance("dead")
        dead.healthy = False

        import pytest as _pytest
        with _pytest.raises(TimeoutError):
            await sched.register_instance(dead)
        assert sched.num_active() == 0
        # a healthy one joins through the same gate
        ok = FakeInstance("ok")
        await sched.register_instance(ok)
        assert sched.num_active() == 1
        await sched.close()
    run(go())
