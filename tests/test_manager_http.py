"""Scheduler HTTP facade (ASGI, no sockets): the reference manager's route
surface over a live RolloutScheduler (rollout-manager/src/main.rs:56-70)."""
import asyncio

import pytest

from polyrl_amd.scheduler import (FakeInstance, RolloutScheduler,
                                  SchedulerConfig)
from polyrl_amd.scheduler.http_api import create_manager_app


def test_manager_routes():
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        await sched.register_instance(FakeInstance("loc", is_local=True),
                                      skip_health_check=True)
        await sched.register_instance(FakeInstance("rem", is_local=False),
                                      skip_health_check=True)
        app = create_manager_app(sched)
        import httpx
        async with httpx.AsyncClient(
                transport=httpx.ASGITransport(app=app),
                base_url="http://mgr") as c:
            r = await c.get("/health")
            assert r.json()["status"] == "ok"

            r = await c.get("/get_instances_status")
            d = r.json()
            assert len(d["instances"]) == 2
            assert d["latest_weight_version"] == 0

            # version bump clears remote from the pool
            r = await c.post("/update_weight_version", json={"version": 1})
            assert r.status_code == 200
            d = (await c.get("/get_instances_status")).json()
            by_id = {i["id"]: i for i in d["instances"]}
            assert by_id["loc"]["active"] and not by_id["rem"]["active"]

            # transfer-plane negotiation: CAS receive list, then activate
            r = await c.post("/get_receive_instances")
            assert r.json()["instances"] == ["rem"]
            r = await c.post("/update_weights",
                             json={"version": 1, "instances": ["rem"]})
            assert r.json()["results"]["rem"]
            d = (await c.get("/get_instances_status")).json()
            assert all(i["active"] for i in d["instances"])

            # metrics feedback returns the new local time-box
            r = await c.post("/update_metrics", json={
                "step_time_s": 100.0, "trainer_bubble_time_s": 5.0,
                "step_throughput": 3.0})
            assert "new_max_gen_s" in r.json()

            # abort + eviction
            r = await c.post("/abort_local_requests")
            assert r.json()["status"] == "ok"
            r = await c.post("/shutdown_instances",
                             json={"instances": ["rem"]})
            assert r.json()["status"] == "ok"
            d = (await c.get("/get_instances_status")).json()
            assert [i["id"] for i in d["instances"]] == ["loc"]
        await sched.close()
    asyncio.run(go())


def test_manager_generate_and_batch_stream():
    """The manager's generation surface (handlers.rs /generate +
    /batch_generate_requests NDJSON contract): single group relay and the
    streamed batch with notifier-first framing."""
    import json

    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        for i in range(2):
            await sched.register_instance(
                FakeInstance(f"i{i}", token_time_s=0.001),
                skip_health_check=True)
        app = create_manager_app(sched)
        import httpx
        async with httpx.AsyncClient(
                transport=httpx.ASGITransport(app=app),
                base_url="http://mgr") as c:
            # single group
            r = await c.post("/generate", json={
                "input_ids": [5, 6, 7],
                "sampling_params": {"n": 2, "max_new_tokens": 4}})
            d = r.json()
            assert len(d["samples"]) == 2
            exp = [(7 + 1 + t) % 50000 for t in range(4)]
            assert d["samples"][0]["output_ids"] == exp
            assert len(d["samples"][0]["output_logprobs"]) == 4

            # streamed batch: NDJSON, notifier first, one line per group
            reqs = [[g, {"input_ids": [11], "sampling_params":
                         {"n": 1, "max_new_tokens": 3}}] for g in range(4)]
            lines = []
            async with c.stream("POST", "/batch_generate_requests",
                                json={"requests": reqs,
                                      "max_local_gen_s": 0}) as resp:
                assert resp.status_code == 200
                async for line in resp.aiter_lines():
                    if line.strip():
                        lines.append(json.loads(line))
            assert lines[0].get("type") == "notifier"
            gids = sorted(x["gid"] for x in lines[1:])
            assert gids == [0, 1, 2, 3]
            exp = [(11 + 1 + t) % 50000 for t in range(3)]
            for x in lines[1:]:
                assert x["samples"][0]["output_ids"] == exp
        await sched.close()
    asyncio.run(go())
