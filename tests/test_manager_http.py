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


def test_weight_sender_registry():
    """PUT /update_weight_senders + round-robin assignment over instances
    (handlers.rs PUT route, state.rs:149-162): senders rotate across
    registrations and re-assign on registry replacement."""
    async def go():
        sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.02))
        app = create_manager_app(sched)
        import httpx
        async with httpx.AsyncClient(
                transport=httpx.ASGITransport(app=app),
                base_url="http://mgr") as c:
            # before any senders: registration yields no endpoint
            await sched.register_instance(FakeInstance("a", is_local=False),
                                          skip_health_check=True)
            assert sched.weight_sender_for("a") is None

            r = await c.put("/update_weight_senders",
                            json={"senders": ["10.0.0.1:7000",
                                              "10.0.0.2:7000"]})
            assert r.json()["num_senders"] == 2
            # replacement re-assigns existing instances round-robin
            assert sched.weight_sender_for("a") == "10.0.0.1:7000"

            await sched.register_instance(FakeInstance("b", is_local=False),
                                          skip_health_check=True)
            await sched.register_instance(FakeInstance("d", is_local=False),
                                          skip_health_check=True)
            assert sched.weight_sender_for("b") == "10.0.0.2:7000"
            assert sched.weight_sender_for("d") == "10.0.0.1:7000"

            d = (await c.get("/get_instances_status")).json()
            eps = {i["id"]: i["weight_sender_endpoint"]
                   for i in d["instances"]}
            assert eps == {"a": "10.0.0.1:7000", "b": "10.0.0.2:7000",
                           "d": "10.0.0.1:7000"}

            # receive-instances response carries the per-instance sender
            await c.post("/update_weight_version", json={"version": 1})
            r = (await c.post("/get_receive_instances")).json()
            assert set(r["instances"]) == {"a", "b", "d"}
            assert r["senders"]["b"] == "10.0.0.2:7000"
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


def test_remote_instance_self_registration_e2e(tmp_path):
    """The reference's elastic-join lifecycle (§3.4 / launch_sglang.sh
    capability): a REAL `engine_server` subprocess started with --manager
    self-registers against a live manager facade; the manager then serves
    a /generate relayed through the joined instance."""
    import asyncio
    import subprocess
    import sys
    import time

    import httpx

    from polyrl_amd.scheduler.manager import RolloutScheduler, SchedulerConfig
    from polyrl_amd.scheduler.http_api import serve_manager

    loop = asyncio.new_event_loop()
    import threading
    threading.Thread(target=loop.run_forever, daemon=True).start()
    sched = RolloutScheduler(SchedulerConfig(stats_interval_s=0.1,
                                             health_check_timeout_s=60.0))
    mgr_port = 31930
    server = serve_manager(sched, host="127.0.0.1", port=mgr_port, loop=loop)
    for _ in range(100):
        try:
            if httpx.get(f"http://127.0.0.1:{mgr_port}/health",
                         timeout=1.0).status_code == 200:
                break
        except Exception:
            time.sleep(0.05)

    import os
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.Popen(
        [sys.executable, "-m", "polyrl_amd.server.engine_server",
         "--model", "llama-debug-cpu", "--host", "127.0.0.1",
         "--port", "31931", "--kv-gb", "0.02",
         "--manager", f"http://127.0.0.1:{mgr_port}",
         "--advertise-addr", "127.0.0.1:31931"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        # join is health-gated on the manager side
        deadline = time.monotonic() + 180   # model init under suite load
        while time.monotonic() < deadline:
            if sched.num_active() >= 1:
                break
            time.sleep(0.2)
        assert sched.num_active() >= 1, "instance never joined the pool"
        # generate THROUGH the manager facade (relay route)
        r = httpx.post(f"http://127.0.0.1:{mgr_port}/generate",
                       json={"input_ids": [5, 6, 7],
                             "sampling_params": {"n": 1,
                                                 "max_new_tokens": 4},
                             "return_logprob": True}, timeout=60.0)
        assert r.status_code == 200, r.text
        out = r.json()
        sample = out["samples"][0]
        assert len(sample["output_ids"]) == 4
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
        server.should_exit = True
