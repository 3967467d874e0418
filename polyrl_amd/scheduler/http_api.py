"""HTTP facade for the in-process scheduler.

The reference rollout-manager is an HTTP service (rollout-manager/src/
main.rs:56-70 — 13 axum routes).  Co-located trainers call the scheduler
in-process (no hop), but ELASTIC REMOTE instances still need an HTTP
surface to join a running job (§3.4 lifecycle) and operators need the
status/metrics routes.  This app wraps a live RolloutScheduler with the
reference's route names.
"""
from __future__ import annotations

import asyncio
from typing import Optional

from fastapi import FastAPI, Request


def create_manager_app(scheduler, loop: Optional[asyncio.AbstractEventLoop]
                       = None, remote_weight_state_fn=None) -> "FastAPI":
    """App over a RolloutScheduler.  If the scheduler's asyncio loop is a
    different thread's loop, pass it so mutations run there.  Newly joining
    remote instances get ``remote_weight_state_fn`` as their TCP-plane
    weight source (version -> state dict)."""
    app = FastAPI(title="polyrl-amd rollout manager")
    app.state.scheduler = scheduler

    async def _run(coro):
        if loop is None or loop is asyncio.get_running_loop():
            return await coro
        return await asyncio.wrap_future(
            asyncio.run_coroutine_threadsafe(coro, loop))

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.post("/register_rollout_instance")
    async def register_rollout_instance(request: Request):
        """A remote instance announces itself (handlers.rs:40-86): health-
        gate it, then it joins once its weight version is current."""
        body = await request.json()
        addr = body["addr"]
        from ..server import HttpInstance
        inst = HttpInstance(addr, weight_state_fn=remote_weight_state_fn)
        await _run(scheduler.register_instance(inst))
        return {"status": "registered", "instance_id": inst.instance_id,
                "weight_version": scheduler.latest_weight_version,
                "weight_sender_endpoint":
                    scheduler.weight_sender_for(inst.instance_id)}

    @app.post("/register_local_rollout_instances")
    async def register_local_rollout_instances(request: Request):
        """Batch-register co-located engines by address
        (handlers.rs route; local = trainer-GPU engines subject to the
        local time-box)."""
        body = await request.json()
        from ..server import HttpInstance
        ids = []
        for addr in body["addrs"]:
            inst = HttpInstance(addr, is_local=True,
                                weight_state_fn=remote_weight_state_fn)
            await _run(scheduler.register_instance(inst))
            ids.append(inst.instance_id)
        return {"status": "registered", "instance_ids": ids}

    def _parse_group(gid, body) -> "GroupRequest":
        sp = body.get("sampling_params", {})
        from .types import GroupRequest, SamplingSpec
        return GroupRequest(
            gid=int(gid), input_ids=list(body["input_ids"]),
            n=int(sp.get("n", 1)),
            sampling=SamplingSpec(
                temperature=float(sp.get("temperature", 1.0)),
                top_k=int(sp.get("top_k", -1)),
                top_p=float(sp.get("top_p", 1.0)),
                max_new_tokens=int(sp.get("max_new_tokens", 128)),
                stop_token_ids=tuple(sp.get("stop_token_ids", ()))),
            return_logprob=bool(body.get("return_logprob", True)))

    def _sample_json(smp):
        return {"output_ids": smp.output_ids,
                "output_logprobs": smp.output_logprobs,
                "finish_reason": smp.finish_reason,
                "completion_tokens": smp.completion_tokens,
                "num_migrations": smp.num_migrations}

    @app.post("/generate")
    async def generate(request: Request):
        """Relay one prompt group through the scheduler (continuation /
        RR / version gating apply — handlers.rs:330-418)."""
        body = await request.json()
        req = _parse_group(body.get("gid", 0), body)
        res = await _run(scheduler.process_group(req))
        return {"gid": res.gid, "instance_ids": res.instance_ids,
                "samples": [_sample_json(x) for x in res.samples]}

    @app.post("/batch_generate_requests")
    async def batch_generate_requests(request: Request):
        """The reference's NDJSON streaming contract
        (handlers.rs:442-564): body [[gid, generate_request], ...] (+
        optional max_local_gen_s); response streams one JSON line per
        completed group as it finishes, first line = submit notifier."""
        import json as _json

        from fastapi.responses import StreamingResponse
        body = await request.json()
        pairs = body if isinstance(body, list) else body["requests"]
        window = None if isinstance(body, list) \
            else body.get("max_local_gen_s")
        groups = [_parse_group(gid, g) for gid, g in pairs]

        import queue as _q
        out_q: _q.Queue = _q.Queue()

        async def pump():
            try:
                async for item in scheduler.submit_batch(
                        groups, max_local_gen_s=window):
                    if isinstance(item, dict):
                        out_q.put(_json.dumps(item))
                    else:
                        out_q.put(_json.dumps(
                            {"gid": item.gid,
                             "instance_ids": item.instance_ids,
                             "samples": [_sample_json(x)
                                         for x in item.samples]}))
            finally:
                out_q.put(None)

        if loop is None or loop is asyncio.get_running_loop():
            # keep a strong ref: asyncio tasks are weakly referenced and
            # an unreferenced pump can be garbage-collected mid-stream
            t = asyncio.ensure_future(pump())
            _bg = getattr(app.state, "_bg_tasks", None)
            if _bg is None:
                _bg = app.state._bg_tasks = set()
            _bg.add(t)
            t.add_done_callback(_bg.discard)
        else:
            asyncio.run_coroutine_threadsafe(pump(), loop)

        async def lines():
            while True:
                try:
                    item = out_q.get_nowait()
                except _q.Empty:
                    await asyncio.sleep(0.005)
                    continue
                if item is None:
                    break
                yield item + "\n"
        return StreamingResponse(lines(), media_type="application/x-ndjson")

    @app.get("/get_instances_status")
    async def get_instances_status():
        out = []
        for inst in scheduler.instances():
            st = inst.get_stats()
            out.append({
                "id": inst.instance_id,
                "is_local": inst.is_local,
                "active": inst.instance_id in scheduler._active,
                "running": st.num_running,
                "queued": st.num_queued,
                "gen_throughput": st.gen_throughput,
                "weight_version": getattr(inst, "weight_version", 0),
                "weight_sender_endpoint":
                    scheduler.weight_sender_for(inst.instance_id),
            })
        return {"instances": out,
                "latest_weight_version": scheduler.latest_weight_version,
                "max_local_gen_s": scheduler.max_local_gen_s}

    @app.post("/update_weight_version")
    async def update_weight_version(request: Request):
        body = await request.json()
        await _run(scheduler.update_weight_version(int(body["version"])))
        return {"status": "ok"}

    @app.post("/get_receive_instances")
    async def get_receive_instances():
        insts = scheduler.get_receive_instances()
        return {"instances": [i.instance_id for i in insts],
                "senders": {i.instance_id:
                            scheduler.weight_sender_for(i.instance_id)
                            for i in insts}}

    @app.put("/update_weight_senders")
    async def update_weight_senders(request: Request):
        """Trainer registers its weight-sender endpoints; the scheduler
        round-robins them over instances (handlers.rs PUT route +
        state.rs:149-162)."""
        body = await request.json()
        scheduler.update_weight_senders(list(body.get("senders", [])))
        return {"status": "ok", "num_senders": len(body.get("senders", []))}

    @app.post("/update_weights")
    async def update_weights(request: Request):
        body = await request.json()
        version = int(body["version"])
        results = {}
        for iid in body.get("instances", []):
            await _run(scheduler.finish_weight_update(iid, version,
                                                      success=True))
            results[iid] = True
        return {"status": "ok", "results": results}

    @app.post("/update_metrics")
    async def update_metrics(request: Request):
        from .types import MetricsUpdate
        body = await request.json()
        return scheduler.update_metrics(MetricsUpdate(
            step_time_s=float(body.get("step_time_s", 0.0)),
            trainer_bubble_time_s=float(body.get("trainer_bubble_time_s",
                                                 0.0)),
            step_throughput=float(body.get("step_throughput", 0.0))))

    @app.post("/abort_local_requests")
    async def abort_local_requests():
        for inst in scheduler.instances(active_only=True):
            if inst.is_local:
                inst.abort_all()
        return {"status": "ok"}

    @app.post("/shutdown_instances")
    async def shutdown_instances(request: Request):
        body = await request.json()
        for iid in body.get("instances", []):
            await _run(scheduler.evict_instance(iid, shutdown=True))
        return {"status": "ok"}

    return app


def serve_manager(scheduler, host: str = "0.0.0.0", port: int = 5000,
                  loop: Optional[asyncio.AbstractEventLoop] = None,
                  remote_weight_state_fn=None):
    """Run the facade in a daemon thread (the reference spawns the Rust
    manager on the head node, launcher.py:14-51); returns the server."""
    import threading

    import uvicorn
    app = create_manager_app(scheduler, loop, remote_weight_state_fn)
    server = uvicorn.Server(uvicorn.Config(app, host=host, port=port,
                                           log_level="error"))
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    return server
