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
                "weight_version": scheduler.latest_weight_version}

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
        return {"instances": [i.instance_id for i in insts]}

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
