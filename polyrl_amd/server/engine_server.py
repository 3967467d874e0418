"""FastAPI app exposing one Engine as a remote rollout instance.

Route surface mirrors what the reference scheduler + clients consume from a
rollout server (handlers.rs:153,264,399,656,919; sglang_http_async_engine.py:
155-299): token-in/token-out /generate with optional SSE streaming, stats,
abort, weight updates, memory occupation switches.

Weight delivery over HTTP: POST /update_weights_from_agent with a
safetensors payload (body bytes) or a node-local file path — the co-located
fast path never goes through HTTP (transfer/ does hipIPC + RCCL instead).
"""
import asyncio
import json
from typing import List, Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from ..rollout.engine import Engine, SamplingParams
from ..rollout.runner import EngineRunner


def create_app(engine: Engine, runner: Optional[EngineRunner] = None,
               allowed_sender_cidrs: Optional[List[str]] = None):

    app = FastAPI(title="polyrl-amd rollout instance")
    runner = runner or EngineRunner(engine)
    app.state.runner = runner
    app.state.engine = engine
    app.state.allowed_sender_cidrs = allowed_sender_cidrs

    def _sender_ok(request: Request) -> bool:
        from ..transfer.tcp_engine import addr_allowed
        host = request.client.host if request.client else None
        return addr_allowed(host, app.state.allowed_sender_cidrs)

    def _sp(d: dict) -> SamplingParams:
        return SamplingParams(
            temperature=float(d.get("temperature", 1.0)),
            top_k=int(d.get("top_k", -1)),
            top_p=float(d.get("top_p", 1.0)),
            max_new_tokens=int(d.get("max_new_tokens", 128)),
            stop_token_ids=tuple(d.get("stop_token_ids", ())))

    def _meta(o, include_logprobs: bool) -> dict:
        m = {"finish_reason": {"type": o.finish_reason},
             "completion_tokens": len(o.output_ids)}
        if include_logprobs:
            # reference format: [(logprob, token_id), ...]
            m["output_token_logprobs"] = [
                [lp, t] for lp, t in zip(o.output_logprobs, o.output_ids)]
        return m

    @app.post("/generate")
    async def generate(request: Request):
        body = await request.json()
        input_ids: List[int] = body["input_ids"]
        sp = _sp(body.get("sampling_params", {}))
        n = int(body.get("sampling_params", {}).get("n", 1))
        want_lp = bool(body.get("return_logprob", False))
        stream = bool(body.get("stream", False))

        futs = runner.submit_group(input_ids, sp, n)

        if not stream:
            outs = await asyncio.gather(*futs)
            return JSONResponse([
                {"index": i, "output_ids": o.output_ids,
                 "meta_info": _meta(o, want_lp)}
                for i, o in enumerate(outs)])

        async def sse():
            pending = {asyncio.ensure_future(_tag(i, f)): i
                       for i, f in enumerate(futs)}
            while pending:
                done, _ = await asyncio.wait(
                    pending.keys(), return_when=asyncio.FIRST_COMPLETED)
                for d in done:
                    pending.pop(d)
                    i, o = d.result()
                    chunk = {"index": i, "output_ids": o.output_ids,
                             "meta_info": _meta(o, want_lp)}
                    yield f"data: {json.dumps(chunk)}\n\n"
            yield "data: [DONE]\n\n"

        async def _tag(i, f):
            return i, await f

        return StreamingResponse(sse(), media_type="text/event-stream")

    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/health_generate")
    async def health_generate():
        # liveness probe that exercises the engine (one 1-token decode)
        sp = SamplingParams(max_new_tokens=1, temperature=0.0)
        out = await runner.submit([1], sp)
        return {"status": "ok", "tokens": len(out.output_ids)}

    @app.get("/get_server_info")
    async def get_server_info():
        return runner.stats()

    @app.post("/abort_request")
    async def abort_request(request: Request):
        body = await request.json()
        runner.abort(rid=body.get("rid"),
                     abort_all=bool(body.get("abort_all", False)))
        return {"status": "ok"}

    @app.post("/flush_cache")
    async def flush_cache():
        # paged KV frees per request; the retained state is the radix
        # prefix cache — drop it
        engine.flush_radix()
        return {"status": "ok"}

    @app.post("/update_weights_from_agent")
    async def update_weights_from_agent(request: Request):
        if not _sender_ok(request):
            return JSONResponse({"success": False,
                                 "message": "sender not in allowed CIDRs"},
                                403)
        """Install new weights.  Accepts either
        {"version": V, "path": "/node/local/file.safetensors"} or raw
        safetensors bytes with the version in X-Weight-Version header."""
        import torch
        ctype = request.headers.get("content-type", "")
        unlock = False
        if ctype.startswith("application/json"):
            body = await request.json()
            version = int(body["version"])
            # off-policy mode (reference roadmap 'unlock weight update of
            # rollout engines'): keep in-flight generation running across
            # the swap; the trainer's TIS reweighting corrects for it
            unlock = bool(body.get("unlock", False))
            if body.get("ack_only"):
                # weights were delivered out of band (collective broadcast
                # over xGMI, transfer/collective.py): version acknowledgment
                runner.weight_version = version
                return {"success": True,
                        "message": f"ack version {version} (out-of-band)"}
            path = body.get("path")
            if path:
                from safetensors.torch import load_file
                sd = load_file(path)
            else:
                return JSONResponse({"success": False,
                                     "message": "no path given"}, 400)
        else:
            raw = await request.body()
            version = int(request.headers.get("x-weight-version", "0"))
            from safetensors.torch import load
            sd = load(raw)
        # run the blocking swap off the event loop; takes the step lock so
        # it excludes in-flight generation
        await asyncio.get_running_loop().run_in_executor(
            None, lambda: runner.update_weights(
                sd, version, abort_in_flight=not unlock))
        return {"success": True, "message": f"weights at version {version}"}

    @app.post("/update_weights_from_tensor")
    async def update_weights_from_tensor(request: Request):
        """Reference-adapter surface (sglang_http_async_engine.py:155-299
        capability): JSON {"version": V, "tensors": {name: {"data": b64,
        "shape": [...], "dtype": "bfloat16"}}} — base64 raw tensor bytes.
        The agent/TCP routes are the efficient paths; this one exists for
        drop-in client compatibility."""
        if not _sender_ok(request):
            return JSONResponse({"success": False,
                                 "message": "sender not in allowed CIDRs"},
                                403)
        import base64

        import numpy as np
        import torch
        body = await request.json()
        version = int(body.get("version", runner.weight_version + 1))
        unlock = bool(body.get("unlock", False))
        sd = {}
        for name, t in body["tensors"].items():
            dt = getattr(torch, t.get("dtype", "float32"))
            raw = base64.b64decode(t["data"])
            # frombuffer has no bf16: view as uint8 then reinterpret
            ten = torch.frombuffer(bytearray(raw), dtype=torch.uint8) \
                .view(dt).reshape(t["shape"])
            sd[name] = ten
        await asyncio.get_running_loop().run_in_executor(
            None, lambda: runner.update_weights(
                sd, version, abort_in_flight=not unlock))
        return {"success": True, "message": f"weights at version {version}"}

    @app.post("/weights_handshake")
    async def weights_handshake(request: Request):
        """Arm a TCP bulk receive (the reference's receiver-agent bootstrap,
        receiver_agent.py:184-240): allocate/reuse a registered CPU buffer
        sized to the incoming state dict, listen on N stream ports, return
        them to the sender."""
        if not _sender_ok(request):
            return JSONResponse({"success": False,
                                 "message": "sender not in allowed CIDRs"},
                                403)
        import torch

        from ..transfer.tcp_engine import TcpWeightReceiver
        body = await request.json()
        metas = body["metas"]            # [(name, shape, dtype_str), ...]
        num_streams = int(body.get("num_streams", 4))
        total = 0
        for meta in metas:
            shape, dts = meta[1], meta[2]
            n = 1
            for s in shape:
                n *= s
            total += n * torch.tensor([], dtype=getattr(torch, dts)) \
                .element_size()
        st = app.state
        if getattr(st, "wt_buffer", None) is None or \
                st.wt_buffer.numel() < total:
            st.wt_buffer = torch.empty(total, dtype=torch.uint8)
        if getattr(st, "wt_rx", None) is not None:
            st.wt_rx.close()
        st.wt_rx = TcpWeightReceiver(st.wt_buffer[:total],
                                     host=body.get("bind_host", "0.0.0.0"),
                                     num_streams=num_streams)
        st.wt_rx.expect(total)
        st.wt_metas = metas
        return {"ports": st.wt_rx.ports, "total_bytes": total}

    @app.post("/update_weights_from_tcp")
    async def update_weights_from_tcp(request: Request):
        """Install weights from the armed TCP receive: wait for all spans,
        reconstruct named views (patches.py:205-215 capability), swap under
        the step lock."""
        if not _sender_ok(request):
            return JSONResponse({"success": False,
                                 "message": "sender not in allowed CIDRs"},
                                403)
        import torch
        body = await request.json()
        version = int(body["version"])
        st = app.state
        assert getattr(st, "wt_rx", None) is not None, "handshake first"
        ok = await asyncio.get_running_loop().run_in_executor(
            None, lambda: st.wt_rx.wait(timeout=float(
                body.get("timeout_s", 600.0))))
        if not ok:
            return JSONResponse({"success": False,
                                 "message": "tcp receive timed out"}, 500)
        sd = {}
        off = 0
        for meta in st.wt_metas:
            name, shape, dts = meta[0], meta[1], meta[2]
            scale = float(meta[3]) if len(meta) > 3 else None
            dt = getattr(torch, dts)
            n = 1
            for s in shape:
                n *= s
            nbytes = n * torch.tensor([], dtype=dt).element_size()
            view = st.wt_buffer[off:off + nbytes].view(dt).view(shape)
            if scale is not None:      # fp8-compressed: dequantize
                view = view.to(torch.float32) * scale
            sd[name] = view
            off += nbytes
        await asyncio.get_running_loop().run_in_executor(
            None, lambda: runner.update_weights(
                sd, version,
                abort_in_flight=not bool(body.get("unlock", False))))
        st.wt_rx.close()
        st.wt_rx = None
        return {"success": True, "message": f"weights at version {version}"}

    @app.post("/release_memory_occupation")
    async def release_memory_occupation():
        with runner.lock:
            if not engine.has_work():
                engine.release_memory()
                return {"status": "ok", "released": True}
        return {"status": "busy", "released": False}

    @app.post("/resume_memory_occupation")
    async def resume_memory_occupation():
        with runner.lock:
            engine.resume_memory()
        return {"status": "ok"}

    @app.post("/shutdown")
    async def shutdown():
        runner.stop()
        return {"status": "ok"}

    return app


def main():
    """``python -m polyrl_amd.server.engine_server --model qwen2.5-1.5b
    --port 30001 [--manager http://head:5000]`` — launch a remote elastic
    instance and (optionally) self-register with the scheduler's HTTP facade
    (reference: examples/scripts/launch_sglang.sh + patches.py:513-543)."""
    import argparse

    import torch
    import uvicorn

    from ..models.registry import get_model_config

    p = argparse.ArgumentParser()
    p.add_argument("--model", required=True)
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=30001)
    p.add_argument("--dtype", default="bfloat16")
    p.add_argument("--kv-gb", type=float, default=8.0)
    p.add_argument("--manager", default=None,
                   help="scheduler HTTP endpoint to register with")
    p.add_argument("--advertise-addr", default=None,
                   help="address the manager should reach us at (default: "
                        "the interface that routes to the manager)")
    p.add_argument("--load", default=None,
                   help="safetensors checkpoint to load (else random init)")
    p.add_argument("--radix-cache", action="store_true", default=True,
                   help="cross-request KV prefix reuse (default on for "
                        "serving; RL trainers flush it every publish)")
    p.add_argument("--no-radix-cache", dest="radix_cache",
                   action="store_false")
    p.add_argument("--allowed-sender-cidrs", default=None,
                   help="comma-separated CIDRs allowed to push weights "
                        "(reference: allowed_sender_ips)")
    args = p.parse_args()

    cfg = get_model_config(args.model)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = getattr(torch, args.dtype) if device == "cuda" else torch.float32
    engine = Engine(cfg, device=device, dtype=dtype,
                    kv_bytes_budget=int(args.kv_gb * (1 << 30)),
                    enable_radix_cache=args.radix_cache)
    if args.load:
        import os as _os
        if _os.path.isdir(args.load):
            # HF model dir (sharded safetensors): the engine's fused
            # buffers ingest per-name via load_state_dict
            import glob as _glob

            from safetensors.torch import load_file
            sd = {}
            for f in sorted(_glob.glob(_os.path.join(args.load,
                                                     "*.safetensors"))):
                sd.update(load_file(f))
            engine.model.load_state_dict(sd, strict=False)
        else:
            from safetensors.torch import load_file
            engine.model.load_state_dict(load_file(args.load), strict=False)
    else:
        for _, t in engine.model._name_map.items():
            t.normal_(0, 0.02)
    cidrs = args.allowed_sender_cidrs.split(",") \
        if args.allowed_sender_cidrs else None
    app = create_app(engine, allowed_sender_cidrs=cidrs)

    if args.manager:
        # register AFTER our own server answers /health (the manager
        # health-gates joins, instance_manager.rs:5-37 capability) — from
        # a background thread so uvicorn can start below
        import threading

        def _register():
            import time as _t

            import requests
            adv = args.advertise_addr
            if adv is None and args.host not in ("0.0.0.0", "::", ""):
                adv = args.host
            if adv is None:
                # the interface that routes to the manager (multi-node)
                import socket
                from urllib.parse import urlparse
                u = urlparse(args.manager)
                try:
                    sk = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
                    sk.connect((u.hostname, u.port or 80))
                    adv = sk.getsockname()[0]
                    sk.close()
                except OSError:
                    adv = "127.0.0.1"
            if ":" not in adv:
                adv = f"{adv}:{args.port}"
            deadline = _t.monotonic() + 120
            while _t.monotonic() < deadline:
                try:
                    requests.get(f"http://{adv}/health", timeout=2)
                    break
                except Exception:
                    _t.sleep(0.2)
            for _ in range(30):
                try:
                    requests.post(
                        f"{args.manager}/register_rollout_instance",
                        json={"addr": f"http://{adv}"}, timeout=10)
                    return
                except Exception:
                    _t.sleep(1.0)
        threading.Thread(target=_register, daemon=True).start()

    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
