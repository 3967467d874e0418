"""HttpInstance — scheduler-side client for a remote rollout instance.

The remote analog of scheduler.InProcessInstance: speaks the engine HTTP
contract (engine_server.py routes; reference handlers.rs:152-328 relays the
same surface).  Uses httpx; pass ``transport`` to run against an in-process
ASGI app in tests (no sockets).
"""
from __future__ import annotations

import asyncio
import json
from typing import List, Optional

from ..scheduler.instances import RolloutInstance
from ..scheduler.types import GroupRequest, InstanceStats, SampleResult


class HttpInstance(RolloutInstance):
    def __init__(self, base_url: str, instance_id: Optional[str] = None,
                 transport=None, timeout: float = 600.0,
                 weight_file_fn=None, weight_state_fn=None,
                 is_local: bool = False):
        import httpx
        self.base_url = base_url.rstrip("/")
        self.instance_id = instance_id or self.base_url
        self.is_local = is_local
        self._client = httpx.AsyncClient(transport=transport,
                                         base_url=self.base_url,
                                         timeout=timeout)
        self._stats = InstanceStats()
        self._bg_tasks = set()          # strong refs for fire-and-forget
        # weight delivery (checked in order):
        #  * weight_state_fn: version -> state dict, pushed over the
        #    N-stream TCP plane (sender-agent path, tcp_engine.py)
        #  * weight_file_fn: version -> node-local safetensors path
        #  * neither: version ack only (bytes arrive out of band)
        self.weight_file_fn = weight_file_fn
        self.weight_state_fn = weight_state_fn
        self.weight_version = 0

    async def generate_group(self, req: GroupRequest) -> List[SampleResult]:
        payload = {
            "input_ids": list(req.input_ids),
            "sampling_params": {
                "n": req.n,
                "max_new_tokens": req.sampling.max_new_tokens,
                "temperature": req.sampling.temperature,
                "top_p": req.sampling.top_p,
                "top_k": req.sampling.top_k,
                "stop_token_ids": list(req.sampling.stop_token_ids),
            },
            "return_logprob": req.return_logprob,
            "stream": True,
        }
        outs = [SampleResult() for _ in range(req.n)]
        try:
            async with self._client.stream("POST", "/generate",
                                           json=payload) as resp:
                resp.raise_for_status()
                async for line in resp.aiter_lines():
                    if not line.startswith("data: "):
                        continue
                    data = line[len("data: "):]
                    if data == "[DONE]":
                        break
                    chunk = json.loads(data)
                    i = chunk["index"]
                    meta = chunk.get("meta_info", {})
                    o = outs[i]
                    o.output_ids = list(chunk.get("output_ids", []))
                    lps = meta.get("output_token_logprobs")
                    if lps is not None:
                        o.output_logprobs = [float(x[0]) for x in lps]
                    o.finish_reason = meta.get("finish_reason", {}).get(
                        "type", "length")
                    o.completion_tokens = meta.get("completion_tokens",
                                                   len(o.output_ids))
        except Exception as e:
            # stream broke: surface partials for token-level continuation
            # (handlers.rs:152-328 returns partial responses on failure)
            err = RuntimeError(f"instance {self.instance_id} stream failed: {e}")
            err.partials = outs
            raise err from e
        return outs

    def get_stats(self) -> InstanceStats:
        return self._stats

    async def refresh_stats(self):
        """Async stats poll (the scheduler's 1 Hz loop calls get_stats
        synchronously; the server facade refreshes this snapshot)."""
        try:
            r = await self._client.get("/get_server_info", timeout=2.0)
            d = r.json()
            self._stats = InstanceStats(
                num_running=int(d.get("#running_req", 0)),
                num_queued=int(d.get("#queue_req", 0)),
                gen_throughput=float(d.get("last_gen_throughput", 0.0)))
            self.weight_version = int(d.get("weight_version",
                                            self.weight_version))
        except Exception:
            pass

    async def health(self) -> bool:
        try:
            r = await self._client.get("/health_generate", timeout=30.0)
            return r.status_code == 200
        except Exception:
            return False

    async def update_weights(self, version: int, bootstrap: bool = False
                             ) -> bool:
        try:
            if self.weight_state_fn is not None:
                ok = await asyncio.get_running_loop().run_in_executor(
                    None, self._push_tcp, version)
                if ok:
                    self.weight_version = version
                return ok
            if self.weight_file_fn is not None:
                path = self.weight_file_fn(version)
                r = await self._client.post(
                    "/update_weights_from_agent",
                    json={"version": version, "path": path})
            else:
                # no file transport: weights arrive out of band (collective
                # broadcast); acknowledge the version so the scheduler
                # reactivates this instance
                r = await self._client.post(
                    "/update_weights_from_agent",
                    json={"version": version, "ack_only": True})
            ok = r.status_code == 200 and r.json().get("success", False)
        except Exception:
            ok = False
        if ok:
            self.weight_version = version
        return ok

    def _push_tcp(self, version: int) -> bool:
        """Blocking sender-agent push over the TCP plane (run off-loop)."""
        import urllib.parse

        import httpx

        from ..transfer.tcp_engine import push_state_dict_tcp
        sd = self.weight_state_fn(version)
        host = urllib.parse.urlparse(self.base_url).hostname or "127.0.0.1"
        with httpx.Client(base_url=self.base_url) as c:
            return push_state_dict_tcp(sd, c, host, version=version)

    def abort_all(self):
        # fire-and-forget from sync context.  The task MUST be strongly
        # referenced: asyncio holds only weak refs to tasks, and an
        # unreferenced abort task can be garbage-collected before it ever
        # POSTs (observed intermittently: time-box aborts silently lost,
        # the local engine ran to completion with zero migrations).
        async def _abort():
            try:
                await self._client.post("/abort_request",
                                        json={"abort_all": True}, timeout=5.0)
            except Exception:
                pass
        try:
            loop = asyncio.get_running_loop()
            t = loop.create_task(_abort())
            self._bg_tasks.add(t)
            t.add_done_callback(self._bg_tasks.discard)
        except RuntimeError:
            asyncio.run(_abort())

    async def shutdown(self):
        try:
            await self._client.post("/shutdown", timeout=5.0)
        except Exception:
            pass
        await self._client.aclose()
