"""Disaggregated (split) mode: trainer ranks + dedicated rollout ranks on
one RCCL world (BASELINE config #4: e.g. 4 trainer GPUs + 4 rollout GPUs).

Mirrors the reference's disaggregation with MI355X-
native transports:
  * generation: trainer rank 0 drives the in-process RolloutScheduler over
    HttpInstances of the rollout ranks' engine servers (NDJSON-equivalent
    group streaming, token-level continuation, weight-version gating);
    finished groups are sliced into equal per-trainer-rank ibatches and
    broadcast over the trainer group (the reference's RANK_ZERO
    get_stream_batches pump, stream_fsdp_workers.py:497-507).
  * weights: bucketed RCCL broadcast over the WORLD group
    (transfer/collective.py) — trainer ranks all-gather their FSDP shards,
    rank 0's buckets stream over xGMI to every rollout rank, which installs
    them under the engine step lock.  Control plane = one
    broadcast_object_list opcode per step ("publish" | "exit").
"""
from __future__ import annotations

import asyncio
import os
import threading
import time
from typing import Dict, Iterator, List, Optional

import torch
import torch.distributed as dist

from ..protocol import TensorBatch
from ..scheduler import RolloutScheduler, SchedulerConfig
from ..scheduler.manager import StreamingBatchIterator
from ..scheduler.types import GroupRequest, SamplingSpec
from .rollout_coordinator import postprocess_groups


def split_roles(world: int, num_rollout: int):
    assert 0 < num_rollout < world, \
        f"num_rollout_ranks {num_rollout} must be in (0, world {world})"
    t = world - num_rollout
    return list(range(t)), list(range(t, world))


def rollout_port(rank: int, base: int = 30000) -> int:
    return base + rank


def advertise_addr() -> str:
    """The address other nodes reach this rank's HTTP engine server at.
    Multi-node: the interface that routes to MASTER_ADDR (or an explicit
    POLYRL_ADVERTISE_ADDR); single node this resolves to 127.0.0.1."""
    a = os.environ.get("POLYRL_ADVERTISE_ADDR")
    if a:
        return a
    master = os.environ.get("MASTER_ADDR", "127.0.0.1")
    try:
        import socket
        sk = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        sk.connect((master, int(os.environ.get("MASTER_PORT", "29500"))))
        ip = sk.getsockname()[0]
        sk.close()
        return ip
    except OSError:
        return "127.0.0.1"


# ------------------------------------------------------------ rollout rank


def rollout_serve_loop(cfg, model_cfg, rank: int, device: str, dtype,
                       port_base: int = 30000):
    """Main loop of a dedicated rollout rank: engine + HTTP server thread;
    the main thread participates in the control/weight collectives."""
    import uvicorn

    from ..rollout.engine import Engine
    from ..rollout.runner import EngineRunner
    from ..server import create_app
    from ..transfer.collective import CollectiveWeightPlane

    ro = cfg.actor_rollout_ref.rollout
    kv_budget = 64 << 20 if device == "cpu" else None
    if device != "cpu":
        free, _ = torch.cuda.mem_get_info()
        kv_budget = int(free * ro.gpu_memory_utilization * 0.6)
    engine = Engine(model_cfg, device=device, dtype=dtype,
                    page_size=ro.page_size, kv_bytes_budget=kv_budget,
                    max_running_requests=ro.max_running_requests,
                    max_num_batched_tokens=ro.max_num_batched_tokens,
                    max_model_len=ro.prompt_length + ro.response_length,
                    decode_chunk_size=ro.decode_chunk_size,
                    enable_radix_cache=ro.enable_radix_cache,
                    seed=cfg.trainer.seed)
    runner = EngineRunner(engine)
    app = create_app(engine, runner)
    port = rollout_port(rank, port_base)
    # bind all interfaces so trainer ranks on OTHER nodes can reach us;
    # the advertised address travels through main_stream's address exchange
    server = uvicorn.Server(uvicorn.Config(app, host="0.0.0.0", port=port,
                                           log_level="error"))
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    print(f"[rollout rank {rank}] engine serving on "
          f"{advertise_addr()}:{port}", flush=True)

    plane = CollectiveWeightPlane(src=0, device=device)

    def apply_weight(name, tensor):
        engine.model.update_named(name, tensor)

    while True:
        box = [None]
        dist.broadcast_object_list(box, src=0)
        op = box[0]
        if op == "publish":
            with runner.lock:  # excludes in-flight generation during swap
                version = plane.receive(apply_weight)
                engine.flush_radix()
            runner.weight_version = version
        elif op == "exit":
            break
        else:
            raise RuntimeError(f"unknown ctrl opcode {op!r}")
    server.should_exit = True
    runner.stop()


# ------------------------------------------------------------ trainer side


class DisaggPublisher:
    """Weight publication for the split mode: ctrl opcode + bucketed world
    broadcast; scheduler's weight version is bumped first so stale remote
    instances are version-gated (handlers.rs:566-600 contract)."""

    def __init__(self, model, device: str, scheduler: Optional[RolloutScheduler],
                 loop: Optional[asyncio.AbstractEventLoop], rank: int):
        from ..transfer.collective import CollectiveWeightPlane
        self.model = model
        self.plane = CollectiveWeightPlane(src=0, device=device)
        self.scheduler = scheduler
        self.loop = loop
        self.rank = rank
        self.version = 0
        self.last_publish_s = 0.0

    @torch.no_grad()
    def publish(self) -> int:
        t0 = time.perf_counter()
        self.version += 1
        if self.rank == 0:
            dist.broadcast_object_list(["publish"], src=0)
            if self.scheduler is not None:
                fut = asyncio.run_coroutine_threadsafe(
                    self.scheduler.update_weight_version(self.version),
                    self.loop)
                fut.result(timeout=60)
        else:
            box = [None]
            dist.broadcast_object_list(box, src=0)
        sd = dict(self.model.state_dict())
        self.plane.publish(sd, version=self.version)
        if self.rank == 0 and self.scheduler is not None:
            # remote instances got the bytes via the broadcast; mark them
            # current so they rejoin the active pool
            async def _activate():
                for inst in self.scheduler.get_receive_instances():
                    await self.scheduler.finish_weight_update(
                        inst.instance_id, self.version, success=True)
            asyncio.run_coroutine_threadsafe(_activate(), self.loop) \
                .result(timeout=60)
        self.last_publish_s = time.perf_counter() - t0
        return self.version

    def shutdown(self):
        """Release the rollout ranks.  The ctrl broadcast is a WORLD
        collective — every trainer rank must take part."""
        if self.rank == 0:
            dist.broadcast_object_list(["exit"], src=0)
        else:
            box = [None]
            dist.broadcast_object_list(box, src=0)


class DisaggCoordinator:
    """Trainer-side coordinator for split mode (LocalRolloutCoordinator
    surface: submit / stream_batches).  Rank 0 talks to the rollout pool via
    the scheduler; every trainer rank receives equal ibatch slices."""

    def __init__(self, response_length: int, trainer_group,
                 rollout_urls: List, rank: int, n_trainer: int,
                 pad_token_id: int = 0, device="cpu",
                 max_local_gen_s: float = 0.0,
                 manager_port: int = 0,
                 remote_weight_state_fn=None,
                 sched_cfg=None, multi_turn=None):
        """rollout_urls entries are either plain url strings (remote) or
        (url, is_local) tuples.  ``manager_port`` > 0 serves the manager
        HTTP facade (scheduler/http_api.py) on rank 0 so elastic remote
        instances can join at runtime; new joins get
        ``remote_weight_state_fn`` as their TCP weight source."""
        self.response_length = response_length
        self.group = trainer_group
        self.rank = rank
        self.n_trainer = n_trainer
        self.pad = pad_token_id
        self.device = device
        self.max_local_gen_s = max_local_gen_s
        # multi-turn through the scheduler path: dict from
        # rollout_coordinator.load_interaction (reference MultiTurnConfig)
        self.multi_turn = multi_turn
        self._sampling_spec = None
        self._iter: Optional[StreamingBatchIterator] = None
        self._meta: Dict[int, dict] = {}
        self._gid = 0
        self.loop: Optional[asyncio.AbstractEventLoop] = None
        self.scheduler: Optional[RolloutScheduler] = None
        self.manager_server = None
        if rank == 0:
            self.loop = asyncio.new_event_loop()
            t = threading.Thread(target=self.loop.run_forever, daemon=True)
            t.start()
            sc = SchedulerConfig(stats_interval_s=0.2)
            if sched_cfg is not None:      # PPOConfig.scheduler knobs
                sc.max_assigned_batches_per_stats_check = \
                    sched_cfg.max_assigned_batches_per_stats_check
                sc.health_check_interval_s = sched_cfg.health_check_interval_s
                sc.max_retries = sched_cfg.max_retries
                sc.scheduling_policy = getattr(sched_cfg,
                                               "scheduling_policy",
                                               sc.scheduling_policy)
            self.scheduler = RolloutScheduler(sc)

            async def _register():
                from ..server import HttpInstance
                for spec in rollout_urls:
                    url, is_local = spec if isinstance(spec, tuple) \
                        else (spec, False)
                    inst = HttpInstance(
                        url, is_local=is_local,
                        weight_state_fn=None if is_local
                        else remote_weight_state_fn)
                    # engines start at version 0 == scheduler's
                    await self.scheduler.register_instance(inst)
            asyncio.run_coroutine_threadsafe(_register(), self.loop) \
                .result(timeout=600)
            if manager_port > 0:
                from ..scheduler.http_api import serve_manager
                self.manager_server = serve_manager(
                    self.scheduler, port=manager_port, loop=self.loop,
                    remote_weight_state_fn=remote_weight_state_fn)

    # --------------------------------------------------------------- submit
    def submit(self, prompts: TensorBatch, sampling, n: int):
        """Called with the GLOBAL batch on rank 0 (other ranks no-op)."""
        if self.rank != 0:
            return
        ids = prompts["input_ids"]
        mask = prompts["attention_mask"]
        uids = prompts["uid"]
        reqs = []
        for g in range(ids.shape[0]):
            gid = self._gid
            self._gid += 1
            raw = ids[g][mask[g].bool()].tolist()
            self._meta[gid] = {
                "uid": str(uids[g]),
                "extras": {k: v[g] for k, v in prompts.non_tensors.items()
                           if k != "uid"},
                "prompt_ids": ids[g].cpu(),
                "prompt_mask": mask[g].cpu(),
                "raw": raw,
            }
            reqs.append(GroupRequest(
                gid=gid, input_ids=raw, n=n,
                sampling=SamplingSpec(
                    temperature=sampling.temperature,
                    top_k=sampling.top_k, top_p=sampling.top_p,
                    max_new_tokens=sampling.max_new_tokens,
                    stop_token_ids=tuple(sampling.stop_token_ids))))
        self._sampling_spec = reqs[0].sampling if reqs else None
        self._iter = StreamingBatchIterator(
            self.scheduler, reqs,
            max_local_gen_s=self.max_local_gen_s, loop=self.loop)

    def update_metrics(self, step_time_s: float, trainer_bubble_s: float,
                       step_throughput: float) -> float:
        """Trainer feedback -> adaptive local-gen window (the reference's
        /update_metrics loop, stream_ray_trainer.py:691-704).  Rank 0 only;
        returns (and adopts) the new window."""
        if self.scheduler is None:
            return self.max_local_gen_s
        from ..scheduler.types import MetricsUpdate
        out = self.scheduler.update_metrics(MetricsUpdate(
            step_time_s=step_time_s, trainer_bubble_time_s=trainer_bubble_s,
            step_throughput=step_throughput))
        if self.max_local_gen_s > 0:          # adaptive only when time-boxed
            self.max_local_gen_s = out["new_max_gen_s"]
        return self.max_local_gen_s

    # --------------------------------------------------------------- stream
    def stream_batches(self, local_stream: int) -> Iterator[TensorBatch]:
        """Yield equal per-rank batches of local_stream samples until the
        submitted batch is consumed.  Rank 0 pulls from the scheduler stream
        and fans slices out over the trainer group."""
        while True:
            if self.rank == 0:
                shard_box = self._next_shards(local_stream)
            else:
                shard_box = [None]
            dist.broadcast_object_list(shard_box, src=0, group=self.group)
            shards = shard_box[0]
            if shards is None:
                return
            yield shards[self.rank if self.group is None
                         else dist.get_rank(self.group)]

    def _next_shards(self, local_stream: int):
        """Rank 0: gather n_trainer*local_stream finished samples, build one
        TensorBatch per trainer rank.  Returns [None] when drained."""
        need = self.n_trainer * local_stream
        got, groups = 0, []
        for item in self._iter:
            if isinstance(item, dict):     # submit notifier
                continue
            res = item
            meta = self._meta.pop(res.gid)
            if self.multi_turn is not None:
                self._run_multi_turns(meta["raw"], res)
            groups.append((meta, res))
            got += len(res.samples)
            if got >= need:
                break
        if not groups:
            return [None]
        # partition into EQUAL sample shards — required so every trainer
        # rank sees the same cumulative counts (the minibatch-boundary
        # is_opt_step flags must agree across ranks or FSDP deadlocks);
        # config validation guarantees full rounds divide evenly
        assert got % self.n_trainer == 0, \
            f"stream round of {got} samples !% {self.n_trainer} trainer ranks"
        per = got // self.n_trainer
        sizes = [len(res.samples) for _, res in groups]
        if len(set(sizes)) == 1 and len(groups) % self.n_trainer == 0:
            # token-balanced shards (the reference's _balance_batch DP
            # seqlen balance, stream_ray_trainer.py:406-410): equal group
            # COUNTS per rank, response tokens spread karmarkar-karp style
            toks = [sum(len(x.output_ids) for x in res.samples)
                    for _, res in groups]
            from ..core.seqlen import get_seqlen_balanced_partitions
            parts = get_seqlen_balanced_partitions(
                toks, self.n_trainer, equal_size=True)
            return [[self._make_batch([groups[i] for i in sorted(pt)])
                     for pt in parts]]
        out = []
        gi = 0
        for _ in range(self.n_trainer):
            take, taken = [], 0
            while gi < len(groups) and taken < per:
                take.append(groups[gi])
                taken += len(groups[gi][1].samples)
                gi += 1
            assert taken == per, "group sizes must tile the shard evenly"
            out.append(self._make_batch(take))
        return [out]

    def _run_multi_turns(self, raw, res):
        """Scheduler-path multi-turn (reference MultiTurnConfig capability):
        after each finished assistant turn, ask the interaction for the
        next user turn and continue the sample through the scheduler (with
        its full token-level fault tolerance); user tokens carry no loss."""
        from dataclasses import replace as _replace
        mt = self.multi_turn
        for s in res.samples:
            ids = list(s.output_ids)
            lps = list(s.output_logprobs)
            loss = [1] * len(ids)
            a_turns, u_turns = 1, 0
            while (s.finish_reason not in ("abort", "error")
                   and a_turns < mt["max_assistant_turns"]
                   and u_turns < mt["max_user_turns"]):
                budget = self.response_length - len(ids)
                if budget <= 1:
                    break
                user_ids, done = mt["interaction"](list(raw), list(ids))
                if done or not user_ids:
                    break
                user_ids = list(user_ids)[
                    :mt.get("max_tool_response_length", 256)]
                if len(user_ids) >= budget:
                    break
                ids += user_ids
                lps += [0.0] * len(user_ids)
                loss += [0] * len(user_ids)
                u_turns += 1
                mx = self.response_length - len(ids)
                cap = mt.get("per_turn_max_tokens", 0)
                if cap > 0:
                    mx = min(mx, cap)
                cont = GroupRequest(
                    gid=res.gid, input_ids=list(raw) + ids, n=1,
                    sampling=_replace(self._sampling_spec,
                                      max_new_tokens=mx))
                r2 = asyncio.run_coroutine_threadsafe(
                    self.scheduler.process_group(cont),
                    self.loop).result(timeout=600)
                s2 = r2.samples[0]
                ids += s2.output_ids
                lps += s2.output_logprobs
                loss += [1] * len(s2.output_ids)
                s.finish_reason = s2.finish_reason
                s.num_migrations += s2.num_migrations
                res.instance_ids.extend(r2.instance_ids)
                a_turns += 1
            Lr = self.response_length
            s.output_ids = ids[:Lr]
            s.output_logprobs = lps[:Lr]
            s.loss_mask = loss[:Lr]

    def _make_batch(self, pairs) -> TensorBatch:
        prompt_ids = torch.stack([m["prompt_ids"] for m, _ in pairs])
        prompt_mask = torch.stack([m["prompt_mask"] for m, _ in pairs])
        uids = [m["uid"] for m, _ in pairs]

        class _O:                      # adapt SampleResult -> RequestOutput
            def __init__(self, s):
                self.output_ids = s.output_ids
                self.output_logprobs = s.output_logprobs
                self.finish_reason = s.finish_reason
                self.loss_mask = getattr(s, "loss_mask", None)
        outputs = [[_O(s) for s in res.samples] for _, res in pairs]
        out = postprocess_groups(prompt_ids, prompt_mask, uids, outputs,
                                 self.response_length, self.pad,
                                 self.device,
                                 group_extras=[m.get("extras")
                                               for m, _ in pairs])
        # fault-tolerance observability: continuation hops in this slice
        out.meta_info["num_migrations"] = int(sum(
            s.num_migrations for _, res in pairs for s in res.samples))
        return out


class ElasticPublisher:
    """Publisher for the elastic co-located mode: every rank copies full
    params straight into ITS engine (the xGMI-free fast path), rank 0 bumps
    the scheduler version (local instances rejoin immediately — their bytes
    are already current), and stale REMOTE instances receive an async TCP
    push of a host-cached copy (the reference's async sender-agent
    behavior, sender_agent.py:324-647)."""

    def __init__(self, model, engine_model, coordinator, tie: bool,
                 trainer_group=None):
        from ..transfer.weight_transfer import WeightPublisher
        self.inner = WeightPublisher(model, [engine_model],
                                     tie_word_embeddings=tie)
        self.coordinator = coordinator
        self.group = trainer_group
        self.cpu_cache: Dict[str, torch.Tensor] = {}
        self._cache_lock = threading.Lock()
        self.last_publish_s = 0.0

    def snapshot_cache(self) -> Dict[str, torch.Tensor]:
        """Consistent snapshot for async TCP pushes (the next publish may
        refill the cache while a push is in flight)."""
        with self._cache_lock:
            return dict(self.cpu_cache)

    @property
    def version(self) -> int:
        return self.inner.version

    @torch.no_grad()
    def publish(self) -> int:
        t0 = time.time()
        sched = self.coordinator.scheduler
        # all trainer ranks must agree on whether to host-cache (the gather
        # loop is collective); only rank 0 knows the remote roster
        need = [0]
        if sched is not None:
            need = [int(any(not i.is_local for i in sched.instances()))]
        if dist.is_available() and dist.is_initialized():
            dist.broadcast_object_list(need, src=0, group=self.group)
        use_cache = sched is not None and need[0]
        if use_cache:
            staging: Dict[str, torch.Tensor] = {}
            v = self.inner.publish(cpu_cache=staging)
            with self._cache_lock:
                self.cpu_cache = staging
        else:
            v = self.inner.publish()
        if sched is not None:
            loop = self.coordinator.loop

            async def _bump_and_push():
                await sched.update_weight_version(v)
                for inst in sched.get_receive_instances():
                    # TCP push + activation, async w.r.t. training
                    await sched.finish_weight_update(inst.instance_id, v,
                                                     success=True)
            fut = asyncio.run_coroutine_threadsafe(_bump_and_push(), loop)
            if not need[0]:
                fut.result(timeout=60)   # no remotes: cheap, keep ordering
        self.last_publish_s = time.time() - t0
        return v

    def shutdown(self):
        pass
