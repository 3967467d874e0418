"""StreamPPOTrainer — the streamed PPO/GRPO fit loop, SPMD one-process-per-GPU.

Reference capability: StreamRayPPOTrainer.fit (stream_ray_trainer.py:282-689)
re-designed without Ray: every rank runs the same loop; the rollout engine is
in-process per rank; data-dependent control flow is deterministic by
construction (equal per-rank sample counts, exact-size stream batches), so no
cross-rank RPC is needed — only the FSDP/RCCL collectives inside the workers.

Preserved semantics:
  * streamed training: the actor updates on ibatch i while later samples are
    still decoding (engine work interleaves between update calls),
  * minibatch-boundary bookkeeping: optimizer steps fire when the cumulative
    streamed sample count crosses each ppo_mini_batch boundary
    (stream_ray_trainer.py:500-568), LR steps on the last slice,
  * driver-side reward -> old/ref logprob -> values -> KL penalty ->
    advantage (GAE or GRPO group-norm) per ibatch,
  * weight publication to every rollout consumer each iteration (§3.3).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..config import PPOConfig
from ..core import algos
from ..core.metrics import (Tracking, compute_data_metrics,
                            compute_throughput_metrics, compute_timing_metrics,
                            marked_timer, reduce_metrics)
from ..data import SyntheticPromptDataset, epoch_batches
from ..models import create_model, get_model_config
from ..protocol import TensorBatch
from ..reward import load_reward_manager
from ..rollout.engine import Engine, SamplingParams
from ..utils.profiling import step_profiler
from ..transfer.weight_transfer import WeightPublisher
from .checkpoint import CheckpointManager
from .rollout_coordinator import LocalRolloutCoordinator
from .workers import ActorWorker, CriticWorker


def _dist_info():
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(), dist.get_rank()
    return 1, 0


class StreamPPOTrainer:
    def __init__(self, config: PPOConfig, device: Optional[str] = None,
                 reward_fn=None, dataset=None, process_group=None,
                 rollout_addrs: Optional[Dict[int, str]] = None):
        self.config = config
        self.pg = process_group          # trainer subgroup (disagg) or None
        if process_group is not None:
            self.world = dist.get_world_size(process_group)
            self.rank = dist.get_rank(process_group)
        else:
            self.world, self.rank = _dist_info()
        if device is None:
            device = config.trainer.device
        if device == "cuda":
            device = f"cuda:{torch.cuda.current_device()}"
        self.device = device

        self._validate_config()

        arr = config.actor_rollout_ref
        model_cfg = get_model_config(arr.model.path,
                                     override=arr.model.override_config)
        self.model_cfg = model_cfg
        dtype = arr.model.dtype

        def _set_remove_padding(m, mc, enabled, pad_to=0):
            # packed varlen path: GPU kernels cover head_dim 64 and 128
            # (llama/qwen families); the CPU tier runs the torch reference
            # at any dim
            trunk = getattr(m, "model", None)
            if trunk is None or not hasattr(trunk, "use_remove_padding"):
                return
            if enabled and (device == "cpu" or mc.head_dim in (64, 128)):
                trunk.use_remove_padding = True
                # fixed-M padding: every dynamic-budget micro lands on the
                # same (optimal) GEMM shapes (models/llama.py _pack_pad_to)
                if pad_to and device != "cpu":
                    trunk.pack_pad_to = int(pad_to)

        torch.manual_seed(config.trainer.seed)
        actor_model = create_model(model_cfg, kind="actor", dtype=dtype,
                                   device=device)
        if arr.model.load_weights:
            from ..models.hf_loader import load_hf_checkpoint
            miss, unexp = load_hf_checkpoint(actor_model,
                                             arr.model.load_weights)
            if self.rank == 0:
                print(f"[trainer] loaded HF weights from "
                      f"{arr.model.load_weights} "
                      f"(missing {len(miss)}, unexpected {len(unexp)})",
                      flush=True)
        if arr.model.enable_gradient_checkpointing:
            actor_model.gradient_checkpointing_enable()
        if arr.model.lora_rank > 0:
            from ..models.lora import apply_lora
            apply_lora(actor_model, arr.model.lora_rank,
                       alpha=arr.model.lora_alpha)
        actor_pad_to = (arr.actor.ppo_max_token_len_per_gpu *
                        arr.actor.ulysses_sequence_parallel_size
                        if arr.actor.use_dynamic_bsz else 0)
        _set_remove_padding(actor_model, model_cfg,
                            arr.model.use_remove_padding, actor_pad_to)
        self.actor = ActorWorker(actor_model, arr.actor, device=device,
                                 pg=self.pg)

        self.use_ref = arr.actor.use_kl_loss or config.algorithm.use_kl_in_reward
        self.ref = None
        if self.use_ref:
            torch.manual_seed(config.trainer.seed)
            ref_model = create_model(model_cfg, kind="actor", dtype=dtype,
                                     device=device)
            ref_model.load_state_dict(
                {k: v for k, v in actor_model.state_dict().items()})
            for p in ref_model.parameters():
                p.requires_grad_(False)
            _set_remove_padding(ref_model, model_cfg,
                                arr.model.use_remove_padding, actor_pad_to)
            self.ref = ActorWorker(ref_model, arr.actor, device=device,
                                   is_ref=True, pg=self.pg)

        self.use_critic = config.algorithm.adv_estimator == "gae"
        self.critic = None
        if self.use_critic:
            critic_cfg = config.critic
            critic_model_cfg = get_model_config(
                critic_cfg.model.path if critic_cfg.model.path != "llama3-1b"
                or arr.model.path == "llama3-1b"
                else arr.model.path)
            critic_model = create_model(critic_model_cfg, kind="critic",
                                        dtype=dtype, device=device)
            if critic_cfg.model.enable_gradient_checkpointing:
                critic_model.gradient_checkpointing_enable()
            critic_pad_to = (critic_cfg.ppo_max_token_len_per_gpu *
                             getattr(critic_cfg,
                                     "ulysses_sequence_parallel_size", 1)
                             if critic_cfg.use_dynamic_bsz else 0)
            _set_remove_padding(critic_model, critic_model_cfg,
                                critic_cfg.model.use_remove_padding,
                                critic_pad_to)
            self.critic = CriticWorker(critic_model, critic_cfg,
                                       device=device, pg=self.pg)

        # ---------------- rollout plane -------------------------------------
        ro = arr.rollout
        self.disagg = ro.num_rollout_ranks > 0
        self.elastic = ro.name == "elastic" and not self.disagg
        # one tensor-parallel engine across the trainer ranks (config #5:
        # FSDP trainer + TP rollout, on-device FSDP->TP reshard)
        self.tp_rollout = (not self.disagg and not self.elastic
                           and ro.tensor_model_parallel_size > 1)
        # scheduler-coordinated generation (rank-0 pump + shard broadcast)
        self.sched_coordinated = self.disagg or self.elastic
        if self.disagg:
            # split mode (BASELINE config #4): no local engine; the rollout
            # pool lives on the trailing ranks, reached via the scheduler
            from .disagg import (DisaggCoordinator, DisaggPublisher,
                                 rollout_port, split_roles)
            gworld = dist.get_world_size()
            _, rollout_ranks = split_roles(gworld, ro.num_rollout_ranks)
            addrs = rollout_addrs or {}
            urls = [f"http://{addrs.get(r) or '127.0.0.1'}:"
                    f"{rollout_port(r, ro.rollout_port_base)}"
                    for r in rollout_ranks]
            self.engine = None
            from .rollout_coordinator import load_interaction
            self.coordinator = DisaggCoordinator(
                ro.response_length, self.pg, urls, rank=self.rank,
                n_trainer=self.world, pad_token_id=0, device="cpu",
                max_local_gen_s=ro.max_local_gen_s,
                sched_cfg=self.config.scheduler,
                multi_turn=load_interaction(ro.multi_turn))
            self.publisher = DisaggPublisher(
                self.actor.model, device,
                self.coordinator.scheduler, self.coordinator.loop,
                rank=dist.get_rank())
        else:
            eng_dtype = getattr(torch, ro.dtype) if device != "cpu" \
                else torch.float32
            kv_budget = self._kv_budget(ro)
            tp_ctx = None
            eng_seed = config.trainer.seed * 1000 + self.rank
            if ro.tensor_model_parallel_size > 1 and not self.elastic:
                from ..parallel.tp import TPContext
                tp = ro.tensor_model_parallel_size
                assert dist.is_initialized() and \
                    dist.get_world_size() == tp, \
                    f"TP rollout currently requires tp == world ({tp})"
                tp_ctx = TPContext(dist.group.WORLD)
                eng_seed = config.trainer.seed * 1000   # rank-uniform: the
                # TP engine's sampling must agree bitwise across ranks
            self.engine = Engine(model_cfg, device=device, dtype=eng_dtype,
                                 page_size=ro.page_size,
                                 kv_bytes_budget=kv_budget,
                                 max_running_requests=ro.max_running_requests,
                                 max_num_batched_tokens=ro.max_num_batched_tokens,
                                 max_model_len=ro.prompt_length + ro.response_length,
                                 decode_chunk_size=ro.decode_chunk_size,
                                 enable_radix_cache=ro.enable_radix_cache,
                                 tp_ctx=tp_ctx,
                                 seed=eng_seed)
            if self.elastic:
                self._setup_elastic(ro, model_cfg, device)
            else:
                shard = None
                if self.tp_rollout:
                    shard = (self.rank, self.world)
                from .rollout_coordinator import load_interaction
                self.coordinator = LocalRolloutCoordinator(
                    self.engine, ro.response_length, pad_token_id=0,
                    device="cpu", shard=shard,
                    multi_turn=load_interaction(ro.multi_turn))
                self.publisher = WeightPublisher(
                    self.actor.model, [self.engine.model],
                    tie_word_embeddings=model_cfg.tie_word_embeddings)

        # ---------------- data + reward -------------------------------------
        dcfg = config.data
        if dataset is not None:
            self.dataset = dataset
        elif dcfg.train_files:
            # tokenized parquet prompts (data.py schema; see
            # examples/data_preprocess/gsm8k.py for the preprocessor)
            from ..data import ParquetRLHFDataset
            from ..utils.tokenizer import get_tokenizer
            tok = get_tokenizer(config.actor_rollout_ref.model.path)
            self.dataset = ParquetRLHFDataset(
                list(dcfg.train_files), dcfg.max_prompt_length,
                prompt_key=dcfg.prompt_key, tokenizer=tok,
                input_ids_key="input_ids",
                filter_overlong=dcfg.filter_overlong_prompts)
        else:
            self.dataset = SyntheticPromptDataset(
                num_prompts=dcfg.synthetic_num_prompts,
                vocab_size=model_cfg.vocab_size,
                max_prompt_length=dcfg.max_prompt_length,
                seed=dcfg.seed)
        self.val_dataset = None
        if dcfg.val_files:
            from ..data import ParquetRLHFDataset
            from ..utils.tokenizer import get_tokenizer
            self.val_dataset = ParquetRLHFDataset(
                list(dcfg.val_files), dcfg.max_prompt_length,
                prompt_key=dcfg.prompt_key,
                tokenizer=get_tokenizer(config.actor_rollout_ref.model.path),
                input_ids_key="input_ids",
                filter_overlong=dcfg.filter_overlong_prompts)
        self.reward_fn = reward_fn or load_reward_manager("constant")

        self.ckpt_actor = CheckpointManager(config.trainer.default_local_dir,
                                            "actor", pg=self.pg)
        self.ckpt_critic = CheckpointManager(
            config.trainer.default_local_dir, "critic",
            pg=self.pg) if self.use_critic else None
        self.tracking = Tracking(config.trainer.project_name,
                                 config.trainer.experiment_name,
                                 config.trainer.logger if self.rank == 0 else [],
                                 config.trainer.default_local_dir)
        self.global_step = 0
        self._maybe_resume()

    # ------------------------------------------------------------------ setup
    def _setup_elastic(self, ro, model_cfg, device):
        """Elastic co-located mode (the reference's primary shape, §3.4):
        every rank serves its engine over HTTP; rank 0's scheduler drives
        the whole pool as LOCAL instances (time-boxed when remotes exist);
        elastic remote instances join at runtime through the manager facade
        and receive weights over the TCP plane."""
        import threading

        import uvicorn

        from ..rollout.runner import EngineRunner
        from ..server import create_app
        from .disagg import DisaggCoordinator, ElasticPublisher, rollout_port

        grank = dist.get_rank() if dist.is_initialized() else 0
        gworld = dist.get_world_size() if dist.is_initialized() else 1
        self._engine_runner = EngineRunner(self.engine)
        app = create_app(self.engine, self._engine_runner)
        port = rollout_port(grank, ro.rollout_port_base)
        self._engine_server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=port, log_level="error"))
        threading.Thread(target=self._engine_server.run, daemon=True).start()
        if dist.is_initialized():
            dist.barrier(group=self.pg)
        specs = [(f"http://127.0.0.1:{rollout_port(r, ro.rollout_port_base)}",
                  True) for r in range(gworld)]
        from .rollout_coordinator import load_interaction
        self.coordinator = DisaggCoordinator(
            ro.response_length, self.pg, specs, rank=self.rank,
            n_trainer=self.world, pad_token_id=0, device="cpu",
            max_local_gen_s=ro.max_local_gen_s,
            manager_port=ro.rollout_manager_port if self.rank == 0 else 0,
            remote_weight_state_fn=lambda v: self.publisher.snapshot_cache(),
            sched_cfg=self.config.scheduler,
            multi_turn=load_interaction(ro.multi_turn))
        self.publisher = ElasticPublisher(
            self.actor.model, self.engine.model, self.coordinator,
            tie=model_cfg.tie_word_embeddings, trainer_group=self.pg)
        if self.rank == 0:
            # register this trainer as the weight-sender endpoint so the
            # scheduler can assign it to joining instances (the reference's
            # launcher.register_weight_senders PUT at startup)
            from .disagg import advertise_addr
            import asyncio as _aio
            _sched = self.coordinator.scheduler
            _ep = f"{advertise_addr()}:{ro.rollout_manager_port or 0}"

            async def _reg_sender():
                _sched.update_weight_senders([_ep])
            _aio.run_coroutine_threadsafe(
                _reg_sender(), self.coordinator.loop).result(timeout=30)

    def _kv_budget(self, ro) -> int:
        if self.device.startswith("cuda"):
            from ..rollout.kv_cache import PagedKVCache
            mc = get_model_config(self.config.actor_rollout_ref.model.path)
            bt = PagedKVCache.bytes_per_token(
                mc.num_hidden_layers, mc.num_key_value_heads, mc.head_dim)
            # what max_running concurrent sequences at full length need
            need = bt * ro.max_running_requests * \
                (ro.prompt_length + ro.response_length)
            free, total = torch.cuda.mem_get_info()
            return int(min(free * ro.gpu_memory_utilization * 0.5,
                           need * 1.125))
        return 64 << 20

    def _validate_config(self):
        c = self.config
        n = c.actor_rollout_ref.rollout.sampling.n
        tb = c.data.train_batch_size
        mini = c.actor_rollout_ref.actor.ppo_mini_batch_size
        stream = c.actor_rollout_ref.rollout.min_stream_batch_size
        world = self.world
        assert tb % world == 0, f"train_batch_size {tb} % world {world}"
        total = tb * n
        assert total % mini == 0, \
            f"total samples {total} must divide by ppo_mini_batch {mini}"
        assert mini % world == 0
        assert stream % n == 0, f"min_stream_batch {stream} % n {n}"
        assert stream % world == 0
        local_stream = stream // world
        local_total = total // world
        assert local_total % local_stream == 0, \
            f"local total {local_total} % local stream {local_stream}"
        mini_local = mini // world
        assert mini_local % local_stream == 0 or local_stream % mini_local == 0 \
            or mini_local % n == 0, "minibatch/stream sizes must compose"
        # parallelism fences (reference: config/rollout.py:193-202 —
        # EP = TP x DP constraint, PP declared but NotImplemented for >1)
        ro = c.actor_rollout_ref.rollout
        if ro.pipeline_model_parallel_size > 1:
            raise NotImplementedError(
                "pipeline_model_parallel_size > 1 is config-declared but not "
                "implemented (matches the reference fence)")
        if ro.expert_parallel_size > 1:
            assert ro.expert_parallel_size == \
                ro.tensor_model_parallel_size * ro.data_parallel_size, \
                "expert_parallel_size must equal TP x DP"
        sp = c.actor_rollout_ref.actor.ulysses_sequence_parallel_size
        if sp > 1:
            assert world % sp == 0, f"world {world} % sp {sp} != 0"
        # reference fences kept 1:1 (stream_dp_actor.py:145-146, multi-turn)
        if c.actor_rollout_ref.actor.ppo_epochs != 1:
            raise NotImplementedError(
                "ppo_epochs != 1 is fenced (matches the reference: "
                "stream_dp_actor.py:145-146 — streamed minibatches are "
                "consumed once)")
        if ro.multi_turn.enable and ro.multi_turn.interaction_path is None:
            raise ValueError("multi_turn.enable needs an interaction_path "
                             "(python file with the turn-generator fn)")

    # ------------------------------------------------------------------- fit
    def fit(self, max_steps: Optional[int] = None):
        c = self.config
        if c.trainer.val_before_train and self.global_step == 0 \
                and not getattr(self, "_pre_validated", False):
            self._pre_validated = True
            val = self.validate()
            if self.rank == 0 and val:
                self.tracking.log(val, 0)
        total_steps = max_steps or c.trainer.total_training_steps or 1
        ro = c.actor_rollout_ref.rollout
        n = ro.sampling.n
        local_bs = c.data.train_batch_size // self.world
        local_stream = ro.min_stream_batch_size // self.world
        mini_local = c.actor_rollout_ref.actor.ppo_mini_batch_size // self.world
        local_total = local_bs * n

        step_in_run = 0
        skip = self.global_step   # resume: replay the deterministic batch
        for epoch in range(c.trainer.total_epochs):   # order up to the ckpt
            for global_batch in epoch_batches(self.dataset,
                                              c.data.train_batch_size,
                                              shuffle=c.data.shuffle,
                                              seed=c.data.seed + epoch):
                if skip > 0:
                    skip -= 1
                    continue
                if step_in_run >= total_steps:
                    return
                self.global_step += 1
                step_in_run += 1
                metrics: Dict[str, float] = {}
                timing: Dict[str, float] = {}
                with step_profiler(self.global_step in
                                   c.trainer.profile_steps,
                                   c.trainer.profile_dir, self.global_step):
                    with marked_timer("step", timing):
                        batch_metrics = self._run_step(
                            global_batch, n, local_bs, local_stream,
                            mini_local, local_total, timing)
                metrics.update(batch_metrics)
                metrics.update(compute_timing_metrics(
                    self._last_full_batch, timing))
                metrics.update(compute_throughput_metrics(
                    self._last_full_batch, timing,
                    self.world, model_cfg=self.model_cfg,
                    use_critic=self.use_critic))
                metrics["training/global_step"] = self.global_step
                if self.rank == 0:
                    self.tracking.log(metrics, self.global_step)
                if (c.trainer.save_freq > 0 and
                        self.global_step % c.trainer.save_freq == 0) or \
                        self._should_save_esi(timing.get("step", 0.0)):
                    self.save_checkpoint()
                if c.trainer.test_freq > 0 and \
                        self.global_step % c.trainer.test_freq == 0:
                    val = self.validate()
                    if self.rank == 0 and val:
                        self.tracking.log(val, self.global_step)
        return

    @torch.no_grad()
    def validate(self, num_prompts: Optional[int] = None) -> Dict[str, float]:
        """Greedy validation rollouts + reward scoring (the reference's
        _validate capability, stream_ray_trainer.py:305-313; validation
        falls back to local colocated generation like
        sglang_rollout_remote.py:184-196)."""
        c = self.config
        ro = c.actor_rollout_ref.rollout
        ds = self.val_dataset if getattr(self, "val_dataset", None) \
            is not None else self.dataset
        nval = num_prompts or max(c.data.train_batch_size // self.world, 1)
        if self.sched_coordinated or self.tp_rollout:
            nval = max(nval - nval % self.world, self.world)
        idx = list(range(min(nval, len(ds))))
        batch = ds.batch(idx)
        sampling = SamplingParams(temperature=0.0,
                                  max_new_tokens=ro.response_length)
        if self.engine is not None:
            self.engine.resume_memory()      # no-op unless released
        self.publisher.publish()
        self.coordinator.submit(batch, sampling, 1)
        stream = (len(idx) // self.world
                  if (self.sched_coordinated or self.tp_rollout)
                  else len(idx))
        groups = []
        for b in self.coordinator.stream_batches(stream):
            groups.append(b)
        if not groups:
            return {}
        full = TensorBatch.concat(groups)
        scores = self.reward_fn(full)
        seq_scores = scores.sum(-1)
        lens = full["response_mask"].sum(-1).float()
        out = {
            "val/score/mean": float(seq_scores.mean()),
            "val/score/max": float(seq_scores.max()),
            "val/response_length/mean": float(lens.mean()),
            "val/n": float(len(full)),
        }
        # per-data-source breakdown (the reference logs val scores by
        # source via the reward-manager dispatch)
        ds = full.non_tensors.get("data_source")
        if ds is not None:
            import numpy as _np
            for src in sorted({str(x) for x in ds}):
                m = torch.from_numpy(_np.array([str(x) == src for x in ds]))
                out[f"val/score/{src}/mean"] = float(seq_scores[m].mean())
        if dist.is_available() and dist.is_initialized():
            t = torch.tensor([out["val/score/mean"], out["val/n"]])
            dist.all_reduce(t, group=self.pg)
            out["val/score/mean"] = float(t[0] / self.world)
            out["val/n"] = float(t[1])
        if ro.free_cache_engine and self.engine is not None \
                and not self.engine.has_work():
            self.engine.release_memory()     # back to trainer_mode
        return out

    # ------------------------------------------------------------- one step
    def _run_step(self, global_batch: TensorBatch, n: int, local_bs: int,
                  local_stream: int, mini_local: int, local_total: int,
                  timing: Dict[str, float]) -> Dict[str, float]:
        c = self.config
        ro = c.actor_rollout_ref.rollout

        # free_cache_engine: the reference's rollout_mode/trainer_mode
        # memory dance (stream_fsdp_workers.py:467-492) — resume the KV
        # pool for generation, release it once the stream drains.  Off by
        # default on MI355X (288 GB fits both resident).
        if ro.free_cache_engine and self.engine is not None:
            self.engine.resume_memory()
        # 1. publish current weights to the rollout plane (every iteration,
        #    incl. bootstrap — §3.3)
        with marked_timer("weight_sync", timing):
            self.publisher.publish()

        # 2. submit prompts: co-located = this rank's shard to its engine;
        #    disagg = the whole batch through the scheduler (rank 0)
        sampling = SamplingParams(
            temperature=ro.sampling.temperature,
            top_k=ro.sampling.top_k, top_p=ro.sampling.top_p,
            max_new_tokens=ro.response_length)
        if self.sched_coordinated or self.tp_rollout:
            submit_batch = global_batch     # identical on every rank
        else:
            submit_batch = global_batch.slice(
                slice(self.rank * local_bs, (self.rank + 1) * local_bs))
        # ReMax: an extra greedy rollout of the same prompts supplies the
        # per-prompt reward baseline (verl remax recipe; estimator in
        # core/algos.py).  Runs on the freshly published weights.
        if c.algorithm.adv_estimator == "remax":
            with marked_timer("gen_baseline", timing):
                self._collect_remax_baselines(submit_batch, local_bs)
        with marked_timer("gen_submit", timing):
            self.coordinator.submit(submit_batch, sampling, n)

        # 3. stream loop
        all_metrics: Dict[str, List[float]] = {}
        ibatches: List[TensorBatch] = []
        cum = 0                       # cumulative local samples trained
        warmup = c.trainer.critic_warmup
        stream_it = iter(self.coordinator.stream_batches(local_stream))
        while True:
            # engine decode runs inside the generator pull: time it so the
            # step breakdown attributes rollout-vs-update wall explicitly
            with marked_timer("gen_wait", timing):
                ibatch = next(stream_it, None)
            if ibatch is None:
                break
            with marked_timer("prep", timing):
                ibatch = self._prepare_ibatch(ibatch, timing)
            ibatches.append(ibatch)
            # 4. slice at minibatch boundaries (global bookkeeping via equal
            #    per-rank counts)
            with marked_timer("update", timing):
                off = 0
                bs = len(ibatch)
                while off < bs:
                    next_boundary = ((cum // mini_local) + 1) * mini_local
                    take = min(bs - off, next_boundary - cum)
                    sl = ibatch.slice(slice(off, off + take))
                    cum += take
                    off += take
                    is_opt = (cum % mini_local == 0)
                    is_lr = (cum >= local_total)
                    scale = take / mini_local
                    if self.use_critic:
                        m = self.critic.update_critic_stream(
                            sl, is_opt, is_lr, accum_scale=scale)
                        for k, v in m.items():
                            all_metrics.setdefault(k, []).extend(v)
                    if warmup <= self.global_step:
                        m = self.actor.update_policy_stream(
                            sl, is_opt, is_lr, accum_scale=scale)
                        for k, v in m.items():
                            all_metrics.setdefault(k, []).extend(v)
        full = TensorBatch.concat(ibatches) if ibatches else TensorBatch()
        self._last_full_batch = full
        if ro.free_cache_engine and self.engine is not None \
                and not self.engine.has_work():
            self.engine.release_memory()
        metrics = reduce_metrics(all_metrics)
        metrics.update(compute_data_metrics(full, self.use_critic))
        if torch.cuda.is_available():
            metrics["perf/max_memory_allocated_gb"] = \
                torch.cuda.max_memory_allocated() / (1 << 30)
            metrics["perf/max_memory_reserved_gb"] = \
                torch.cuda.max_memory_reserved() / (1 << 30)
        # rollout data dump (stream_ray_trainer.py:585-587 capability)
        dump = getattr(c.trainer, "rollout_data_dir", "")
        if dump and self.rank == 0 and len(full):
            import json as _json
            import os as _os
            _os.makedirs(dump, exist_ok=True)
            with open(_os.path.join(
                    dump, f"step_{self.global_step}.jsonl"), "w") as f:
                scores = full["token_level_scores"].sum(-1)
                rm = full["response_mask"]
                for i in range(len(full)):
                    f.write(_json.dumps({
                        "uid": str(full["uid"][i]),
                        "response_ids":
                            full["responses"][i][rm[i].bool()].tolist(),
                        "score": float(scores[i]),
                    }) + "\n")
        # feedback to the scheduler's adaptive local-gen time-box
        # (stream_ray_trainer.py:691-704 capability)
        if self.sched_coordinated and self.rank == 0 and \
                hasattr(self.coordinator, "update_metrics"):
            step_t = timing.get("step", 0.0)
            # "prep" already contains the reward/adv/logprob sub-timers
            busy = sum(timing.get(k, 0.0) for k in
                       ("update", "prep", "weight_sync", "gen_baseline"))
            bubble = max(step_t - busy, 0.0)
            thr = len(full) / step_t if step_t > 0 else 0.0
            new_window = self.coordinator.update_metrics(step_t, bubble, thr)
            metrics["training/max_local_gen_s"] = new_window
            metrics["training/num_rollout_instances"] = float(
                len(self.coordinator.scheduler.instances()))
            metrics["training/num_migrations"] = float(sum(
                b.meta_info.get("num_migrations", 0) for b in ibatches))
        return metrics

    def _collect_remax_baselines(self, submit_batch: TensorBatch,
                                 local_bs: int) -> None:
        """Greedy (temperature=0, n=1) rollout of this step's prompts;
        sequence reward per prompt becomes the ReMax baseline, keyed by uid
        and all-gathered so every rank can score any sample."""
        ro = self.config.actor_rollout_ref.rollout
        greedy = SamplingParams(temperature=0.0,
                                max_new_tokens=ro.response_length)
        self.coordinator.submit(submit_batch, greedy, 1)
        parts = [b for b in self.coordinator.stream_batches(local_bs)]
        batch = TensorBatch.concat(parts)
        seq_scores = self.reward_fn(batch).sum(dim=-1)
        local = {str(u): float(v)
                 for u, v in zip(batch["uid"], seq_scores.tolist())}
        if dist.is_initialized() and dist.get_world_size(self.pg) > 1:
            box = [None] * dist.get_world_size(self.pg)
            dist.all_gather_object(box, local, group=self.pg)
            merged: Dict[str, float] = {}
            for d in box:
                merged.update(d)
            local = merged
        self._remax_baselines = local

    # ------------------------------------------------- per-ibatch preparation
    def _prepare_ibatch(self, ibatch: TensorBatch,
                        timing: Dict[str, float]) -> TensorBatch:
        c = self.config
        # reward
        with marked_timer("reward", timing):
            scores = self.reward_fn(ibatch)
            ibatch["token_level_scores"] = scores
        # old log probs (actor fwd, exact — reference :425-439)
        with marked_timer("old_log_prob", timing):
            old_lp, _ = self.actor.compute_log_prob(ibatch)
            ibatch["old_log_probs"] = old_lp.cpu()
            if "rollout_log_probs" in ibatch:
                rm = ibatch["response_mask"].float()
                diff = (old_lp.cpu() - ibatch["rollout_log_probs"]).abs()
                ibatch.meta_info["rollout_logprob_diff"] = float(
                    algos.masked_mean(diff, rm))
        # ref log probs
        if self.ref is not None:
            with marked_timer("ref_log_prob", timing):
                ref_lp, _ = self.ref.compute_log_prob(ibatch)
                ibatch["ref_log_probs"] = ref_lp.cpu()
        # values
        if self.use_critic:
            with marked_timer("values", timing):
                ibatch["values"] = self.critic.compute_values(ibatch).cpu()
        # KL penalty into reward + advantage (driver-side, :465-498)
        with marked_timer("adv", timing):
            scores = ibatch["token_level_scores"]
            if c.algorithm.use_kl_in_reward and self.ref is not None:
                rewards, kl = algos.apply_kl_penalty(
                    scores, ibatch["old_log_probs"], ibatch["ref_log_probs"],
                    ibatch["response_mask"].float(),
                    c.algorithm.kl_ctrl.kl_coef, c.algorithm.kl_penalty)
                ibatch["token_level_rewards"] = rewards
            else:
                ibatch["token_level_rewards"] = scores
            baselines = None
            if c.algorithm.adv_estimator == "remax":
                baselines = torch.tensor(
                    [self._remax_baselines[str(u)] for u in ibatch["uid"]],
                    dtype=torch.float32)
            adv, ret = algos.compute_advantage(
                ibatch["token_level_rewards"],
                ibatch["response_mask"].float(),
                c.algorithm.adv_estimator,
                values=ibatch.tensors.get("values"),
                index=ibatch.non_tensors.get("uid"),
                gamma=c.algorithm.gamma, lam=c.algorithm.lam,
                norm_adv_by_std_in_grpo=c.algorithm.norm_adv_by_std_in_grpo,
                reward_baselines=baselines)
            ibatch["advantages"] = adv
            ibatch["returns"] = ret
        return ibatch

    def _should_save_esi(self, last_step_s: float) -> bool:
        """Spot/ESI expiration-aware checkpointing (the reference's
        should_save_ckpt_esi, stream_ray_trainer.py:604-623): when
        POLYRL_ESI_EXPIRE_AT (epoch seconds) is set and another step +
        save would not fit before expiry, save NOW.  Rank-0's decision is
        broadcast so every rank enters the (collective) save together."""
        import os as _os
        import time as _time
        save = False
        if self.rank == 0:
            exp = _os.environ.get("POLYRL_ESI_EXPIRE_AT")
            if exp and not getattr(self, "_esi_saved", False):
                remaining = float(exp) - _time.time()
                margin = max(2.0 * last_step_s, 30.0)
                save = remaining < margin
        if dist.is_initialized() and dist.get_world_size(self.pg) > 1:
            box = [save]
            dist.broadcast_object_list(box, src=0, group=self.pg)
            save = box[0]
        if save:
            self._esi_saved = True
        return save

    # ------------------------------------------------------------ checkpoint
    def save_checkpoint(self):
        extra = {"global_step": self.global_step}
        contents = self.config.actor_rollout_ref.actor.checkpoint_contents
        opt = self.actor.optimizer if "optimizer" in contents else None
        lr = self.actor.lr_scheduler if "extra" in contents else None
        self.ckpt_actor.save(self.global_step, self.actor.model, opt, lr,
                             extra)
        if self.use_critic:
            copt = self.critic.optimizer if "optimizer" in contents else None
            clr = self.critic.lr_scheduler if "extra" in contents else None
            self.ckpt_critic.save(self.global_step, self.critic.model,
                                  copt, clr, extra)

    def _maybe_resume(self):
        tr = self.config.trainer
        if tr.resume_mode == "disable":
            return
        step = None
        if tr.resume_mode == "resume_path":
            # explicit checkpoint dir: .../global_step_N
            import re as _re
            assert tr.resume_from_path, \
                "resume_mode=resume_path needs trainer.resume_from_path"
            m = _re.search(r"global_step_(\d+)", tr.resume_from_path)
            assert m, f"no global_step_N in {tr.resume_from_path!r}"
            step = int(m.group(1))
        ex = self.ckpt_actor.load(self.actor.model, self.actor.optimizer,
                                  self.actor.lr_scheduler, step=step)
        if ex is not None:
            self.global_step = int(ex.get("global_step", 0))
            if self.use_critic:
                self.ckpt_critic.load(self.critic.model, self.critic.optimizer,
                                      self.critic.lr_scheduler)
            if self.rank == 0:
                print(f"[resume] restored global_step={self.global_step}")
