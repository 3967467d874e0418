"""FSDP2 actor / critic / reference workers — SPMD, one process per GPU.

Replaces the reference's Ray worker layer (StreamActorRolloutRefWorker /
StreamCriticWorker, stream_fsdp_workers.py; StreamDataParallelPPOActor,
stream_dp_actor.py:85-231) with a collective SPMD design: every rank runs the
same method at the same time; FSDP2 (``fully_shard``) provides param
all-gather / grad reduce-scatter on RCCL over xGMI.

Streaming semantics preserved from the reference:
  * gradient accumulation ACROSS update calls — optimizer steps only when the
    cumulative streamed sample count crosses a ppo_mini_batch boundary
    (is_opt_step; stream_dp_actor.py:226-230),
  * dynamic micro-batching by token budget (ppo_max_token_len_per_gpu),
  * loss scaled per-slice so a minibatch's gradient matches the non-streamed
    equivalent.
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from ..config import ActorConfig, CriticConfig, OptimConfig
from ..core import algos
from ..core.seqlen import fixed_micro_batches, prepare_dynamic_batch, restore_dynamic_batch
from ..protocol import TensorBatch


def _maybe_fully_shard(model: nn.Module, mixed_precision: bool = True,
                       pg=None, reshard_after_forward: bool = True,
                       param_offload: bool = False,
                       param_dtype: str = "bfloat16",
                       reduce_dtype: str = "float32"):
    """Apply FSDP2 per decoder layer + root.  Works for world_size 1..N on
    nccl(RCCL) and gloo alike.  ``pg`` restricts sharding to a subgroup
    (disaggregated split: FSDP over the trainer ranks only).
    ``reshard_after_forward=False`` keeps gathered params resident across
    the step's many micro passes — ONE all-gather per step instead of one
    per micro fwd/bwd, paid with +full-param memory (288 GB affords it for
    the 8B benchmark config).  ``param_offload`` puts sharded params (and
    their optimizer state) in pinned host memory between uses — the 70B
    PPO memory lever (actor+critic+engine beyond 288 GB)."""
    if not (dist.is_available() and dist.is_initialized()):
        return model
    from torch.distributed.fsdp import MixedPrecisionPolicy, fully_shard
    mp = None
    if mixed_precision and next(model.parameters()).dtype == torch.bfloat16:
        mp = MixedPrecisionPolicy(param_dtype=getattr(torch, param_dtype),
                                  reduce_dtype=getattr(torch, reduce_dtype))
    kwargs = {"mp_policy": mp} if mp else {}
    kwargs["reshard_after_forward"] = reshard_after_forward
    if param_offload and torch.cuda.is_available():
        from torch.distributed.fsdp import CPUOffloadPolicy
        kwargs["offload_policy"] = CPUOffloadPolicy()
    if pg is not None:
        from torch.distributed.device_mesh import DeviceMesh
        dev_type = "cuda" if torch.cuda.is_available() else "cpu"
        kwargs["mesh"] = DeviceMesh.from_group(pg, dev_type)
    layers = None
    if hasattr(model, "model") and hasattr(model.model, "layers"):
        layers = model.model.layers
    elif hasattr(model, "h"):
        layers = model.h
    elif hasattr(model, "trunk"):
        layers = model.trunk.h
    if layers is not None:
        for layer in layers:
            fully_shard(layer, **kwargs)
    fully_shard(model, **kwargs)
    return model


def _build_optimizer(params, cfg: OptimConfig):
    """AdamW; fused (single multi-tensor kernel) on GPU — the optimizer
    step is pure bandwidth (~24 B/param of state+grad+param traffic), so
    the foreach default's extra kernel launches and intermediate reads
    are measurable at 8B params."""
    params = list(params)
    # fused only for plain CUDA tensors: DTensor (FSDP2 world>1) support
    # for the fused multi-tensor kernel is not validated on this stack
    fused = bool(params) and all(
        isinstance(p, torch.Tensor) and p.is_cuda
        and not hasattr(p, "placements") for p in params)
    try:
        return torch.optim.AdamW(params, lr=cfg.lr, betas=tuple(cfg.betas),
                                 eps=cfg.eps, weight_decay=cfg.weight_decay,
                                 fused=fused)
    except (RuntimeError, ValueError):     # fused unsupported (CPU/dtype)
        return torch.optim.AdamW(params, lr=cfg.lr, betas=tuple(cfg.betas),
                                 eps=cfg.eps, weight_decay=cfg.weight_decay)


def offload_optimizer_state(optimizer: torch.optim.Optimizer) -> None:
    """Move optimizer state tensors to host between steps (verl's
    offload_fsdp_optimizer capability — the fsdp.optimizer_offload knob).
    Paid with an H2D/D2H pass around each optimizer step; buys ~2 x 4 bytes
    x params of HBM for 70B-scale PPO."""
    for st in optimizer.state.values():
        for k, v in st.items():
            if torch.is_tensor(v) and v.device.type != "cpu":
                st[k] = v.to("cpu", non_blocking=True)


def load_optimizer_state(optimizer: torch.optim.Optimizer, device) -> None:
    for st in optimizer.state.values():
        for k, v in st.items():
            if torch.is_tensor(v) and str(v.device) != str(device):
                st[k] = v.to(device, non_blocking=True)


def _build_lr_scheduler(optimizer, cfg: OptimConfig):
    warmup = cfg.lr_warmup_steps
    if warmup <= 0 and cfg.lr_warmup_steps_ratio > 0 and cfg.total_training_steps > 0:
        warmup = int(cfg.lr_warmup_steps_ratio * cfg.total_training_steps)

    def lr_lambda(step):
        if warmup > 0 and step < warmup:
            return (step + 1) / warmup
        if cfg.warmup_style == "cosine" and cfg.total_training_steps > 0:
            prog = (step - warmup) / max(cfg.total_training_steps - warmup, 1)
            return 0.5 * (1 + math.cos(math.pi * min(prog, 1.0)))
        return 1.0

    return torch.optim.lr_scheduler.LambdaLR(optimizer, lr_lambda)


def _clip_grad_norm(model: nn.Module, max_norm: float) -> float:
    """Grad-norm clip that understands DTensor (FSDP2) gradients."""
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    if not grads:
        return 0.0
    total = torch.nn.utils.get_total_norm(grads)
    if hasattr(total, "full_tensor"):
        total = total.full_tensor()
    torch.nn.utils.clip_grads_with_norm_(
        (p for p in model.parameters() if p.grad is not None),
        max_norm, total)
    return float(total)


def _sync_num_micro(k_local: int, pg=None) -> int:
    """FSDP2 unshard/reshard are collectives: every rank must run the SAME
    number of micro-batch fwd (and bwd) passes.  All-reduce the max count;
    ranks short of it run dummy passes (outputs discarded / zero-weighted)."""
    if not (dist.is_available() and dist.is_initialized()) or \
            dist.get_world_size(pg) == 1:
        return k_local
    t = torch.tensor([k_local], dtype=torch.int64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX, group=pg)
    return int(t.item())


def _model_inputs(batch: TensorBatch, device):
    ids = batch["input_ids"].to(device)
    am = batch["attention_mask"].to(device)
    pos = batch["position_ids"].to(device)
    resp = batch["responses"].to(device)
    return ids, am, pos, resp


_SP_GROUPS: Dict[int, tuple] = {}   # sp_size -> (sp_group, dp_group)


def register_sp_groups(sp_size: int, sp_group, dp_group) -> None:
    """Pre-register SP/DP groups built over a rank SUBSET (disaggregated
    trainer subgroup).  dist.new_group is a world collective, so the groups
    must be created on every rank BEFORE the trainer/rollout role branch;
    the trainer ranks then hand them to the workers through this hook."""
    _SP_GROUPS[sp_size] = (sp_group, dp_group)


def _get_sp_groups(sp_size: int):
    """Build (once) and cache the SP/DP process groups for this rank.
    Safe to call from actor+critic+ref constructors: the group-creation
    order is identical on every rank."""
    if sp_size not in _SP_GROUPS:
        from ..parallel.ulysses import build_sp_groups
        _SP_GROUPS[sp_size] = build_sp_groups(sp_size)
    return _SP_GROUPS[sp_size]


def _setup_ulysses(model: nn.Module, sp_size: int):
    """Install the Ulysses context on a llama-family trunk; returns the SP
    group (or None)."""
    if sp_size <= 1 or not (dist.is_available() and dist.is_initialized()):
        return None
    trunk = getattr(model, "model", None)
    if trunk is None or not hasattr(trunk, "ulysses"):
        raise ValueError("ulysses SP requires the llama-family trainer model")
    from ..parallel.ulysses import UlyssesContext
    sp_group, _ = _get_sp_groups(sp_size)
    trunk.ulysses = UlyssesContext(sp_group)
    return sp_group


def _gather_rows(batch: TensorBatch, group):
    """Pool batch rows across the SP group so every rank processes the SAME
    rows (the reference slices sequences of a shared batch the same way —
    verl FSDPUlyssesShardingManager preprocess capability).  Returns
    (pooled_batch, slice_of_this_ranks_rows)."""
    sp = dist.get_world_size(group)
    r = dist.get_rank(group)
    if sp == 1:
        return batch, slice(0, len(batch))
    boxes = [None] * sp
    dist.all_gather_object(boxes, batch, group=group)
    off = sum(len(b) for b in boxes[:r])
    return TensorBatch.concat(boxes), slice(off, off + len(batch))


class ActorWorker:
    """Policy model under FSDP2: compute_log_prob + update_policy_stream."""

    def __init__(self, model: nn.Module, cfg: ActorConfig, device="cpu",
                 is_ref: bool = False, pg=None):
        self.cfg = cfg
        self.device = device
        self.is_ref = is_ref
        self.pg = pg
        self.model = _maybe_fully_shard(
            model, pg=pg,
            reshard_after_forward=cfg.fsdp.reshard_after_forward,
            param_offload=cfg.fsdp.param_offload,
            param_dtype=cfg.fsdp.mixed_precision_dtype,
            reduce_dtype=cfg.fsdp.reduce_dtype)
        if pg is not None and cfg.ulysses_sequence_parallel_size > 1:
            # disagg: SP groups over the trainer subgroup must have been
            # pre-built (a world collective) before the role branch —
            # main_stream.py does this via register_sp_groups()
            assert cfg.ulysses_sequence_parallel_size in _SP_GROUPS, \
                ("Ulysses SP inside a disaggregated trainer subgroup needs "
                 "pre-built groups: call workers.register_sp_groups() on the "
                 "trainer ranks (main_stream.py wires this)")
        self.sp_group = _setup_ulysses(model,
                                       cfg.ulysses_sequence_parallel_size)
        self.sp_size = cfg.ulysses_sequence_parallel_size \
            if self.sp_group is not None else 1
        if not is_ref:
            self.optimizer = _build_optimizer(self.model.parameters(), cfg.optim)
            self.lr_scheduler = _build_lr_scheduler(self.optimizer, cfg.optim)
        self._accum_tokens = 0.0  # diagnostics

    # ----------------------------------------------------------- log prob
    @torch.no_grad()
    def compute_log_prob(self, batch: TensorBatch, want_entropy: bool = False
                         ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
        """old/ref log-probs of the sampled responses.  (B, resp_len) fp32."""
        self.model.eval()
        my_rows = slice(0, len(batch))
        if self.sp_group is not None:
            batch, my_rows = _gather_rows(batch, self.sp_group)
        micro, parts = self._split(batch)
        k = _sync_num_micro(len(micro), self.pg)
        lps, ents = [], []
        for mb in micro:
            lp, ent = self._forward_logprobs(mb, want_entropy)
            lps.append(lp)
            ents.append(ent)
        for _ in range(k - len(micro)):  # dummy collective-alignment passes
            self._forward_logprobs(micro[0].slice(slice(0, 1)), want_entropy)
        lp = restore_dynamic_batch(torch.cat(lps), parts)[my_rows]
        ent = (restore_dynamic_batch(torch.cat(ents), parts)[my_rows]
               if want_entropy else None)
        return lp, ent

    def _split(self, batch: TensorBatch):
        if self.cfg.use_dynamic_bsz:
            budget = self.cfg.ppo_max_token_len_per_gpu * \
                self.cfg.ulysses_sequence_parallel_size
            return prepare_dynamic_batch(batch, budget)
        mbs = self.cfg.ppo_micro_batch_size_per_gpu or len(batch)
        return fixed_micro_batches(batch, mbs)

    def _forward_logprobs(self, mb: TensorBatch, want_entropy: bool,
                          grad: bool = False):
        ids, am, pos, resp = _model_inputs(mb, self.device)
        Lr = resp.shape[1]
        ctx = torch.enable_grad() if grad else torch.no_grad()
        with ctx:
            if self.sp_group is not None:
                return self._forward_logprobs_sp(ids, am, pos, Lr,
                                                 want_entropy)
            logits = self.model(ids, attention_mask=am, position_ids=pos,
                                logits_slice=slice(-Lr - 1, -1))
            fused_ok = (logits.is_cuda and logits.dtype == torch.bfloat16
                        and getattr(self.cfg, "use_fused_kernels", True))
            if fused_ok and not want_entropy:
                # fused CE-style gather: no fp32 logits materialization,
                # no saved log_softmax (ops/csrc/logprobs.hip train path)
                import polyrl_amd.ops as pops
                B = logits.shape[0]
                flat = logits.reshape(-1, logits.shape[-1])
                if grad:
                    lp = pops.gather_logprobs_train(flat, resp.reshape(-1))
                else:
                    lp = pops.gather_logprobs(flat, resp.reshape(-1))
                lp = lp.view(B, Lr)
                return lp, torch.zeros_like(lp)
            if fused_ok and want_entropy and not grad:
                import polyrl_amd.ops as pops
                B = logits.shape[0]
                flat = logits.reshape(-1, logits.shape[-1])
                lp, ent = pops.gather_logprobs(flat, resp.reshape(-1),
                                               want_entropy=True)
                return lp.view(B, Lr), ent.view(B, Lr)
            logits = logits.float()
            lp = algos.logprobs_from_logits(logits, resp)
            ent = algos.entropy_from_logits(logits) if want_entropy else \
                torch.zeros_like(lp)
        return lp, ent

    def _forward_logprobs_sp(self, ids, am, pos, Lr, want_entropy):
        """Ulysses path: full inputs on every SP rank; per-shard logits ->
        per-shard logprobs -> autograd-aware sequence gather -> response
        slice (verl's gather_outputs_and_unpad capability).

        When the trunk takes the packed x SP path (use_remove_padding with
        heads % sp == 0) logits come back as the rank's (Tp/sp, V) PACKED
        token shard; labels are packed the same way and the gathered
        logprobs are scattered onto the (B, L) grid at the end."""
        from ..parallel.ulysses import gather_seq, pad_to_multiple, slice_for_rank
        sp = self.sp_size
        B, L = ids.shape
        ids_p = pad_to_multiple(ids, sp, 1)
        am_p = pad_to_multiple(am, sp, 1)
        pos_p = pad_to_multiple(pos, sp, 1)
        logits = self.model(ids_p, attention_mask=am_p,
                            position_ids=pos_p).float()
        labels_p = torch.cat(
            [ids_p[:, 1:], torch.zeros(B, 1, dtype=ids.dtype,
                                       device=ids.device)], dim=1)
        trunk = getattr(self.model, "model", None)
        meta = getattr(trunk, "pack_sp_meta", None)
        if meta is not None:                       # packed x SP
            labels_pk = labels_p[meta["valid"]]
            pad = meta["Tp"] - meta["T"]
            if pad:
                labels_pk = torch.cat([labels_pk, labels_pk.new_zeros(pad)])
            r = torch.distributed.get_rank(self.sp_group)
            sh = meta["shard"]
            labels_shard = labels_pk[r * sh:(r + 1) * sh]
            lp_shard = algos.logprobs_from_logits(logits, labels_shard)
            lp_pk = gather_seq(lp_shard, 0, self.sp_group)[:meta["T"]]
            lp_grid = torch.zeros(B, ids_p.shape[1], dtype=lp_pk.dtype,
                                  device=lp_pk.device)
            lp_grid[meta["valid"]] = lp_pk
            lp = lp_grid[:, L - Lr - 1:L - 1]
            if want_entropy:
                ent_shard = algos.entropy_from_logits(logits)
                ent_pk = gather_seq(ent_shard, 0, self.sp_group)[:meta["T"]]
                ent_grid = torch.zeros_like(lp_grid)
                ent_grid[meta["valid"]] = ent_pk
                ent = ent_grid[:, L - Lr - 1:L - 1]
            else:
                ent = torch.zeros_like(lp)
            return lp, ent
        labels_shard = slice_for_rank(labels_p, 1, self.sp_group)
        lp_shard = algos.logprobs_from_logits(logits, labels_shard)
        lp = gather_seq(lp_shard, 1, self.sp_group)[:, :L]
        lp = lp[:, L - Lr - 1:L - 1]
        if want_entropy:
            ent_shard = algos.entropy_from_logits(logits)
            ent = gather_seq(ent_shard, 1, self.sp_group)[:, :L]
            ent = ent[:, L - Lr - 1:L - 1]
        else:
            ent = torch.zeros_like(lp)
        return lp, ent

    # ------------------------------------------------------------- update
    def update_policy_stream(self, batch: TensorBatch, is_opt_step: bool,
                             is_lr_step: bool, accum_scale: float
                             ) -> Dict[str, List[float]]:
        """fwd/bwd on one streamed slice; optimizer step at minibatch boundary.

        accum_scale = slice_samples / minibatch_samples normalization factor so
        the accumulated gradient equals the full-minibatch gradient.
        """
        assert not self.is_ref
        self.model.train()
        metrics: Dict[str, List[float]] = {}
        if self.sp_group is not None:
            batch, _ = _gather_rows(batch, self.sp_group)
        micro, _ = self._split(batch)
        k = _sync_num_micro(len(micro), self.pg)
        # dummy zero-weight passes keep FSDP fwd/bwd collectives aligned
        dummies = [micro[0].slice(slice(0, 1)) for _ in range(k - len(micro))]
        n_total = len(batch)
        loss_fn = algos.get_policy_loss_fn(self.cfg.policy_loss_type)
        for mi, mb in enumerate(micro + dummies):
            is_dummy = mi >= len(micro)
            response_mask = mb["response_mask"].to(self.device)
            old_log_prob = mb["old_log_probs"].to(self.device)
            advantages = mb["advantages"].to(self.device)
            log_prob, ent = self._forward_logprobs(
                mb, want_entropy=bool(self.cfg.entropy_coeff), grad=True)
            extra = {}
            if self.cfg.tis_imp_ratio_cap > 0 and \
                    "rollout_log_probs" in mb.tensors and \
                    self.cfg.policy_loss_type == "vanilla":
                # truncated importance sampling: rollout happened on the
                # engine's (possibly stale) weights; reweight each token by
                # min(pi_old/pi_rollout, cap) — stream_dp_actor.py:153-224
                rollout_lp = mb["rollout_log_probs"].to(self.device)
                extra["importance_weights"] = torch.exp(
                    (old_log_prob - rollout_lp).clamp(-20, 20)
                ).clamp(max=self.cfg.tis_imp_ratio_cap)
            pg_loss, pg_clipfrac, ppo_kl, pg_clipfrac_lower = loss_fn(
                old_log_prob=old_log_prob, log_prob=log_prob,
                advantages=advantages, response_mask=response_mask,
                clip_ratio=self.cfg.clip_ratio,
                clip_ratio_low=self.cfg.clip_ratio_low,
                clip_ratio_high=self.cfg.clip_ratio_high,
                loss_agg_mode=self.cfg.loss_agg_mode, **extra)
            loss = pg_loss
            if self.cfg.entropy_coeff:
                loss = loss - self.cfg.entropy_coeff * algos.agg_loss(
                    ent, response_mask, self.cfg.loss_agg_mode)
            if self.cfg.use_kl_loss and "ref_log_probs" in mb.tensors:
                ref_lp = mb["ref_log_probs"].to(self.device)
                kld = algos.kl_penalty(log_prob, ref_lp, self.cfg.kl_loss_type)
                kl_loss = algos.agg_loss(kld, response_mask, self.cfg.loss_agg_mode)
                loss = loss + self.cfg.kl_loss_coef * kl_loss
                if not is_dummy:
                    metrics.setdefault("actor/kl_loss", []).append(kl_loss.item())
            # micro-batch weight within slice x slice weight within minibatch;
            # x sp_size compensates FSDP's world-mean over SP-duplicated rows
            w = 0.0 if is_dummy else \
                (len(mb) / n_total) * accum_scale * self.sp_size
            (loss * w).backward()
            if not is_dummy:
                metrics.setdefault("actor/pg_loss", []).append(pg_loss.item())
                metrics.setdefault("actor/pg_clipfrac", []).append(pg_clipfrac.item())
                metrics.setdefault("actor/ppo_kl", []).append(ppo_kl.item())
        if is_opt_step:
            gn = self._optimizer_step()
            metrics.setdefault("actor/grad_norm", []).append(gn)
        if is_lr_step:
            self.lr_scheduler.step()
            metrics.setdefault("actor/lr", []).append(
                self.lr_scheduler.get_last_lr()[0])
        return metrics

    def _optimizer_step(self) -> float:
        gn = _clip_grad_norm(self.model, self.cfg.optim.grad_clip)
        if not math.isfinite(gn):
            self.optimizer.zero_grad()
            return gn
        offl = self.cfg.fsdp.optimizer_offload and self.device != "cpu"
        if offl:
            load_optimizer_state(self.optimizer, self.device)
        self.optimizer.step()
        if offl:
            offload_optimizer_state(self.optimizer)
        self.optimizer.zero_grad()
        return gn


class CriticWorker:
    """Value model under FSDP2: compute_values + update_critic_stream."""

    def __init__(self, model: nn.Module, cfg: CriticConfig, device="cpu",
                 pg=None):
        self.cfg = cfg
        self.device = device
        self.pg = pg
        self.model = _maybe_fully_shard(
            model, pg=pg,
            reshard_after_forward=cfg.fsdp.reshard_after_forward,
            param_offload=cfg.fsdp.param_offload,
            param_dtype=cfg.fsdp.mixed_precision_dtype,
            reduce_dtype=cfg.fsdp.reduce_dtype)
        if pg is not None and \
                getattr(cfg, "ulysses_sequence_parallel_size", 1) > 1:
            assert cfg.ulysses_sequence_parallel_size in _SP_GROUPS, \
                ("Ulysses SP inside a disaggregated trainer subgroup needs "
                 "pre-built groups: call workers.register_sp_groups() on the "
                 "trainer ranks (main_stream.py wires this)")
        self.sp_group = _setup_ulysses(
            model, getattr(cfg, "ulysses_sequence_parallel_size", 1))
        self.sp_size = cfg.ulysses_sequence_parallel_size \
            if self.sp_group is not None else 1
        self.optimizer = _build_optimizer(self.model.parameters(), cfg.optim)
        self.lr_scheduler = _build_lr_scheduler(self.optimizer, cfg.optim)

    def _forward_values(self, mb: TensorBatch) -> torch.Tensor:
        """(B, Lr) value predictions; SP-aware (shard -> gather -> slice)."""
        ids, am, pos, resp = _model_inputs(mb, self.device)
        Lr = resp.shape[1]
        if self.sp_group is None:
            values = self.model(ids, attention_mask=am, position_ids=pos)
            return values[:, -Lr - 1:-1].float()
        from ..parallel.ulysses import gather_seq, pad_to_multiple
        sp = self.sp_size
        L = ids.shape[1]
        ids_p = pad_to_multiple(ids, sp, 1)
        v_shard = self.model(ids_p,
                             attention_mask=pad_to_multiple(am, sp, 1),
                             position_ids=pad_to_multiple(pos, sp, 1))
        trunk = getattr(self.model, "model", None)
        meta = getattr(trunk, "pack_sp_meta", None)
        if meta is not None:                       # packed x SP token shard
            v_pk = gather_seq(v_shard.float(), 0,
                              self.sp_group)[:meta["T"]]
            v_grid = torch.zeros(ids_p.shape, dtype=v_pk.dtype,
                                 device=v_pk.device)
            v_grid[meta["valid"]] = v_pk
            return v_grid[:, L - Lr - 1:L - 1]
        v = gather_seq(v_shard.float(), 1, self.sp_group)[:, :L]
        return v[:, L - Lr - 1:L - 1]

    def _split(self, batch: TensorBatch):
        if self.cfg.use_dynamic_bsz:
            budget = self.cfg.ppo_max_token_len_per_gpu * \
                getattr(self.cfg, "ulysses_sequence_parallel_size", 1)
            return prepare_dynamic_batch(batch, budget)
        mbs = self.cfg.ppo_micro_batch_size_per_gpu or len(batch)
        return fixed_micro_batches(batch, mbs)

    @torch.no_grad()
    def compute_values(self, batch: TensorBatch) -> torch.Tensor:
        self.model.eval()
        my_rows = slice(0, len(batch))
        if self.sp_group is not None:
            batch, my_rows = _gather_rows(batch, self.sp_group)
        micro, parts = self._split(batch)
        k = _sync_num_micro(len(micro), self.pg)
        outs = []
        for mi in range(k):
            mb = micro[mi] if mi < len(micro) else micro[0].slice(slice(0, 1))
            values = self._forward_values(mb)
            if mi < len(micro):
                outs.append(values)
        return restore_dynamic_batch(torch.cat(outs), parts)[my_rows]

    def update_critic_stream(self, batch: TensorBatch, is_opt_step: bool,
                             is_lr_step: bool, accum_scale: float
                             ) -> Dict[str, List[float]]:
        self.model.train()
        metrics: Dict[str, List[float]] = {}
        if self.sp_group is not None:
            batch, _ = _gather_rows(batch, self.sp_group)
        micro, _ = self._split(batch)
        k = _sync_num_micro(len(micro), self.pg)
        dummies = [micro[0].slice(slice(0, 1)) for _ in range(k - len(micro))]
        n_total = len(batch)
        for mi, mb in enumerate(micro + dummies):
            is_dummy = mi >= len(micro)
            response_mask = mb["response_mask"].to(self.device)
            values = mb["values"].to(self.device)
            returns = mb["returns"].to(self.device)
            vpreds = self._forward_values(mb)
            vf_loss, vf_clipfrac = algos.compute_value_loss(
                vpreds, returns, values, response_mask,
                cliprange_value=self.cfg.cliprange_value,
                loss_agg_mode=self.cfg.loss_agg_mode)
            w = 0.0 if is_dummy else \
                (len(mb) / n_total) * accum_scale * self.sp_size
            (vf_loss * w).backward()
            if not is_dummy:
                metrics.setdefault("critic/vf_loss", []).append(vf_loss.item())
                metrics.setdefault("critic/vf_clipfrac", []).append(vf_clipfrac.item())
        if is_opt_step:
            gn = _clip_grad_norm(self.model, self.cfg.optim.grad_clip)
            if math.isfinite(gn):
                offl = self.cfg.fsdp.optimizer_offload and self.device != "cpu"
                if offl:
                    load_optimizer_state(self.optimizer, self.device)
                self.optimizer.step()
                if offl:
                    offload_optimizer_state(self.optimizer)
            self.optimizer.zero_grad()
            metrics.setdefault("critic/grad_norm", []).append(gn)
        if is_lr_step:
            self.lr_scheduler.step()
        return metrics
