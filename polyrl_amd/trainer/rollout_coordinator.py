"""Rollout coordination: submit prompt groups, stream finished groups back.

Streaming contract mirrors the reference: submission returns
immediately; finished prompt GROUPS (all n samples of one prompt) stream back
while the trainer updates on earlier ibatches; postprocess_samples builds the
padded TensorBatch (sglang_rollout_remote.py:318-391 capability).

Co-located mode: each trainer rank owns one in-process Engine on its GPU and
generates its own shard (instance-level DP).  Split / elastic mode goes
through the scheduler (polyrl_amd/scheduler) with the same group-stream
interface.
"""
from __future__ import annotations

from typing import Dict, Iterator, List, Optional

import numpy as np
import torch

from ..protocol import TensorBatch
from ..rollout.engine import Engine, RequestOutput, SamplingParams


def postprocess_groups(
    prompt_ids: torch.Tensor,        # (G, Lp) left-padded local prompts
    prompt_mask: torch.Tensor,       # (G, Lp)
    group_uids: List[str],
    outputs: List[List[RequestOutput]],  # per group: n sample outputs
    response_length: int,
    pad_token_id: int = 0,
    device="cpu",
    group_extras: Optional[List[dict]] = None,
) -> TensorBatch:
    """Build the training batch for a set of finished prompt groups.
    ``group_extras``: per-group dict of extra per-row values (data_source,
    ground_truth, ...) carried through from the dataset so reward managers
    can score real data."""
    G = len(outputs)
    n = len(outputs[0])
    Lp = prompt_ids.shape[1]
    B = G * n
    resp = torch.full((B, response_length), pad_token_id, dtype=torch.long)
    resp_mask = torch.zeros(B, response_length, dtype=torch.long)
    presence = torch.zeros(B, response_length, dtype=torch.long)
    rollout_lp = torch.zeros(B, response_length, dtype=torch.float32)
    for g in range(G):
        for s, out in enumerate(outputs[g]):
            i = g * n + s
            L = min(len(out.output_ids), response_length)
            resp[i, :L] = torch.tensor(out.output_ids[:L], dtype=torch.long)
            presence[i, :L] = 1
            lm = getattr(out, "loss_mask", None)
            if lm is not None:
                # multi-turn: user/tool tokens attend but carry no loss
                resp_mask[i, :L] = torch.tensor(lm[:L], dtype=torch.long)
            else:
                resp_mask[i, :L] = 1
            rollout_lp[i, :L] = torch.tensor(out.output_logprobs[:L])
    prompts = prompt_ids.repeat_interleave(n, dim=0)
    pmask = prompt_mask.repeat_interleave(n, dim=0)
    input_ids = torch.cat([prompts, resp], dim=1)
    attention_mask = torch.cat([pmask, presence], dim=1)
    position_ids = torch.clamp(torch.cumsum(attention_mask, dim=1) - 1, min=0)
    uids = np.array([group_uids[g] for g in range(G) for _ in range(n)],
                    dtype=object)
    non_tensors = {"uid": uids}
    if group_extras and any(group_extras):
        keys = set()
        for e in group_extras:
            keys.update((e or {}).keys())
        for k in sorted(keys):
            non_tensors[k] = np.array(
                [(group_extras[g] or {}).get(k) for g in range(G)
                 for _ in range(n)], dtype=object)
    return TensorBatch.from_dict(
        tensors={
            "prompts": prompts,
            "responses": resp,
            "input_ids": input_ids,
            "attention_mask": attention_mask,
            "response_mask": resp_mask,
            "position_ids": position_ids,
            "rollout_log_probs": rollout_lp,
        },
        non_tensors=non_tensors,
    ).to(device)


def load_interaction(mt_cfg) -> Optional[dict]:
    """Build the multi-turn kwargs dict from a MultiTurnConfig: loads the
    interaction fn from the configured python file (same pattern as the
    custom reward loader; reference: interaction_config_path,
    config/rollout.py:54)."""
    if mt_cfg is None or not getattr(mt_cfg, "enable", False):
        return None
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "polyrl_interaction", mt_cfg.interaction_path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    fn = getattr(mod, mt_cfg.interaction_name)
    return {"interaction": fn,
            "max_assistant_turns": mt_cfg.max_assistant_turns,
            "max_user_turns": mt_cfg.max_user_turns,
            "max_tool_response_length": mt_cfg.max_tool_response_length,
            "per_turn_max_tokens": getattr(mt_cfg, "per_turn_max_tokens", 0)}


class LocalRolloutCoordinator:
    """One in-process engine per trainer rank (co-located mode).

    ``shard=(rank, world)`` enables the TP-rollout mode (BASELINE config
    #5): every rank submits the SAME global batch to its shard of the ONE
    tensor-parallel engine (SPMD lockstep — the engine's collectives align
    because scheduling is deterministic and identical), and stream_batches
    yields this rank's 1/world slice of each finished round for training."""

    def __init__(self, engine: Engine, response_length: int,
                 pad_token_id: int = 0, device="cpu",
                 shard=None, multi_turn=None, async_decode: bool = True):
        self.engine = engine
        self.response_length = response_length
        self.pad_token_id = pad_token_id
        self.device = device
        self.shard = shard
        # multi-turn (reference MultiTurnConfig capability): dict with
        # interaction fn + turn caps; see load_interaction()
        self.multi_turn = multi_turn
        self._mt_state: Dict[str, dict] = {}
        self._groups: Dict[str, dict] = {}
        self._submit_counter = 0
        # --- async decode pump (the in-process analog of the reference's
        # separate SGLang server process): a side thread steps the engine
        # on its own HIP stream so decode overlaps the trainer's update
        # kernels on the default stream.  Semantics are IDENTICAL to
        # serial decode: the engine holds its own weight buffers at the
        # step's version, and publish happens only between steps when the
        # pump is idle (stream_batches drains everything first).
        # Default OFF: measured ZERO overlap on uniform-length synthetic
        # benches (every sample finishes decode at the same iteration, so
        # updates cannot start earlier) at ~1.5% thread/stream overhead.
        # Turn ON (POLYRL_ASYNC_DECODE=1 / rollout.async_decode) for real
        # EOS-terminated workloads, where staggered completions let decode
        # of the tail overlap updates on the head.
        import os
        self.async_decode = (async_decode and not self.shard
                             and os.environ.get("POLYRL_ASYNC_DECODE", "0")
                             == "1")
        self._pump = None
        self._pump_err = None
        self._ready_q = None
        self._side_stream = None

    # ---------------------------------------------------------- decode pump
    def _pump_loop(self):
        import torch as _t
        try:
            use_stream = str(self.engine.device).startswith("cuda")
            if use_stream:
                if self._side_stream is None:
                    self._side_stream = _t.cuda.Stream()
                # order: step-start weight publish (default stream) must be
                # visible before this step's decode kernels
                self._side_stream.wait_stream(_t.cuda.current_stream())
                ctx = _t.cuda.stream(self._side_stream)
            else:
                import contextlib
                ctx = contextlib.nullcontext()
            with ctx:
                while self.engine.has_work():
                    for grp in self.poll():
                        self._ready_q.put(grp)
            if use_stream:
                # make emitted work visible to the default stream consumers
                _t.cuda.current_stream().synchronize()
        except BaseException as e:                 # noqa: BLE001
            self._pump_err = e
        finally:
            self._ready_q.put(None)               # end-of-stream sentinel

    def _start_pump(self):
        import queue
        import threading
        self._ready_q = queue.Queue()
        self._pump_err = None
        self._pump = threading.Thread(target=self._pump_loop, daemon=True)
        self._pump.start()

    # --------------------------------------------------------------- submit
    def submit(self, prompts: TensorBatch, sampling: SamplingParams, n: int):
        """Enqueue n samples per prompt.  prompts: tensors input_ids (G, Lp),
        attention_mask (G, Lp), non_tensor uid (G,)."""
        ids = prompts["input_ids"]
        mask = prompts["attention_mask"]
        uids = prompts["uid"]
        G = ids.shape[0]
        base = self._submit_counter
        self._submit_counter += 1
        for g in range(G):
            gid = f"b{base}-g{g}"
            raw = ids[g][mask[g].bool()].tolist()
            self._groups[gid] = {
                "uid": str(uids[g]),
                "extras": {k: v[g] for k, v in prompts.non_tensors.items()
                           if k != "uid"},
                "prompt_ids": ids[g].cpu(),
                "prompt_mask": mask[g].cpu(),
                "raw": raw,
                "sampling": sampling,
                "outputs": [None] * n,
                "done": 0,
                "n": n,
            }
            # shared prompt prefill + shared full KV pages for the group
            samp = sampling
            if self.multi_turn is not None and \
                    self.multi_turn.get("per_turn_max_tokens", 0) > 0:
                import copy as _copy
                samp = _copy.copy(sampling)
                samp.max_new_tokens = min(
                    samp.max_new_tokens,
                    self.multi_turn["per_turn_max_tokens"])
            self.engine.add_request_group(gid, raw, samp, n)

    # --------------------------------------------------------------- stream
    def pending_groups(self) -> int:
        return len(self._groups)

    def has_work(self) -> bool:
        return bool(self._groups)

    def poll(self, max_steps: int = 1) -> List[dict]:
        """Step the engine; return finished groups (dicts)."""
        done = []
        for _ in range(max_steps):
            if not self.engine.has_work():
                break
            for out in self.engine.step():
                gid, s = out.rid.rsplit("-s", 1)
                grp = self._groups.get(gid)
                if grp is None:
                    continue
                if self.multi_turn is not None:
                    out = self._multi_turn_step(grp, out)
                    if out is None:       # resubmitted for the next turn
                        continue
                grp["outputs"][int(s)] = out
                grp["done"] += 1
                if grp["done"] == grp["n"]:
                    done.append(self._groups.pop(gid))
        return done

    def _multi_turn_step(self, grp: dict, out: RequestOutput
                         ) -> Optional[RequestOutput]:
        """Absorb one finished assistant turn; either resubmit the request
        with the interaction's next user turn appended (returns None) or
        finalize the sample with a loss mask over assistant tokens only
        (reference capability: MultiTurnConfig + interaction, SURVEY.md
        §2.1 RolloutConfig row)."""
        mt = self.multi_turn
        st = self._mt_state.get(out.rid)
        if st is None:
            st = {"a_turns": 0, "u_turns": 0, "ids": [], "lp": [], "loss": []}
            self._mt_state[out.rid] = st
        st["ids"] += out.output_ids
        st["lp"] += out.output_logprobs
        st["loss"] += [1] * len(out.output_ids)
        st["a_turns"] += 1

        def finalize(reason: str) -> RequestOutput:
            self._mt_state.pop(out.rid, None)
            L = self.response_length
            return RequestOutput(rid=out.rid, output_ids=st["ids"][:L],
                                 output_logprobs=st["lp"][:L],
                                 finish_reason=reason,
                                 loss_mask=st["loss"][:L])

        budget = self.response_length - len(st["ids"])
        if (out.finish_reason == "abort"
                or st["a_turns"] >= mt["max_assistant_turns"]
                or st["u_turns"] >= mt["max_user_turns"]
                or budget <= 1):
            return finalize(out.finish_reason)
        user_ids, done = mt["interaction"](grp["raw"], list(st["ids"]))
        if done or not user_ids:
            return finalize("stop")
        user_ids = list(user_ids)[: mt.get("max_tool_response_length", 256)]
        if len(user_ids) >= budget:       # no room left to answer
            return finalize("length")
        st["ids"] += user_ids
        st["lp"] += [0.0] * len(user_ids)
        st["loss"] += [0] * len(user_ids)
        st["u_turns"] += 1
        import copy
        samp = copy.copy(grp["sampling"])
        samp.max_new_tokens = self.response_length - len(st["ids"])
        cap = mt.get("per_turn_max_tokens", 0)
        if cap > 0:
            samp.max_new_tokens = min(samp.max_new_tokens, cap)
        self.engine.add_request(out.rid, grp["raw"] + st["ids"], samp)
        return None

    def stream_batches(self, stream_size: int) -> Iterator[TensorBatch]:
        """Yield TensorBatches of EXACTLY stream_size samples (whole groups;
        requires stream_size % n == 0) until all submitted groups are
        consumed; the final batch carries any remainder.  Exact sizing keeps
        every DP rank's ibatch count identical — a requirement for the SPMD
        FSDP collectives (each ibatch triggers collective fwd/bwd).

        With ``shard=(r, w)``: collect w*stream_size samples per round and
        yield rank r's group-slice (deterministic, identical on all ranks —
        no communication)."""
        round_size = stream_size * (self.shard[1] if self.shard else 1)
        if self.async_decode and self._groups:
            yield from self._stream_batches_async(round_size)
            return
        ready: List[dict] = []
        ready_samples = 0
        while self._groups or ready:
            if self._groups:
                for grp in self.poll():
                    ready.append(grp)
                    ready_samples += grp["n"]
            while ready_samples >= round_size and ready:
                take, taken = [], 0
                while ready and taken < round_size:
                    g = ready.pop(0)
                    take.append(g)
                    taken += g["n"]
                ready_samples -= taken
                yield self._emit(take)
            if not self._groups and ready:  # tail (partial final batch)
                yield self._emit(ready)
                ready = []
                ready_samples = 0

    def _stream_batches_async(self, round_size: int):
        """Pump-thread variant: decode runs on a side thread/stream while
        the caller (trainer) updates between yields."""
        self._start_pump()
        ready: List[dict] = []
        ready_samples = 0
        done = False
        try:
            while not done or ready:
                if not done:
                    grp = self._ready_q.get()
                    if grp is None:
                        done = True
                        if self._pump_err is not None:
                            raise self._pump_err
                    else:
                        ready.append(grp)
                        ready_samples += grp["n"]
                while ready_samples >= round_size and ready:
                    take, taken = [], 0
                    while ready and taken < round_size:
                        g = ready.pop(0)
                        take.append(g)
                        taken += g["n"]
                    ready_samples -= taken
                    yield self._emit(take)
                if done and ready:               # tail
                    yield self._emit(ready)
                    ready = []
                    ready_samples = 0
        finally:
            if self._pump is not None:
                self._pump.join(timeout=600.0)
                self._pump = None

    def _emit(self, groups: List[dict]) -> TensorBatch:
        if not self.shard:
            return self._make_batch(groups)
        r, w = self.shard
        total = sum(g["n"] for g in groups)
        assert total % w == 0, f"round of {total} samples !% {w} shards"
        per = total // w
        start = 0
        for k in range(w):
            take, taken = [], 0
            while start < len(groups) and taken < per:
                take.append(groups[start])
                taken += groups[start]["n"]
                start += 1
            assert taken == per, "groups must tile shards evenly"
            if k == r:
                return self._make_batch(take)
        raise AssertionError("shard index out of range")

    def _make_batch(self, groups: List[dict]) -> TensorBatch:
        prompt_ids = torch.stack([g["prompt_ids"] for g in groups])
        prompt_mask = torch.stack([g["prompt_mask"] for g in groups])
        uids = [g["uid"] for g in groups]
        outputs = [g["outputs"] for g in groups]
        return postprocess_groups(prompt_ids, prompt_mask, uids, outputs,
                                  self.response_length, self.pad_token_id,
                                  self.device,
                                  group_extras=[g.get("extras")
                                                for g in groups])

    # -------------------------------------------------------------- weights
    def update_weights(self, state_dict):
        self.engine.model.load_state_dict(state_dict, strict=False)
        self.engine.flush_radix()
