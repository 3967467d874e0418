"""In-process continuous-batching decoder for MI355X.

Replaces the reference's external inference engine (SGLang
behind HTTP).  Token-in / token-out.  The model forward here is a separate
no-autograd path over the hand-written HIP kernels (ops/): fused-add-RMSNorm,
table-RoPE, paged decode attention, MFMA varlen prefill attention, SiLU-mul,
fused sampling with logprob capture; projections on hipBLASLt via torch.matmul
against fused QKV / gate-up weight buffers.

Engine weights live in their own contiguous per-layer buffers, decoupled from
the trainer's FSDP parameters, so weight-version updates are plain copies
(transfer/weight_transfer.py) gated against in-flight generation.
"""
from __future__ import annotations

import math
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

import polyrl_amd.ops as ops
from ..models.registry import DecoderConfig
from .kv_cache import PagedKVCache, RadixCache


def _mm_nt(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """y = x @ w.T; decode-sized GEMMs (M <= 512) go through the
    algo-pinned hipBLASLt path (ops.tuned_mm_nt) — the heuristic pick runs
    skinny decode projections far off the weight-read bound.  Prefill Ms
    are ragged (chunk sums), so they keep the torch path to bound the
    number of shape searches."""
    if x.is_cuda and x.dtype == torch.bfloat16 and x.shape[0] <= 512 \
            and _tuned_decode_enabled():
        return ops.tuned_mm_nt(x, w)
    return x @ w.t()


# graphed-decode batch buckets (POLYRL_DECODE_BUCKETS=1): capture counts
# chosen so padding waste stays <= 50% while EOS-staggered batches reuse
# at most ~16 captures over the whole 1..256 running-batch range
DECODE_BUCKETS = (1, 2, 3, 4, 6, 8, 12, 16, 24, 32, 48, 64, 96, 128,
                  192, 256)


def decode_bucket(b: int) -> int:
    """Smallest bucket >= b (b beyond the table maps to itself)."""
    for x in DECODE_BUCKETS:
        if x >= b:
            return x
    return b


def _tuned_decode_enabled() -> bool:
    import os
    return os.environ.get("POLYRL_TUNED_DECODE", "1") == "1"


@dataclass
class SamplingParams:
    temperature: float = 1.0
    top_k: int = -1
    top_p: float = 1.0
    max_new_tokens: int = 128
    stop_token_ids: tuple = ()


@dataclass
class Request:
    rid: str
    input_ids: List[int]
    sampling: SamplingParams
    # runtime state
    output_ids: List[int] = field(default_factory=list)
    output_logprobs: List[float] = field(default_factory=list)
    prefill_pos: int = 0          # tokens already prefilled
    finished: bool = False
    finish_reason: str = ""       # stop | length | abort
    arrival_t: float = 0.0
    # prefix sharing: a group parent prefills once for n samples; children
    # are forked at first-token sampling (radix-cache capability)
    group_children: Optional[List[str]] = None

    @property
    def seq_len(self) -> int:
        return len(self.input_ids) + len(self.output_ids)


@dataclass
class RequestOutput:
    rid: str
    output_ids: List[int]
    output_logprobs: List[float]
    finish_reason: str
    # multi-turn: 1 = assistant token (carries loss), 0 = user/tool token;
    # None = single-turn (all tokens are assistant)
    loss_mask: Optional[List[int]] = None


class LayerWeights:
    """Fused per-layer weight buffers (views kept for named updates).
    With tp > 1 the buffers hold this rank's TP shard (rows of qkv/gate/up,
    cols of o/down — parallel/tp.py docstring)."""

    def __init__(self, cfg: DecoderConfig, device, dtype, tp: int = 1):
        h = cfg.hidden_size
        assert cfg.num_attention_heads % tp == 0 and \
            cfg.num_key_value_heads % tp == 0, \
            f"heads must divide tp={tp}"
        hq = cfg.num_attention_heads * cfg.head_dim // tp
        hk = cfg.num_key_value_heads * cfg.head_dim // tp
        i = cfg.intermediate_size // tp
        self.wqkv = torch.empty(hq + 2 * hk, h, device=device, dtype=dtype)
        self.bqkv = (torch.zeros(hq + 2 * hk, device=device, dtype=dtype)
                     if cfg.attention_bias else None)
        self.wo = torch.empty(h, hq, device=device, dtype=dtype)
        self.w_gate_up = torch.empty(2 * i, h, device=device, dtype=dtype)
        self.w_down = torch.empty(h, i, device=device, dtype=dtype)
        self.tp = tp
        self.input_ln = torch.empty(h, device=device, dtype=dtype)
        self.post_ln = torch.empty(h, device=device, dtype=dtype)
        self._hq, self._hk, self._i = hq, hk, i

    def named_slices(self, prefix: str) -> Dict[str, torch.Tensor]:
        hq, hk, i = self._hq, self._hk, self._i
        out = {
            f"{prefix}.self_attn.q_proj.weight": self.wqkv[:hq],
            f"{prefix}.self_attn.k_proj.weight": self.wqkv[hq:hq + hk],
            f"{prefix}.self_attn.v_proj.weight": self.wqkv[hq + hk:],
            f"{prefix}.self_attn.o_proj.weight": self.wo,
            f"{prefix}.mlp.gate_proj.weight": self.w_gate_up[:i],
            f"{prefix}.mlp.up_proj.weight": self.w_gate_up[i:],
            f"{prefix}.mlp.down_proj.weight": self.w_down,
            f"{prefix}.input_layernorm.weight": self.input_ln,
            f"{prefix}.post_attention_layernorm.weight": self.post_ln,
        }
        if self.bqkv is not None:
            out[f"{prefix}.self_attn.q_proj.bias"] = self.bqkv[:hq]
            out[f"{prefix}.self_attn.k_proj.bias"] = self.bqkv[hq:hq + hk]
            out[f"{prefix}.self_attn.v_proj.bias"] = self.bqkv[hq + hk:]
        return out


class Gpt2LayerWeights:
    """GPT-2 layer buffers (CPU plumbing tier, BASELINE config #1)."""

    def __init__(self, cfg: DecoderConfig, device, dtype):
        h, i = cfg.hidden_size, cfg.intermediate_size
        z = lambda *s: torch.empty(*s, device=device, dtype=dtype)
        self.ln1_w, self.ln1_b = z(h), z(h)
        self.ln2_w, self.ln2_b = z(h), z(h)
        self.wqkv, self.bqkv = z(3 * h, h), z(3 * h)
        self.wo, self.bo = z(h, h), z(h)
        self.w_fc, self.b_fc = z(i, h), z(i)
        self.w_proj, self.b_proj = z(h, i), z(h)

    def named_slices(self, prefix: str):
        return {
            f"{prefix}.ln_1.weight": self.ln1_w, f"{prefix}.ln_1.bias": self.ln1_b,
            f"{prefix}.ln_2.weight": self.ln2_w, f"{prefix}.ln_2.bias": self.ln2_b,
            f"{prefix}.attn_qkv.weight": self.wqkv, f"{prefix}.attn_qkv.bias": self.bqkv,
            f"{prefix}.attn_out.weight": self.wo, f"{prefix}.attn_out.bias": self.bo,
            f"{prefix}.mlp_fc.weight": self.w_fc, f"{prefix}.mlp_fc.bias": self.b_fc,
            f"{prefix}.mlp_proj.weight": self.w_proj, f"{prefix}.mlp_proj.bias": self.b_proj,
        }


class InferenceModel:
    """No-autograd decoder forward over paged KV on the HIP kernel suite.

    ``tp_ctx`` (parallel/tp.py) enables the tensor-parallel mode: buffers
    hold this rank's shard, o/down projections all-reduce, lm_head is
    vocab-parallel with an all-gather before sampling."""

    def __init__(self, cfg: DecoderConfig, device="cuda",
                 dtype=torch.bfloat16, tp_ctx=None):
        from ..parallel.tp import TPContext
        self.tp = tp_ctx if tp_ctx is not None else TPContext(None)
        tp = self.tp.size
        if cfg.arch == "gpt2":
            assert tp == 1, "gpt2 engine is the CPU plumbing tier (tp=1)"
            self._init_gpt2(cfg, device, dtype)
            return
        assert cfg.arch in ("llama", "qwen2"), \
            "rollout engine serves the llama/qwen2/gpt2 families"
        assert cfg.vocab_size % tp == 0, "vocab must divide tp"
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        self.layers = [LayerWeights(cfg, device, dtype, tp)
                       for _ in range(cfg.num_hidden_layers)]
        h = cfg.hidden_size
        self.embed = torch.empty(cfg.vocab_size, h, device=device, dtype=dtype)
        self.final_norm = torch.empty(h, device=device, dtype=dtype)
        self.lm_head = torch.empty(cfg.vocab_size // tp, h, device=device,
                                   dtype=dtype)
        self.rope = ops.RopeTable(cfg.head_dim, cfg.max_position_embeddings,
                                  cfg.rope_theta, device)
        # name -> engine buffer view, for weight updates
        self._name_map: Dict[str, torch.Tensor] = {
            "model.embed_tokens.weight": self.embed,
            "model.norm.weight": self.final_norm,
            "lm_head.weight": self.lm_head,
        }
        for li, lw in enumerate(self.layers):
            self._name_map.update(lw.named_slices(f"model.layers.{li}"))

    def _shard_incoming(self, name: str, src: torch.Tensor,
                        buf: torch.Tensor) -> torch.Tensor:
        """Slice a FULL tensor down to this rank's shard when the buffer is
        TP-sharded (receiver-side resharding, patches.py:196-241 contract)."""
        if self.tp.size == 1 or src.shape == buf.shape:
            return src
        from ..transfer.collective import tp_slice
        if name == "lm_head.weight":
            step = buf.shape[0]
            return src[self.tp.rank * step:(self.tp.rank + 1) * step]
        return tp_slice(name, src, self.tp.rank, self.tp.size, self.cfg.arch)

    def _init_gpt2(self, cfg: DecoderConfig, device, dtype):
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        h = cfg.hidden_size
        self.layers = [Gpt2LayerWeights(cfg, device, dtype)
                       for _ in range(cfg.num_hidden_layers)]
        self.embed = torch.empty(cfg.vocab_size, h, device=device, dtype=dtype)
        self.wpe = torch.empty(cfg.max_position_embeddings, h, device=device,
                               dtype=dtype)
        self.lnf_w = torch.empty(h, device=device, dtype=dtype)
        self.lnf_b = torch.empty(h, device=device, dtype=dtype)
        self.lm_head = torch.empty(cfg.vocab_size, h, device=device, dtype=dtype)
        self._name_map = {
            "wte.weight": self.embed,
            "wpe.weight": self.wpe,
            "ln_f.weight": self.lnf_w,
            "ln_f.bias": self.lnf_b,
            "lm_head.weight": self.lm_head,
        }
        for li, lw in enumerate(self.layers):
            self._name_map.update(lw.named_slices(f"h.{li}"))

    def weight_bytes(self) -> int:
        return sum(v.numel() * v.element_size() for v in self._name_map.values())

    @torch.no_grad()
    def load_state_dict(self, sd: Dict[str, torch.Tensor], strict: bool = True):
        seen = set()
        tied = self.cfg.tie_word_embeddings
        for name, buf in self._name_map.items():
            src = sd.get(name)
            if src is None and name == "lm_head.weight" and tied:
                src = sd.get("model.embed_tokens.weight")
            if src is None:
                if strict:
                    raise KeyError(f"missing weight {name}")
                continue
            src = self._shard_incoming(name, src, buf)
            buf.copy_(src.to(device=buf.device, dtype=buf.dtype,
                             non_blocking=True))
            seen.add(name)
        if strict and len(seen) != len(self._name_map):
            missing = set(self._name_map) - seen
            raise KeyError(f"missing weights: {sorted(missing)[:5]} ...")

    @torch.no_grad()
    def update_named(self, name: str, tensor: torch.Tensor) -> bool:
        buf = self._name_map.get(name)
        if buf is None:
            return False
        tensor = self._shard_incoming(name, tensor, buf)
        buf.copy_(tensor.to(device=buf.device, dtype=buf.dtype,
                            non_blocking=True))
        return True

    # ------------------------------------------------------------- forward
    @torch.no_grad()
    def forward_tokens(self, token_ids: torch.Tensor, positions: torch.Tensor,
                       kv: PagedKVCache, slot_mapping: torch.Tensor,
                       attn_fn) -> torch.Tensor:
        """Shared trunk: embed -> L x (norm, qkv, rope, kv-append, attn(fn),
        o-proj, norm, mlp) -> final norm.  Returns hidden (N, H)."""
        if self.cfg.arch == "gpt2":
            return self._forward_tokens_gpt2(token_ids, positions, kv,
                                             slot_mapping, attn_fn)
        cfg = self.cfg
        tp = self.tp.size
        Hq = cfg.num_attention_heads // tp
        Hk = cfg.num_key_value_heads // tp
        D = cfg.head_dim
        x = self.embed[token_ids]                     # (N, H) gather
        residual = x.clone()
        hidden = None
        for li, lw in enumerate(self.layers):
            if li == 0:
                hidden = ops.rmsnorm(residual, lw.input_ln, cfg.rms_norm_eps)
            else:
                hidden, residual = ops.fused_add_rmsnorm(
                    hidden, residual, lw.input_ln, cfg.rms_norm_eps)
            qkv = _mm_nt(hidden, lw.wqkv)
            if lw.bqkv is not None:
                qkv = qkv + lw.bqkv
            # strided head views into the fused qkv row (no copies: the
            # rope/kv-append/attention kernels take a free token stride)
            q = qkv.narrow(1, 0, Hq * D).unflatten(1, (Hq, D))
            k = qkv.narrow(1, Hq * D, Hk * D).unflatten(1, (Hk, D))
            v = qkv.narrow(1, (Hq + Hk) * D, Hk * D).unflatten(1, (Hk, D))
            ops.apply_rope_inplace(q, k, positions, self.rope)
            ops.kv_cache_append(kv.k_cache[li], kv.v_cache[li], k, v,
                                slot_mapping)
            attn_out = attn_fn(li, q, k, v)           # (N, Hq, D)
            hidden = _mm_nt(attn_out.reshape(-1, Hq * D), lw.wo)
            self.tp.all_reduce_(hidden)               # col-parallel o_proj
            hidden, residual = ops.fused_add_rmsnorm(
                hidden, residual, lw.post_ln, cfg.rms_norm_eps)
            gate_up = _mm_nt(hidden, lw.w_gate_up)
            i_sz = cfg.intermediate_size // tp
            hidden = _mm_nt(ops.silu_mul(gate_up.narrow(1, 0, i_sz),
                                         gate_up.narrow(1, i_sz, i_sz)),
                            lw.w_down)
            self.tp.all_reduce_(hidden)               # col-parallel down_proj
        # final residual add + norm (fused; residual buffer is dead after)
        normed, _ = ops.fused_add_rmsnorm(hidden, residual, self.final_norm,
                                          cfg.rms_norm_eps)
        return normed

    @torch.no_grad()
    def _forward_tokens_gpt2(self, token_ids, positions, kv, slot_mapping,
                             attn_fn):
        import torch.nn.functional as F
        cfg = self.cfg
        H = cfg.num_attention_heads
        D = cfg.head_dim
        x = self.embed[token_ids] + self.wpe[positions.long()]
        for li, lw in enumerate(self.layers):
            h1 = F.layer_norm(x, (cfg.hidden_size,), lw.ln1_w, lw.ln1_b,
                              cfg.layer_norm_eps)
            qkv = h1 @ lw.wqkv.t() + lw.bqkv
            q, k, v = qkv.split(H * D, dim=-1)
            q = q.view(-1, H, D).contiguous()
            k = k.view(-1, H, D).contiguous()
            v = v.view(-1, H, D).contiguous()
            ops.kv_cache_append(kv.k_cache[li], kv.v_cache[li], k, v,
                                slot_mapping)
            attn = attn_fn(li, q, k, v).view(-1, H * D)
            x = x + attn @ lw.wo.t() + lw.bo
            h2 = F.layer_norm(x, (cfg.hidden_size,), lw.ln2_w, lw.ln2_b,
                              cfg.layer_norm_eps)
            x = x + F.gelu(h2 @ lw.w_fc.t() + lw.b_fc) @ lw.w_proj.t() + lw.b_proj
        return F.layer_norm(x, (cfg.hidden_size,), self.lnf_w, self.lnf_b,
                            cfg.layer_norm_eps)

    @torch.no_grad()
    def logits(self, hidden: torch.Tensor) -> torch.Tensor:
        local = _mm_nt(hidden, self.lm_head)
        return self.tp.all_gather_cat(local, dim=-1)


class Engine:
    """Continuous-batching engine: waiting queue + running batch + step()."""

    def __init__(self, cfg: DecoderConfig, device="cuda",
                 dtype=torch.bfloat16, page_size: int = 16,
                 kv_bytes_budget: Optional[int] = None,
                 max_running_requests: int = 256,
                 max_num_batched_tokens: int = 8192,
                 max_model_len: Optional[int] = None,
                 decode_chunk_size: int = 16,
                 tp_ctx=None,
                 seed: int = 0,
                 enable_radix_cache: bool = False):
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        self.model = InferenceModel(cfg, device, dtype, tp_ctx=tp_ctx)
        self.model._engine_owner = self   # weight publishers flush the radix
        self.tp = self.model.tp
        self.max_running = max_running_requests
        self.max_batched_tokens = max_num_batched_tokens
        self.max_model_len = max_model_len or cfg.max_position_embeddings
        # KV cache holds this rank's local KV heads (gpt2 has no GQA/TP)
        hk_local = (cfg.num_attention_heads if cfg.arch == "gpt2"
                    else cfg.num_key_value_heads // self.tp.size)
        bt = PagedKVCache.bytes_per_token(cfg.num_hidden_layers,
                                          hk_local, cfg.head_dim)
        if kv_bytes_budget is None:
            kv_bytes_budget = 1 << 30  # 1 GiB default (tests); callers size it
        num_pages = max(int(kv_bytes_budget // (bt * page_size)), 8)
        self.kv = PagedKVCache(cfg.num_hidden_layers, hk_local,
                               cfg.head_dim, num_pages, page_size,
                               dtype=dtype, device=device)
        self.waiting: List[Request] = []
        self.running: List[Request] = []
        self._seq_counter = 0
        self._seq_ids: Dict[str, int] = {}
        self._step_counter = 0
        self._seed = seed
        self._abort_all = False
        self._scale = 1.0 / math.sqrt(cfg.head_dim)
        self._gen = torch.Generator().manual_seed(seed)  # CPU ref path RNG
        self.decode_chunk_size = max(decode_chunk_size, 1)
        self.enable_prefix_sharing = True
        # cross-request KV prefix reuse (SGLang radix-cache capability);
        # deterministic trie ops, so TP ranks stay in lockstep
        self.radix = RadixCache(self.kv) if enable_radix_cache else None
        # hipGraph-captured decode iteration (one replay per token):
        # removes ~300 kernel-launch round-trips per decode step.  Disabled
        # under TP (capturing RCCL collectives in a graph is unvalidated).
        self.enable_hip_graphs = (device != "cpu") and self.tp.size == 1 and \
            bool(int(__import__("os").environ.get("POLYRL_HIP_GRAPHS", "1")))
        self._graphs: Dict[int, dict] = {}      # batch size -> capture state
        self._graph_pool = None
        self._graph_max_pages = (self.max_model_len + page_size - 1) // page_size
        # EXPERIMENTAL (round-3 candidate, default OFF): pad the graphed
        # decode batch up to a bucket size so EOS-staggered serving (B
        # shrinking every few steps) replays existing captures instead of
        # capturing a fresh graph per exact B.  Dummy rows write KV into a
        # reserved scratch slot and their outputs are discarded.
        self._decode_buckets = bool(int(__import__("os").environ.get(
            "POLYRL_DECODE_BUCKETS", "0")))

    # ------------------------------------------------------------ public API
    def add_request(self, rid: str, input_ids: List[int],
                    sampling: SamplingParams):
        assert len(input_ids) > 0
        req = Request(rid=rid, input_ids=list(input_ids), sampling=sampling,
                      arrival_t=time.time())
        self.waiting.append(req)
        return rid

    def add_request_group(self, rid_prefix: str, input_ids: List[int],
                          sampling: SamplingParams, n: int):
        """n samples of one prompt with a SHARED prompt prefill + shared
        full KV pages (SURVEY.md §2.4.3 radix/prefix-cache capability): one
        request prefills; at its first sampling step the engine draws n
        first tokens and forks n-1 children whose page tables reference the
        parent's full prompt pages (refcounted; a partial trailing page is
        copied per child so decode writes never touch shared pages).
        Child rids are '<prefix>-s0' .. '<prefix>-s<n-1>'."""
        assert n >= 1
        if n == 1 or not self.enable_prefix_sharing:
            for s_ in range(n):
                self.add_request(f"{rid_prefix}-s{s_}", input_ids, sampling)
            return
        parent = Request(rid=f"{rid_prefix}-s0", input_ids=list(input_ids),
                         sampling=sampling, arrival_t=time.time())
        parent.group_children = [f"{rid_prefix}-s{s_}" for s_ in range(1, n)]
        self.waiting.append(parent)

    def abort_request(self, rid: Optional[str] = None, abort_all: bool = False):
        """Mark requests aborted; they are emitted with partial output on the
        next step (token-level continuation happens at the scheduler)."""
        for r in self.running + self.waiting:
            if abort_all or r.rid == rid:
                r.finished = True
                r.finish_reason = "abort"

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def release_memory(self):
        """Free the KV cache tensors (the reference's
        /release_memory_occupation capability, sglang_http_async_engine.py:
        257-284 — used when trainer and rollout time-share a tight GPU).
        Requires an idle engine; resume_memory() reallocates."""
        assert not self.has_work(), "release_memory needs an idle engine"
        self._kv_shape = (self.kv.num_layers, self.kv.num_kv_heads,
                          self.kv.head_dim, self.kv.num_pages,
                          self.kv.page_size)
        if self.radix is not None:
            self.radix.reset()
        self.kv.k_cache = []
        self.kv.v_cache = []
        self._kv_released = True
        # Captured decode graphs hold device pointers into the freed KV
        # tensors; replaying them after a release/resume cycle would read
        # and write stale memory.  Drop them so they re-capture against the
        # new PagedKVCache (the pool is dropped too — it owns the old
        # allocations).
        self._graphs = {}
        self._graph_pool = None
        if str(self.device).startswith("cuda"):
            torch.cuda.empty_cache()

    def resume_memory(self):
        if not getattr(self, "_kv_released", False):
            return
        L, Hk, D, P, ps = self._kv_shape
        self.kv = PagedKVCache(L, Hk, D, P, ps, dtype=self.dtype,
                               device=self.device)
        if self.radix is not None:
            self.radix = RadixCache(self.kv)
        self._kv_released = False

    def num_queued(self) -> int:
        return len(self.waiting)

    def num_running(self) -> int:
        return len(self.running)

    # ---------------------------------------------------------------- step
    @torch.no_grad()
    def step(self) -> List[RequestOutput]:
        """One engine iteration: emit aborted/finished, admit+prefill (chunked),
        decode one token for the running batch.  Returns finished outputs."""
        finished: List[RequestOutput] = []

        # sweep finished/aborted
        still = []
        for r in self.running:
            if r.finished:
                finished.extend(self._emit_all(r))
            else:
                still.append(r)
        self.running = still
        self.waiting = [r for r in self.waiting if not r.finished or
                        finished.extend(self._emit_all(r))]  # aborted waiters

        # ---- admission + chunked prefill -----------------------------------
        token_budget = self.max_batched_tokens
        prefill_reqs: List[Request] = []
        prefill_lens: List[int] = []
        # continue partially-prefilled first, then admit new
        candidates = [r for r in self.running if r.prefill_pos < len(r.input_ids)]
        admit_budget = token_budget - sum(
            min(len(r.input_ids) - r.prefill_pos, token_budget) for r in candidates)
        while self.waiting and len(self.running) < self.max_running \
                and admit_budget > 0:
            nxt = self.waiting[0]
            radix_pages: List[int] = []
            radix_len = 0
            if self.radix is not None and nxt.prefill_pos == 0:
                radix_pages, radix_len = self.radix.match(nxt.input_ids)
            need = min(len(nxt.input_ids) - nxt.prefill_pos - radix_len,
                       admit_budget)
            if not self._ensure_pages(need):
                if radix_pages:
                    self.kv.unref_pages(radix_pages)
                break
            admit_budget -= need
            self.waiting.pop(0)
            sid = self._seq_counter
            self._seq_counter += 1
            self._seq_ids[nxt.rid] = sid
            if radix_len:
                # cached prompt pages seed the page table; prefill resumes
                # at the page-aligned boundary (chunked-prefill path
                # already attends to the full cached history)
                self.kv.seed_seq(sid, radix_pages, radix_len)
                nxt.prefill_pos = radix_len
            self.running.append(nxt)
            candidates.append(nxt)
        for r in candidates:
            if token_budget <= 0:
                break
            need = len(r.input_ids) - r.prefill_pos
            take = min(need, token_budget)
            if take <= 0:
                continue
            if not self._ensure_pages(take) or \
                    not self.kv.allocate(self._seq_ids[r.rid], take):
                break
            prefill_reqs.append(r)
            prefill_lens.append(take)
            token_budget -= take
        if prefill_reqs:
            self._run_prefill(prefill_reqs, prefill_lens)

        # ---- decode ---------------------------------------------------------
        decode_reqs = [r for r in self.running
                       if r.prefill_pos >= len(r.input_ids) and not r.finished]
        if decode_reqs:
            ok = []
            for r in decode_reqs:
                if self._ensure_pages(1) and \
                        self.kv.allocate(self._seq_ids[r.rid], 1):
                    ok.append(r)
                else:
                    # out of KV pages: abort (scheduler continues elsewhere)
                    r.finished = True
                    r.finish_reason = "abort"
                    finished.extend(self._emit_all(r))
            if ok:
                self._run_decode(ok)

        # sweep newly finished
        still = []
        for r in self.running:
            if r.finished:
                finished.extend(self._emit_all(r))
            else:
                still.append(r)
        self.running = still
        self._step_counter += 1
        return finished

    def generate(self, prompts: List[List[int]], sampling: SamplingParams,
                 rid_prefix: str = "req") -> List[RequestOutput]:
        """Synchronous batch generate (test / co-located convenience)."""
        for i, p in enumerate(prompts):
            self.add_request(f"{rid_prefix}-{i}", p, sampling)
        outs: Dict[str, RequestOutput] = {}
        while self.has_work():
            for o in self.step():
                outs[o.rid] = o
        return [outs[f"{rid_prefix}-{i}"] for i in range(len(prompts))]

    # -------------------------------------------------------------- internals
    def _ensure_pages(self, num_tokens: int) -> bool:
        """Free-list check with radix-cache LRU eviction under pressure."""
        if self.kv.can_allocate(num_tokens):
            return True
        if self.radix is not None:
            need = (num_tokens + self.kv.page_size - 1) // self.kv.page_size \
                - self.kv.free_pages
            self.radix.evict(need)
        return self.kv.can_allocate(num_tokens)

    def flush_radix(self):
        """Drop cached prefixes (stale after a weight update — the
        reference flushes SGLang's cache post-update, patches.py:360-387)."""
        if self.radix is not None:
            self.radix.reset()

    def _emit(self, r: Request) -> RequestOutput:
        sid = self._seq_ids.pop(r.rid, None)
        if sid is not None:
            # Aborted requests (weight-update time-box) must NOT donate KV
            # to the radix cache: they are emitted on the step AFTER
            # flush_radix(), so their pages hold OLD-weight KV and a
            # token-exact continuation of the same request would radix-match
            # that stale prefix and skip recomputing it under new weights.
            if self.radix is not None and r.finish_reason != "abort":
                written = self.kv.seq_len(sid)   # last sampled token has no KV
                toks = (r.input_ids + r.output_ids)[:written]
                self.radix.insert(toks, self.kv._seq_pages.get(sid, []))
            self.kv.free_seq(sid)
        if not r.finish_reason:
            r.finish_reason = "length"
        return RequestOutput(rid=r.rid, output_ids=list(r.output_ids),
                             output_logprobs=list(r.output_logprobs),
                             finish_reason=r.finish_reason)

    def _emit_all(self, r: Request) -> List[RequestOutput]:
        """Emit r; a group parent finishing BEFORE its fork (abort / OOM)
        also emits one aborted output per unforked child rid."""
        outs = [self._emit(r)]
        if r.group_children:
            for crid in r.group_children:
                outs.append(RequestOutput(rid=crid, output_ids=[],
                                          output_logprobs=[],
                                          finish_reason=outs[0].finish_reason))
            r.group_children = None
        return outs

    def _sample_last(self, reqs: List[Request], hidden_rows: torch.Tensor):
        """Sample the next token for each req from its last hidden row."""
        logits = self.model.logits(hidden_rows)
        n = len(reqs)
        temp = torch.tensor([r.sampling.temperature for r in reqs],
                            dtype=torch.float32, device=self.device)
        tk = torch.tensor([r.sampling.top_k for r in reqs],
                          dtype=torch.int32, device=self.device)
        tp = torch.tensor([r.sampling.top_p for r in reqs],
                          dtype=torch.float32, device=self.device)
        seed = (self._seed * 0x9E3779B9 + self._step_counter) & 0x7FFFFFFFFFFF
        no_filter = all(
            (r.sampling.top_k <= 0 or r.sampling.top_k >= logits.shape[-1])
            and r.sampling.top_p >= 1.0 for r in reqs)
        tokens, lps = ops.sample(logits, temp, tk, tp, seed,
                                 generator=self._gen, no_filter=no_filter)
        tokens_l = tokens.tolist()
        lps_l = lps.tolist()
        for i, r in enumerate(reqs):
            t = int(tokens_l[i])
            r.output_ids.append(t)
            r.output_logprobs.append(float(lps_l[i]))
            if t in r.sampling.stop_token_ids:
                r.finished = True
                r.finish_reason = "stop"
            elif len(r.output_ids) >= r.sampling.max_new_tokens:
                r.finished = True
                r.finish_reason = "length"
            elif r.seq_len >= self.max_model_len:
                r.finished = True
                r.finish_reason = "length"

    def _run_prefill(self, reqs: List[Request], lens: List[int]):
        dev = self.device
        tok_list, pos_list, slot_list = [], [], []
        cu_q = [0]
        cu_k = [0]
        for r, take in zip(reqs, lens):
            sid = self._seq_ids[r.rid]
            start = r.prefill_pos
            tok_list.extend(r.input_ids[start:start + take])
            pos_list.extend(range(start, start + take))
            slot_list.append(self.kv.slots_for(sid, start, take))
            cu_q.append(cu_q[-1] + take)
            cu_k.append(cu_k[-1] + start + take)
        tokens = torch.tensor(tok_list, dtype=torch.long, device=dev)
        positions = torch.tensor(pos_list, dtype=torch.int32, device=dev)
        slots = torch.cat(slot_list).to(dev)
        cu_q_t = torch.tensor(cu_q, dtype=torch.int32, device=dev)
        cu_k_t = torch.tensor(cu_k, dtype=torch.int32, device=dev)

        seq_ids = [self._seq_ids[r.rid] for r in reqs]
        page_table = self.kv.page_table(seq_ids).to(dev)
        ctx_lens = torch.tensor([s + t for s, t in
                                 zip((r.prefill_pos for r in reqs), lens)],
                                dtype=torch.int32, device=dev)

        def attn_fn(li, q, k, v):
            # chunked prefill: queries attend to the FULL cached history
            # (cu_k spans include prior chunks) — gather K/V from the cache
            # pages for this layer.
            if all(r.prefill_pos == 0 for r in reqs):
                return ops.varlen_prefill_attention(q, k, v, cu_q_t, cu_k_t,
                                                    self._scale, causal=True)
            kf, vf = self._gather_kv(li, seq_ids, ctx_lens)
            return ops.varlen_prefill_attention(q, kf, vf, cu_q_t, cu_k_t,
                                                self._scale, causal=True)

        hidden = self.model.forward_tokens(tokens, positions, self.kv, slots,
                                           attn_fn)
        # requests whose prefill completes sample their first token;
        # group parents fork their children first (shared prompt pages)
        done_rows, done_reqs = [], []
        fork_list = []
        for i, (r, take) in enumerate(zip(reqs, lens)):
            r.prefill_pos += take
            if r.prefill_pos >= len(r.input_ids):
                if r.group_children:
                    fork_list.append((r, cu_q[i + 1] - 1))
                else:
                    done_rows.append(cu_q[i + 1] - 1)
                    done_reqs.append(r)
        if done_reqs:
            rows = hidden[torch.tensor(done_rows, dtype=torch.long, device=dev)]
            self._sample_last(done_reqs, rows)
        for r, row in fork_list:
            self._fork_group(r, hidden[row])

    def _fork_group(self, parent: Request, last_hidden: torch.Tensor):
        """Fork n-1 children off a freshly prefilled group parent: shared
        full prompt pages (refcounted), per-child copy of the partial
        trailing page, then ONE logits row sampled n times (per-row RNG
        keys give independent draws)."""
        child_rids = parent.group_children
        n = len(child_rids) + 1
        Lp = len(parent.input_ids)
        children = [Request(rid=crid, input_ids=parent.input_ids,
                            sampling=parent.sampling, prefill_pos=Lp,
                            arrival_t=parent.arrival_t)
                    for crid in child_rids]
        child_sids = []
        for c in children:
            sid = self._seq_counter
            self._seq_counter += 1
            self._seq_ids[c.rid] = sid
            child_sids.append(sid)
        copies = self.kv.fork_seq(self._seq_ids[parent.rid], child_sids, Lp)
        if copies is None:
            # out of pages for the partial-page copies: children re-prefill
            # independently (correct, just unshared)
            for c, sid in zip(children, child_sids):
                self._seq_ids.pop(c.rid, None)
                c.prefill_pos = 0
                self.waiting.append(c)
            parent.group_children = None
            self._sample_last([parent], last_hidden.unsqueeze(0))
            return
        if copies:
            src = torch.tensor([a for a, _ in copies], dtype=torch.long,
                               device=self.device)
            dst = torch.tensor([b for _, b in copies], dtype=torch.long,
                               device=self.device)
            rem = Lp % self.kv.page_size
            for li in range(self.cfg.num_hidden_layers):
                self.kv.k_cache[li][dst, :rem] = self.kv.k_cache[li][src, :rem]
                self.kv.v_cache[li][dst, :rem] = self.kv.v_cache[li][src, :rem]
        parent.group_children = None
        rows = last_hidden.unsqueeze(0).expand(n, -1).contiguous()
        self.running.extend(children)
        self._sample_last([parent] + children, rows)

    def _gather_kv(self, li: int, seq_ids: List[int], ctx_lens: torch.Tensor):
        """Materialize contiguous K/V for chunked-prefill history (per layer)."""
        ks, vs = [], []
        flat_k = self.kv.k_cache[li].view(-1, self.cfg.num_key_value_heads,
                                          self.cfg.head_dim)
        flat_v = self.kv.v_cache[li].view(-1, self.cfg.num_key_value_heads,
                                          self.cfg.head_dim)
        for i, sid in enumerate(seq_ids):
            L = int(ctx_lens[i])
            slots = self.kv.slots_for(sid, 0, L).to(flat_k.device)
            ks.append(flat_k[slots.long()])
            vs.append(flat_v[slots.long()])
        return torch.cat(ks), torch.cat(vs)

    def _decode_chunk_len(self, reqs: List[Request]) -> int:
        """How many decode iterations can run device-side without a host
        sync: bounded by the chunk size, every request's remaining token
        budget, and the model-len fence.  Stop-token requests decode 1 at a
        time (token-exact stop)."""
        if any(r.sampling.stop_token_ids for r in reqs):
            return 1
        rem = min(r.sampling.max_new_tokens - len(r.output_ids) for r in reqs)
        rem = min(rem, min(self.max_model_len - r.seq_len for r in reqs))
        return max(1, min(self.decode_chunk_size, rem))

    def _run_decode(self, reqs: List[Request]):
        """C decode iterations with all state device-resident: one H2D
        (initial tokens/positions/slots) and one D2H sync (the chunk's
        sampled tokens + logprobs) per C tokens, instead of per token."""
        dev = self.device
        C = self._decode_chunk_len(reqs)
        B = len(reqs)
        seq_ids = [self._seq_ids[r.rid] for r in reqs]
        p0 = [r.seq_len - 1 for r in reqs]
        if C > 1:
            # slots for positions [p0, p0+C) were allocated by the caller
            # for step 0 only; grow the allocation for the rest now
            ok = []
            for r, sid in zip(reqs, seq_ids):
                if self.kv.allocate(sid, C - 1):
                    ok.append(True)
                else:
                    ok.append(False)
            if not all(ok):
                # roll back to single-token decode for this round — and
                # roll back the successful growths too, or their _seq_len
                # would count slots that never get written (the radix
                # cache would donate garbage KV for those positions)
                for o, sid in zip(ok, seq_ids):
                    if o:
                        self.kv.shrink_seq(sid, C - 1)
                C = 1

        tokens = torch.tensor([r.output_ids[-1] for r in reqs],
                              dtype=torch.long, device=dev)
        pos0 = torch.tensor(p0, dtype=torch.int32, device=dev)
        # (C, B) slot matrix: row s is the (contiguous) slot mapping of step s
        slots_all = torch.stack(
            [self.kv.slots_for(sid, p, C) for sid, p in zip(seq_ids, p0)],
            dim=1).to(dev)
        page_table = self.kv.page_table(seq_ids).to(dev)
        ctx = pos0 + 1

        temp = torch.tensor([r.sampling.temperature for r in reqs],
                            dtype=torch.float32, device=dev)
        tk = torch.tensor([r.sampling.top_k for r in reqs],
                          dtype=torch.int32, device=dev)
        tp = torch.tensor([r.sampling.top_p for r in reqs],
                          dtype=torch.float32, device=dev)
        no_filter = all(
            (r.sampling.top_k <= 0 or r.sampling.top_k >= self.cfg.vocab_size)
            and r.sampling.top_p >= 1.0 for r in reqs)
        out_tokens = torch.empty(C, B, dtype=torch.int64, device=dev)
        out_lps = torch.empty(C, B, dtype=torch.float32, device=dev)

        seed0 = (self._seed * 0x9E3779B9 + self._step_counter * 131) \
            & 0x7FFFFFFFFFFF
        if self.enable_hip_graphs and no_filter and dev != "cpu" \
                and self.cfg.arch != "gpt2":
            try:
                self._decode_graphed(B, C, tokens, pos0, slots_all,
                                     page_table, temp, tk, tp, seed0,
                                     out_tokens, out_lps)
                C_done = True
            except Exception as e:
                import traceback
                print(f"[engine] hipGraph decode capture failed, falling "
                      f"back to eager: {e!r}", flush=True)
                traceback.print_exc()
                self.enable_hip_graphs = False   # fall back permanently
                C_done = False
        else:
            C_done = False
        if not C_done:
            def attn_fn(li, q, k, v):
                return ops.paged_attention_decode(
                    q, self.kv.k_cache[li], self.kv.v_cache[li], page_table,
                    ctx, self._scale)

            for s in range(C):
                positions = pos0 + s
                hidden = self.model.forward_tokens(tokens, positions, self.kv,
                                                   slots_all[s], attn_fn)
                logits = self.model.logits(hidden)
                tokens, lps = ops.sample(logits, temp, tk, tp, seed0 + s,
                                         generator=self._gen,
                                         no_filter=no_filter)
                out_tokens[s] = tokens
                out_lps[s] = lps
                if s + 1 < C:
                    ctx = ctx + 1

        # single sync for the whole chunk
        toks_h = out_tokens.t().cpu().tolist()
        self._finish_decode_chunk(reqs, C, toks_h, out_lps)

    _SCRATCH_SEQ = -7777       # sentinel seq id owning the scratch page

    def _decode_scratch_slot(self) -> int:
        """Reserve (once per KV cache lifetime) one page as the dummy-row
        KV sink for bucketed decode; zero it so dummy-row attention reads
        finite bf16.  Re-reserves automatically after release/resume
        (the KV cache is rebuilt there and seq maps reset)."""
        if self.kv.seq_len(self._SCRATCH_SEQ) == 0:
            if not self.kv.allocate(self._SCRATCH_SEQ, 1):
                raise RuntimeError("no free page for decode-bucket scratch")
            page = int(self.kv.slots_for(self._SCRATCH_SEQ, 0, 1)[0]) \
                // self.kv.page_size
            for li in range(len(self.kv.k_cache)):
                self.kv.k_cache[li][page].zero_()
                self.kv.v_cache[li][page].zero_()
        return int(self.kv.slots_for(self._SCRATCH_SEQ, 0, 1)[0])

    def _decode_graphed(self, B, C, tokens, pos0, slots_all, page_table,
                        temp, tk, tp, seed0, out_tokens, out_lps):
        """hipGraph decode: capture one decode iteration per batch size and
        replay it per token.  All state (tokens/positions/ctx/slots/page
        table/seed) lives in static device buffers; positions, ctx and the
        sampling seed advance INSIDE the graph, the slot column is a tiny
        D2D copy per replay."""
        dev = self.device
        # bucketed capture (EXPERIMENTAL, default off): pad to Bb rows; rows
        # [B:Bb) are dummies writing KV into a reserved scratch slot
        Bb = decode_bucket(B) if self._decode_buckets else B
        scratch = self._decode_scratch_slot() if Bb > B else 0
        st = self._graphs.get(Bb)
        if st is None:
            st = {
                "tokens": torch.zeros(Bb, dtype=torch.int64, device=dev),
                "pos": torch.zeros(Bb, dtype=torch.int32, device=dev),
                "ctx": torch.ones(Bb, dtype=torch.int32, device=dev),
                "slots": torch.full((Bb,), scratch, dtype=torch.int32,
                                    device=dev),
                "ptab": torch.zeros(Bb, self._graph_max_pages,
                                    dtype=torch.int32, device=dev),
                "temp": torch.ones(Bb, dtype=torch.float32, device=dev),
                "tk": torch.full((Bb,), -1, dtype=torch.int32, device=dev),
                "tp": torch.ones(Bb, dtype=torch.float32, device=dev),
                "seed": torch.zeros(1, dtype=torch.int64, device=dev),
                "out_t": torch.zeros(Bb, dtype=torch.int64, device=dev),
                "out_l": torch.zeros(Bb, dtype=torch.float32, device=dev),
            }
            if Bb > B:
                st["ptab"].fill_(scratch // self.kv.page_size)

            def body():
                def attn_fn(li, q, k, v):
                    return ops.paged_attention_decode(
                        q, self.kv.k_cache[li], self.kv.v_cache[li],
                        st["ptab"], st["ctx"], self._scale)
                hidden = self.model.forward_tokens(
                    st["tokens"], st["pos"], self.kv, st["slots"], attn_fn)
                logits = self.model.logits(hidden)
                ops.sample(logits, st["temp"], st["tk"], st["tp"], 0,
                           no_filter=True, seed_dev=st["seed"],
                           out=(st["out_t"], st["out_l"]))
                st["tokens"].copy_(st["out_t"])
                st["pos"] += 1
                st["ctx"] += 1
                st["seed"] += 1

            st["body"] = body
            # warmup on a side stream (kv writes land in slots that the
            # first real replay overwrites with identical data)
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            st["tokens"][:B].copy_(tokens)
            st["pos"][:B].copy_(pos0)
            st["ctx"][:B].copy_(pos0 + 1)
            st["slots"][:B].copy_(slots_all[0])
            mp = min(page_table.shape[1], self._graph_max_pages)
            st["ptab"][:B].zero_()
            st["ptab"][:B, :mp].copy_(page_table[:, :mp])
            with torch.cuda.stream(side):
                for _ in range(2):
                    body()
            torch.cuda.current_stream().wait_stream(side)
            g = torch.cuda.CUDAGraph()
            if self._graph_pool is None:
                with torch.cuda.graph(g):
                    body()
                self._graph_pool = g.pool()
            else:
                with torch.cuda.graph(g, pool=self._graph_pool):
                    body()
            st["graph"] = g
            self._graphs[Bb] = st

        # load chunk state into the static buffers
        st["tokens"][:B].copy_(tokens)
        st["pos"][:B].copy_(pos0)
        st["ctx"][:B].copy_(pos0 + 1)
        mp = min(page_table.shape[1], self._graph_max_pages)
        st["ptab"][:B].zero_()
        st["ptab"][:B, :mp].copy_(page_table[:, :mp])
        st["temp"][:B].copy_(temp)
        st["tk"][:B].copy_(tk)
        st["tp"][:B].copy_(tp)
        st["seed"].fill_(seed0)
        if Bb > B:
            # reset dummy rows (a previous call on this bucket with a
            # larger real B may have left real state here): KV writes go
            # to the scratch slot, attention reads ctx=1 from the zeroed
            # scratch page, outputs are discarded below
            st["pos"][B:].zero_()
            st["ctx"][B:].fill_(1)
            st["slots"][B:].fill_(scratch)
            st["ptab"][B:].fill_(scratch // self.kv.page_size)
            st["tokens"][B:].zero_()
        g = st["graph"]
        for s in range(C):
            st["slots"][:B].copy_(slots_all[s])
            g.replay()
            out_tokens[s].copy_(st["out_t"][:B])
            out_lps[s].copy_(st["out_l"][:B])

    def _finish_decode_chunk(self, reqs, C, toks_h, out_lps):
        lps_h = out_lps.t().cpu().tolist()
        for i, r in enumerate(reqs):
            for s in range(C):
                t = int(toks_h[i][s])
                r.output_ids.append(t)
                r.output_logprobs.append(float(lps_h[i][s]))
                if t in r.sampling.stop_token_ids:
                    r.finished = True
                    r.finish_reason = "stop"
                    break
            if not r.finished:
                if len(r.output_ids) >= r.sampling.max_new_tokens:
                    r.finished = True
                    r.finish_reason = "length"
                elif r.seq_len >= self.max_model_len:
                    r.finished = True
                    r.finish_reason = "length"
