"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, reference PyTorch on CPU.

Policy (deliberate): on a CUDA/HIP device these functions REQUIRE the in-tree
extension ``polyrl_amd._hip`` — a missing extension raises instead of silently
falling back to eager, so a GPU run can never pass on a non-native path.
On CPU they use ops.ref (the same functions the numerics tests compare
against).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import ref

_EXT = None
_EXT_ERR: Optional[str] = None
try:
    from polyrl_amd import _hip as _EXT  # type: ignore
except Exception as e:  # pragma: no cover - import-environment dependent
    _EXT_ERR = repr(e)


def extension_loaded() -> bool:
    return _EXT is not None


def _require_ext():
    if _EXT is None:
        raise RuntimeError(
            "polyrl_amd._hip extension is required on GPU but failed to load "
            f"({_EXT_ERR}). Build it in-tree: "
            "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace")
    return _EXT


# --------------------------------------------------------------------- norms


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if x.is_cuda:
        ext = _require_ext()
        out = torch.empty_like(x)
        ext.rmsnorm(out, x.contiguous(), weight.contiguous(), eps)
        return out
    return ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float = 1e-6
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """returns (normed, new_residual); on GPU, residual is updated IN PLACE
    and returned as new_residual."""
    if x.is_cuda:
        ext = _require_ext()
        out = torch.empty_like(x)
        ext.fused_add_rmsnorm(out, residual, x.contiguous(), weight.contiguous(), eps)
        return out, residual
    return ref.fused_add_rmsnorm(x, residual, weight, eps)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """gate/up may be strided row views into a fused gate_up projection."""
    if gate.is_cuda:
        ext = _require_ext()
        out = torch.empty(gate.shape, dtype=gate.dtype, device=gate.device)
        ext.silu_mul(out, gate, up)
        return out
    return ref.silu_mul(gate, up)


# ---------------------------------------------------------------------- RoPE


class RopeTable:
    """Host-precomputed cos/sin tables (guide App. B: no on-device trig)."""

    def __init__(self, head_dim: int, max_positions: int, theta: float,
                 device, dtype=torch.float32):
        pos = torch.arange(max_positions, device=device)
        cos, sin = ref.rope_cos_sin(pos, head_dim, theta, dtype=torch.float32)
        self.cos = cos.contiguous()
        self.sin = sin.contiguous()
        self.head_dim = head_dim
        self.theta = theta
        self.max_positions = max_positions


def apply_rope_inplace(q: torch.Tensor, k: torch.Tensor,
                       positions: torch.Tensor, table: RopeTable
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """q: (N, Hq, D), k: (N, Hk, D), positions (N,) int32. In place on GPU."""
    if q.is_cuda:
        ext = _require_ext()
        ext.rope_inplace(q, k, positions.int(), table.cos, table.sin)
        return q, k
    cos = table.cos[positions.long()]
    sin = table.sin[positions.long()]
    q2, k2 = ref.apply_rope(q, k, cos, sin)
    q.copy_(q2)
    k.copy_(k2)
    return q, k


# ------------------------------------------------------------------ logprobs


def gather_logprobs(logits: torch.Tensor, labels: torch.Tensor,
                    want_entropy: bool = False):
    """(N, V), (N,) -> lp (N,) fp32 [, entropy (N,) fp32]"""
    if logits.is_cuda:
        ext = _require_ext()
        N = logits.size(0)
        lp = torch.empty(N, dtype=torch.float32, device=logits.device)
        ent = torch.empty(N if want_entropy else 0, dtype=torch.float32,
                          device=logits.device)
        ext.gather_logprobs(lp, ent, logits.contiguous(), labels.long(),
                            want_entropy)
        return (lp, ent) if want_entropy else lp
    if want_entropy:
        return ref.gather_logprobs_entropy(logits, labels)
    return ref.gather_logprobs(logits, labels)


# ------------------------------------------------------------------ sampling


def sample(logits: torch.Tensor, temperature: torch.Tensor,
           top_k: torch.Tensor, top_p: torch.Tensor, seed: int,
           generator: Optional[torch.Generator] = None,
           no_filter: Optional[bool] = None,
           seed_dev: Optional[torch.Tensor] = None,
           out: Optional[Tuple[torch.Tensor, torch.Tensor]] = None
           ) -> Tuple[torch.Tensor, torch.Tensor]:
    """(N, V) -> (tokens (N,) int64, logprobs (N,) fp32 under raw softmax).

    ``no_filter=True`` asserts every row has top_k<=0 and top_p>=1 (the
    common RL rollout setting) and takes the fused single-pass kernel; pass
    it from the caller to avoid a device sync here.  ``seed_dev`` ((1,)
    int64 device tensor) overrides ``seed`` so the call is hipGraph-
    capturable with fresh randomness per replay; ``out`` supplies static
    output buffers for the same reason."""
    if logits.is_cuda:
        ext = _require_ext()
        N = logits.size(0)
        if out is not None:
            tokens, lps = out
        else:
            tokens = torch.empty(N, dtype=torch.int64, device=logits.device)
            lps = torch.empty(N, dtype=torch.float32, device=logits.device)
        if no_filter is None:
            no_filter = bool(((top_k <= 0) | (top_k >= logits.size(1))).all()
                             and (top_p >= 1.0).all())
        ext.sample(tokens, lps, logits.contiguous(), temperature.float(),
                   top_k.int(), top_p.float(), seed, no_filter,
                   seed_dev if seed_dev is not None else torch.Tensor())
        return tokens, lps
    return ref.top_k_top_p_sample(logits, temperature, top_k, top_p, generator)


# ------------------------------------------------------------------ KV cache


def kv_cache_append(k_cache: torch.Tensor, v_cache: torch.Tensor,
                    k: torch.Tensor, v: torch.Tensor,
                    slot_mapping: torch.Tensor):
    """Scatter new token K/V rows (N, Hk, D) into page slots."""
    if k_cache.is_cuda:
        ext = _require_ext()
        ext.kv_cache_append(k_cache, v_cache, k, v, slot_mapping.int())
        return
    page_size = k_cache.shape[1]
    flat_k = k_cache.view(-1, *k_cache.shape[2:])
    flat_v = v_cache.view(-1, *v_cache.shape[2:])
    valid = slot_mapping >= 0
    flat_k[slot_mapping[valid].long()] = k[valid]
    flat_v[slot_mapping[valid].long()] = v[valid]


# ----------------------------------------------------------------- attention


def paged_attention_decode(q: torch.Tensor, k_cache: torch.Tensor,
                           v_cache: torch.Tensor, page_table: torch.Tensor,
                           context_lens: torch.Tensor, scale: float
                           ) -> torch.Tensor:
    if q.is_cuda:
        ext = _require_ext()
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        ext.paged_attention_decode(out, q, k_cache, v_cache,
                                   page_table.int(), context_lens.int(), scale)
        return out
    return ref.paged_attention_decode(q, k_cache, v_cache, page_table,
                                      context_lens, scale)


def _build_prefill_tiles(cu_seqlens_q: torch.Tensor, qtile: int = 64):
    """tile tables (tile_seq, tile_q0) for the prefill kernel grid."""
    cu = cu_seqlens_q.cpu()
    tile_seq, tile_q0 = [], []
    for s in range(cu.numel() - 1):
        L = int(cu[s + 1] - cu[s])
        for q0 in range(0, L, qtile):
            tile_seq.append(s)
            tile_q0.append(q0)
    return (torch.tensor(tile_seq, dtype=torch.int32),
            torch.tensor(tile_q0, dtype=torch.int32))


def varlen_prefill_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                             cu_seqlens_q: torch.Tensor,
                             cu_seqlens_k: torch.Tensor, scale: float,
                             causal: bool = True, return_lse: bool = False):
    if q.is_cuda:
        ext = _require_ext()
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        tile_seq, tile_q0 = _build_prefill_tiles(cu_seqlens_q)
        lse = (torch.empty(q.shape[0], q.shape[1], dtype=torch.float32,
                           device=q.device) if return_lse else torch.Tensor())
        ext.varlen_prefill_attention(
            out, q, k, v, cu_seqlens_q.int(),
            cu_seqlens_k.int(), tile_seq.to(q.device), tile_q0.to(q.device),
            scale, causal, lse)
        return (out, lse) if return_lse else out
    out = ref.varlen_prefill_attention(q, k, v, cu_seqlens_q, cu_seqlens_k,
                                       scale, causal)
    if return_lse:
        return out, ref.varlen_lse(q, k, cu_seqlens_q, cu_seqlens_k, scale,
                                   causal)
    return out


class _FlashAttnVarlen(torch.autograd.Function):
    """Training-path varlen causal attention: MFMA forward (with LSE) and
    hand-written CDNA4 backward (attention_backward.hip) -- the flash-attn
    capability of the reference training stack."""

    @staticmethod
    def forward(ctx, q, k, v, cu_seqlens, scale, causal,
                t64_seq, t64_q0, t32_seq, t32_k0):
        ext = _require_ext()
        qc, kc, vc = q.contiguous(), k.contiguous(), v.contiguous()
        out = torch.empty(qc.shape, dtype=qc.dtype, device=qc.device)
        lse = torch.empty(qc.shape[0], qc.shape[1], dtype=torch.float32,
                          device=qc.device)
        ext.varlen_prefill_attention(out, qc, kc, vc, cu_seqlens, cu_seqlens,
                                     t64_seq, t64_q0, scale, causal, lse)
        ctx.save_for_backward(qc, kc, vc, out, lse, cu_seqlens,
                              t32_seq, t32_k0, t64_seq, t64_q0)
        ctx.scale = scale
        ctx.causal = causal
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse, cu, t32_seq, t32_k0, t64_seq, t64_q0 = \
            ctx.saved_tensors
        ext = _require_ext()
        import os as _os
        ver = _os.environ.get("POLYRL_ATTN_BWD", "v3")
        if ver == "v3":
            # atomic-free dkv+dq split: every output row written exactly
            # once, so the grad buffers need no zero-fill
            dq = torch.empty(q.shape, dtype=torch.float32, device=q.device)
            dk = torch.empty(k.shape, dtype=torch.float32, device=q.device)
            dv = torch.empty(v.shape, dtype=torch.float32, device=q.device)
            ext.varlen_attention_backward_v3(
                dq, dk, dv, q, k, v, out, dout.contiguous(), lse,
                cu, cu, t64_seq, t64_q0, ctx.scale, ctx.causal)
            return (dq.to(q.dtype), dk.to(q.dtype), dv.to(q.dtype),
                    None, None, None, None, None, None, None)
        dq = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        dk = torch.zeros(k.shape, dtype=torch.float32, device=q.device)
        dv = torch.zeros(v.shape, dtype=torch.float32, device=q.device)
        use_v2 = ver == "v2" or \
            _os.environ.get("POLYRL_ATTN_BWD_V2", "0") == "1"
        ext.varlen_attention_backward(
            dq, dk, dv, q, k, v, out, dout.contiguous(), lse,
            cu, cu, t32_seq, t32_k0, ctx.scale, ctx.causal, use_v2)
        return (dq.to(q.dtype), dk.to(q.dtype), dv.to(q.dtype),
                None, None, None, None, None, None, None)


class _RMSNormTrain(torch.autograd.Function):
    """Trainer RMSNorm with optional fused residual add — hand-written
    CDNA4 fwd+bwd (rmsnorm_train.hip).  Returns (y, h) where h = x + res
    (the next residual); grads flow through both outputs."""

    @staticmethod
    def forward(ctx, x, res, w, eps):
        ext = _require_ext()
        res_t = res if res is not None else torch.Tensor()
        y, h, rstd = ext.rmsnorm_train_fwd(x.contiguous(), res_t, w, eps)
        ctx.save_for_backward(h, w, rstd)
        ctx.has_res = res is not None
        return y, h

    @staticmethod
    def backward(ctx, dy, dh_out):
        h, w, rstd = ctx.saved_tensors
        ext = _require_ext()
        dx, dw = ext.rmsnorm_train_bwd(dy, h, w, rstd)
        if dh_out is not None:
            dx = dx + dh_out
        return (dx, dx if ctx.has_res else None, dw.to(w.dtype), None)


def _fused_norm_enabled() -> bool:
    import os
    return os.environ.get("POLYRL_FUSED_NORM", "1") == "1"


def rmsnorm_train(x: torch.Tensor, w: torch.Tensor, eps: float
                  ) -> torch.Tensor:
    """Differentiable RMSNorm (trainer path)."""
    y, _ = _RMSNormTrain.apply(x, None, w, eps)
    return y


def fused_add_rmsnorm_train(x: torch.Tensor, res: torch.Tensor,
                            w: torch.Tensor, eps: float):
    """(norm(x+res), x+res) — one kernel for the residual add + norm."""
    return _RMSNormTrain.apply(x, res, w, eps)


class _SiluMulTrain(torch.autograd.Function):
    """Differentiable silu(gate)*up in one kernel each way."""

    @staticmethod
    def forward(ctx, gate, up):
        ext = _require_ext()
        shp = gate.shape
        g = gate.reshape(-1, shp[-1]).contiguous()
        u = up.reshape(-1, shp[-1]).contiguous()
        ctx.save_for_backward(g, u)
        ctx.shp = shp
        out = torch.empty_like(g)
        ext.silu_mul(out, g, u)
        return out.view(shp)

    @staticmethod
    def backward(ctx, dy):
        g, u = ctx.saved_tensors
        ext = _require_ext()
        dgate = torch.empty_like(g)
        dup = torch.empty_like(u)
        ext.silu_mul_bwd(dgate, dup,
                         dy.reshape(-1, dy.shape[-1]).contiguous(), g, u)
        return dgate.view(ctx.shp), dup.view(ctx.shp)


def silu_mul_train(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return _SiluMulTrain.apply(gate, up)


class _GatherLogprobsTrain(torch.autograd.Function):
    """Differentiable per-token logprob gather over the vocab: fwd saves
    only (bf16 logits, labels, lse) instead of autograd's fp32 log_softmax
    output; bwd is one streaming pass (logprobs.hip)."""

    @staticmethod
    def forward(ctx, logits, labels):
        ext = _require_ext()
        N = logits.size(0)
        lp = torch.empty(N, dtype=torch.float32, device=logits.device)
        lse = torch.empty(N, dtype=torch.float32, device=logits.device)
        ext.gather_logprobs_train_fwd(lp, lse, logits, labels)
        ctx.save_for_backward(logits, labels, lse)
        return lp

    @staticmethod
    def backward(ctx, dlp):
        logits, labels, lse = ctx.saved_tensors
        ext = _require_ext()
        return ext.gather_logprobs_train_bwd(logits, labels, lse,
                                             dlp.float()), None


def gather_logprobs_train(logits: torch.Tensor, labels: torch.Tensor
                          ) -> torch.Tensor:
    """(N, V) bf16 contiguous + (N,) int64 -> (N,) fp32, differentiable."""
    return _GatherLogprobsTrain.apply(logits.contiguous(), labels.long())


class _RopeTrain(torch.autograd.Function):
    """Trainer NEOX RoPE in one kernel pass each way (rope_train.hip)."""

    @staticmethod
    def forward(ctx, x, cos_t, sin_t):
        ext = _require_ext()
        ctx.save_for_backward(cos_t, sin_t)
        return ext.rope_train_apply(x.contiguous(), cos_t, sin_t, False)

    @staticmethod
    def backward(ctx, dy):
        cos_t, sin_t = ctx.saved_tensors
        ext = _require_ext()
        return (ext.rope_train_apply(dy.contiguous(), cos_t, sin_t, True),
                None, None)


def rope_train(x: torch.Tensor, cos_t: torch.Tensor, sin_t: torch.Tensor
               ) -> torch.Tensor:
    """(T, H, D) bf16 with (T, D/2) fp32 tables — differentiable."""
    return _RopeTrain.apply(x, cos_t, sin_t)


class _TunedLinear(torch.autograd.Function):
    """nn.Linear matmuls through the per-shape hipBLASLt algo search
    (gemm_tuned.cpp): fwd nt, dgrad nn, wgrad tn — the trainer's three hot
    layouts, each pinned to its measured-fastest algorithm in-process."""

    @staticmethod
    def forward(ctx, x, w):
        ext = _require_ext()
        ctx.save_for_backward(x, w)
        return ext.tuned_linear_fwd(x, w)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        ext = _require_ext()
        dy = dy.contiguous()
        dx = ext.tuned_linear_dgrad(dy, w) if ctx.needs_input_grad[0] else None
        dw = ext.tuned_linear_wgrad(dy, x) if ctx.needs_input_grad[1] else None
        return dx, dw


def _tuned_gemm_enabled() -> bool:
    import os
    return os.environ.get("POLYRL_TUNED_GEMM", "1") == "1"


def tuned_mm_nt(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Inference-path y = x @ w.T with per-shape algo pinning (no autograd).
    Used by the engine's decode projections: skinny-M GEMMs measured up to
    7x off the weight-read bound on the heuristic pick
    (profiles/PROFILES.md r02 kernel stats)."""
    ext = _require_ext()
    return ext.tuned_linear_fwd(x, w)


def tuned_linear(x: torch.Tensor, w: torch.Tensor,
                 b: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear with per-shape algo pinning on GPU bf16 (training trunk
    shapes); falls through to F.linear elsewhere (CPU tier, tiny dims,
    non-bf16, DTensor-wrapped FSDP params)."""
    if (x.is_cuda and x.dtype == torch.bfloat16
            and w.dtype == torch.bfloat16 and w.shape[0] >= 64
            and w.shape[1] >= 64 and _tuned_gemm_enabled()
            and not hasattr(w, "placements")
            and not hasattr(x, "placements")):
        shp = x.shape
        y = _TunedLinear.apply(x.reshape(-1, shp[-1]).contiguous(), w)
        y = y.view(*shp[:-1], w.shape[0])
        return y if b is None else y + b
    return torch.nn.functional.linear(x, w, b)


def build_varlen_tiles(cu_seqlens_cpu: torch.Tensor, device):
    """Precompute the (64-row forward, 32-key backward) tile tables once per
    packed batch; reused by every layer's flash_attn_varlen call."""
    t64 = _build_prefill_tiles(cu_seqlens_cpu, qtile=64)
    t32 = _build_prefill_tiles(cu_seqlens_cpu, qtile=32)
    return (t64[0].to(device), t64[1].to(device),
            t32[0].to(device), t32[1].to(device))


def flash_attn_varlen(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      cu_seqlens: torch.Tensor, scale: float,
                      causal: bool = True, tiles=None) -> torch.Tensor:
    """Differentiable varlen causal attention over packed (total, H, D).
    GPU: custom MFMA fwd/bwd kernels (tiles from build_varlen_tiles, else
    derived here with a device sync); CPU: the plain-torch reference (its
    autograd supplies the backward)."""
    if q.is_cuda:
        if tiles is None:
            tiles = build_varlen_tiles(cu_seqlens.cpu(), q.device)
        return _FlashAttnVarlen.apply(q, k, v, cu_seqlens.int(), scale,
                                      causal, *tiles)
    return ref.varlen_prefill_attention(q, k, v, cu_seqlens, cu_seqlens,
                                        scale, causal).to(q.dtype)
