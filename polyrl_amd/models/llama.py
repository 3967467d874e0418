"""Native dense decoder family (Llama-3 / Qwen2.5) for training and rollout.

One weight set, two execution paths:
  * training forward (this file): PyTorch-ROCm autograd path — SDPA attention,
    used by the FSDP actor/critic workers for fwd/bwd.
  * rollout forward (rollout/engine.py): no-autograd paged path on the
    hand-written CDNA4 HIP kernels, reading the SAME parameters.

Parameter names are HF-compatible (model.layers.N.self_attn.q_proj.weight ...)
so the checkpoint layout interchanges (SURVEY.md §5.4 north star).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from .registry import DecoderConfig


def _norm(mod, x: torch.Tensor) -> torch.Tensor:
    """RMSNorm through the fused trainer kernel on GPU bf16; eager
    reference elsewhere (CPU tier / fp32)."""
    import polyrl_amd.ops as pops
    w = mod.weight
    if x.is_cuda and x.dtype == torch.bfloat16 \
            and w.dtype == torch.bfloat16 \
            and not hasattr(w, "placements") \
            and pops._fused_norm_enabled():
        return pops.rmsnorm_train(x, w, mod.variance_epsilon)
    return mod(x)


def _lin(mod: nn.Module, x: torch.Tensor) -> torch.Tensor:
    """Route trainer linears through the algo-pinned hipBLASLt path
    (ops.tuned_linear); identical math to mod(x).  Wrapped modules (LoRA)
    keep their own forward."""
    if type(mod) is not nn.Linear:
        return mod(x)
    import polyrl_amd.ops as pops
    return pops.tuned_linear(x, mod.weight, mod.bias)


def _pack_align() -> int:
    """GEMM M-alignment for the packed varlen path (0/1 disables)."""
    import os
    try:
        return max(int(os.environ.get("POLYRL_PACK_ALIGN", "256")), 1)
    except ValueError:
        return 256


def _pack_pad_to(model) -> int:
    """Fixed-M padding target for the packed path (0 disables).  Measured
    (profiles/PROFILES.md round 2): hipBLASLt loses 20-45% at mid-size M
    from CU-grid tail quantization — M=8192 fills 256 CUs in exactly 2
    waves (1.6 PF nt) while M~7000 strands a partial wave (0.88-1.27 PF),
    and 256-alignment alone does NOT recover it.  Padding every micro to
    the token budget makes all trunk GEMM shapes fixed AND optimal."""
    import os
    env = os.environ.get("POLYRL_PACK_PAD_TO")
    if env is not None:
        try:
            return max(int(env), 0)
        except ValueError:
            return 0
    return int(getattr(model, "pack_pad_to", 0) or 0)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.variance_epsilon = eps

    def forward(self, x):
        # fused F.rms_norm (fp32 internal); weight applied in model dtype to
        # match the HF/engine convention ((x_norm).to(dt) * w)
        dt = x.dtype
        normed = F.rms_norm(x.float(), (x.shape[-1],), None,
                            self.variance_epsilon)
        return normed.to(dt) * self.weight


class RotaryCache(nn.Module):
    """fp32 cos/sin cache, grown on demand."""

    def __init__(self, head_dim: int, theta: float):
        super().__init__()
        self.head_dim = head_dim
        self.theta = theta
        self.register_buffer("cos_cached", torch.empty(0), persistent=False)
        self.register_buffer("sin_cached", torch.empty(0), persistent=False)

    def get(self, positions: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        need = int(positions.max().item()) + 1 if positions.numel() else 1
        if self.cos_cached.numel() == 0 or self.cos_cached.shape[0] < need or \
                self.cos_cached.device != positions.device:
            n = max(need, 512)
            inv = 1.0 / (self.theta ** (
                torch.arange(0, self.head_dim, 2, device=positions.device,
                             dtype=torch.float32) / self.head_dim))
            t = torch.arange(n, device=positions.device, dtype=torch.float32)
            freqs = torch.outer(t, inv)
            self.cos_cached = freqs.cos()
            self.sin_cached = freqs.sin()
        idx = positions.long()
        return self.cos_cached[idx], self.sin_cached[idx]


class Attention(nn.Module):
    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.num_kv_heads = cfg.num_key_value_heads
        self.head_dim = cfg.head_dim
        h = cfg.hidden_size
        bias = cfg.attention_bias
        self.q_proj = nn.Linear(h, self.num_heads * self.head_dim, bias=bias)
        self.k_proj = nn.Linear(h, self.num_kv_heads * self.head_dim, bias=bias)
        self.v_proj = nn.Linear(h, self.num_kv_heads * self.head_dim, bias=bias)
        self.o_proj = nn.Linear(self.num_heads * self.head_dim, h, bias=False)

    def forward_packed(self, x, cos, sin, cu_seqlens, sp_group=None):
        """Packed varlen path (use_remove_padding): x (total, H); cos/sin
        (total, D/2) per token; causal flash attention over the block-diag
        layout via the hand-written MFMA fwd/bwd kernels (ops.flash_attn_
        varlen — the reference's flash-attn capability).

        With ``sp_group`` (Ulysses SP x packed): x is this
        rank's contiguous 1/sp token shard of the packed stream; the
        all-to-all trades it for a head shard over the FULL packed stream,
        the flash kernel runs on cu_seqlens unchanged with H/sp heads, and
        the reverse all-to-all restores the token shard."""
        import polyrl_amd.ops as pops
        T = x.shape[0]
        q = _lin(self.q_proj, x).view(T, self.num_heads, self.head_dim)
        k = _lin(self.k_proj, x).view(T, self.num_kv_heads, self.head_dim)
        v = _lin(self.v_proj, x).view(T, self.num_kv_heads, self.head_dim)
        if q.is_cuda and q.dtype == torch.bfloat16 \
                and self.head_dim in (64, 128):
            # single-pass fused rotation fwd/bwd (ops/csrc/rope_train.hip)
            cos_f = cos.float().contiguous()
            sin_f = sin.float().contiguous()
            q = pops.rope_train(q, cos_f, sin_f)
            k = pops.rope_train(k, cos_f, sin_f)
        else:
            cos = cos.to(q.dtype).unsqueeze(1)     # (T, 1, D/2)
            sin = sin.to(q.dtype).unsqueeze(1)
            d = self.head_dim // 2
            q = torch.cat([q[..., :d] * cos - q[..., d:] * sin,
                           q[..., d:] * cos + q[..., :d] * sin], dim=-1)
            k = torch.cat([k[..., :d] * cos - k[..., d:] * sin,
                           k[..., d:] * cos + k[..., :d] * sin], dim=-1)
        if sp_group is not None:
            from ..parallel.ulysses import all_to_all_4d
            import torch.distributed as _dist
            sp = _dist.get_world_size(sp_group)
            assert self.num_heads % sp == 0 and self.num_kv_heads % sp == 0, \
                f"heads ({self.num_heads}/{self.num_kv_heads}) % sp {sp} != 0"
            # (1, T/sp, H, D) -> scatter heads, gather tokens -> (1, T, H/sp, D)
            q = all_to_all_4d(q.unsqueeze(0), 2, 1, sp_group).squeeze(0)
            k = all_to_all_4d(k.unsqueeze(0), 2, 1, sp_group).squeeze(0)
            v = all_to_all_4d(v.unsqueeze(0), 2, 1, sp_group).squeeze(0)
        scale = 1.0 / math.sqrt(self.head_dim)
        o = pops.flash_attn_varlen(q, k, v, cu_seqlens[0], scale,
                                   causal=True, tiles=cu_seqlens[1])
        o = o.to(x.dtype)
        if sp_group is not None:
            from ..parallel.ulysses import all_to_all_4d
            # (1, T, H/sp, D) -> scatter tokens, gather heads -> (1, T/sp, H, D)
            o = all_to_all_4d(o.unsqueeze(0), 1, 2, sp_group).squeeze(0)
        return _lin(self.o_proj, o.reshape(x.shape[0], -1))

    def forward(self, x, cos, sin, attn_bias_mask: Optional[torch.Tensor],
                sp_group=None):
        """cos/sin: (B, 1, L, D/2) fp32 over the FULL sequence.  With
        ``sp_group`` (Ulysses SP) x is the rank's 1/sp sequence shard; the
        all-to-all trades it for a head shard over the full sequence
        (parallel/ulysses.py)."""
        B, Ls, _ = x.shape
        q = _lin(self.q_proj, x).view(B, Ls, self.num_heads, self.head_dim)
        k = _lin(self.k_proj, x).view(B, Ls, self.num_kv_heads, self.head_dim)
        v = _lin(self.v_proj, x).view(B, Ls, self.num_kv_heads, self.head_dim)
        if sp_group is not None:
            from ..parallel.ulysses import all_to_all_4d
            import torch.distributed as _dist
            sp = _dist.get_world_size(sp_group)
            assert self.num_heads % sp == 0 and self.num_kv_heads % sp == 0, \
                f"heads ({self.num_heads}/{self.num_kv_heads}) % sp {sp} != 0"
            q = all_to_all_4d(q, 2, 1, sp_group)   # (B, L, Hq/sp, D)
            k = all_to_all_4d(k, 2, 1, sp_group)
            v = all_to_all_4d(v, 2, 1, sp_group)
        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        cos = cos.to(q.dtype)
        sin = sin.to(q.dtype)
        d = self.head_dim // 2
        q = torch.cat([q[..., :d] * cos - q[..., d:] * sin,
                       q[..., d:] * cos + q[..., :d] * sin], dim=-1)
        k = torch.cat([k[..., :d] * cos - k[..., d:] * sin,
                       k[..., d:] * cos + k[..., :d] * sin], dim=-1)
        gqa = k.shape[1] != q.shape[1]
        if attn_bias_mask is not None:
            o = F.scaled_dot_product_attention(q, k, v, attn_mask=attn_bias_mask,
                                               enable_gqa=gqa)
        else:
            o = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                               enable_gqa=gqa)
        o = o.transpose(1, 2)                      # (B, L[, Hq/sp], D)
        if sp_group is not None:
            from ..parallel.ulysses import all_to_all_4d
            o = all_to_all_4d(o, 1, 2, sp_group)   # back to (B, Ls, Hq, D)
        o = o.reshape(B, Ls, -1)
        return self.o_proj(o)


class MLP(nn.Module):
    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        h, i = cfg.hidden_size, cfg.intermediate_size
        self.gate_proj = nn.Linear(h, i, bias=cfg.mlp_bias)
        self.up_proj = nn.Linear(h, i, bias=cfg.mlp_bias)
        self.down_proj = nn.Linear(i, h, bias=cfg.mlp_bias)

    def forward(self, x):
        g = _lin(self.gate_proj, x)
        u = _lin(self.up_proj, x)
        if g.is_cuda and g.dtype == torch.bfloat16 \
                and g.shape[-1] % 8 == 0:
            import polyrl_amd.ops as pops
            return _lin(self.down_proj, pops.silu_mul_train(g, u))
        return _lin(self.down_proj, F.silu(g) * u)


class DecoderLayer(nn.Module):
    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        self.self_attn = Attention(cfg)
        self.mlp = MLP(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)

    def forward(self, x, cos, sin, attn_bias_mask, sp_group=None,
                cu_seqlens=None):
        if cu_seqlens is not None:
            attn_out = self.self_attn.forward_packed(
                _norm(self.input_layernorm, x), cos, sin, cu_seqlens,
                sp_group)
            import polyrl_amd.ops as pops
            _w2 = self.post_attention_layernorm.weight
            if attn_out.is_cuda and attn_out.dtype == torch.bfloat16 \
                    and not hasattr(_w2, "placements") \
                    and pops._fused_norm_enabled():
                # fused residual-add + norm: h = x + attn_out computed
                # inside the norm kernel (one pass; rmsnorm_train.hip)
                m_in, h = pops.fused_add_rmsnorm_train(
                    attn_out, x, self.post_attention_layernorm.weight,
                    self.post_attention_layernorm.variance_epsilon)
            else:
                h = x + attn_out
                m_in = self.post_attention_layernorm(h)
            return h + self.mlp(m_in)
        x = x + self.self_attn(self.input_layernorm(x), cos, sin,
                               attn_bias_mask, sp_group)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class DecoderModel(nn.Module):
    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        self.config = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            DecoderLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.rotary = RotaryCache(cfg.head_dim, cfg.rope_theta)
        self.gradient_checkpointing = False
        self.ulysses = None        # UlyssesContext set by the worker (SP>1)
        self.use_remove_padding = False  # packed varlen path (worker-set)

    def _forward_packed_sp(self, input_ids, attention_mask, position_ids):
        """Ulysses SP x packed varlen (VERDICT r1 weakness #6: SP long-
        context training must not fall back to padded SDPA).  Every SP rank
        gets the SAME full (B, L) inputs; the packed token stream is padded
        to sp*align and split into contiguous per-rank shards.  Everything
        except attention runs on the 1/sp token shard; attention exchanges
        tokens for heads (forward_packed sp path).  Returns the rank's
        (T/sp, H) hidden shard; ``pack_sp_meta`` carries the scatter info
        the worker needs (valid mask, true token count T, padded Tp)."""
        import torch.distributed as _dist
        B, L = input_ids.shape
        sp = self.ulysses.size
        r = self.ulysses.rank
        valid = attention_mask.bool()
        seqlens = valid.sum(-1).int()
        cu = torch.zeros(B + 1, dtype=torch.int32, device=input_ids.device)
        torch.cumsum(seqlens, 0, out=cu[1:])
        ids_p = input_ids[valid]
        pos_p = position_ids[valid]
        T = int(ids_p.shape[0])
        chunk = sp * _pack_align()
        pad_to = _pack_pad_to(self)
        if pad_to and 0.75 * pad_to <= T <= pad_to:
            Tp = -(-pad_to // chunk) * chunk
        else:
            Tp = -(-T // chunk) * chunk
        if Tp != T:
            padn = Tp - T
            ids_p = torch.cat([ids_p, ids_p.new_zeros(padn)])
            pos_p = torch.cat([pos_p, torch.arange(
                padn, device=pos_p.device, dtype=pos_p.dtype)])
            cu = torch.cat([cu, cu.new_full((1,), Tp)])
        shard = Tp // sp
        sl = slice(r * shard, (r + 1) * shard)
        cos, sin = self.rotary.get(pos_p[sl])      # shard positions only
        x = self.embed_tokens(ids_p[sl])
        if input_ids.is_cuda:
            import polyrl_amd.ops as pops
            tiles = pops.build_varlen_tiles(cu.cpu(), input_ids.device)
        else:
            tiles = None
        cu_pack = (cu, tiles)
        grp = self.ulysses.group
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, cos, sin, None, grp, cu_pack,
                    use_reentrant=False)
            else:
                x = layer(x, cos, sin, None, grp, cu_pack)
        x = _norm(self.norm, x)
        self.pack_sp_meta = {"valid": valid, "T": T, "Tp": Tp,
                             "shard": shard}
        return x                                    # (Tp/sp, H)

    def _forward_packed(self, input_ids, attention_mask, position_ids):
        """Remove-padding path (reference: use_remove_padding + flash varlen,
        SURVEY.md §5.7): pack the valid tokens of all rows, run every layer
        on (total, H) with block-diagonal causal flash attention, scatter
        back to (B, L, H).  12-25%% fewer tokens through every GEMM on
        typical left-padded prompt batches, and no SDPA mask fallback."""
        B, L = input_ids.shape
        valid = attention_mask.bool()
        seqlens = valid.sum(-1).int()
        cu = torch.zeros(B + 1, dtype=torch.int32, device=input_ids.device)
        torch.cumsum(seqlens, 0, out=cu[1:])
        ids_p = input_ids[valid]                     # (total,)
        pos_p = position_ids[valid]
        # Pad the packed token count to a multiple of PACK_ALIGN: dynamic
        # token-budget micros produce arbitrary totals (6472, 7216, ...)
        # and hipBLASLt loses ~35-45% on the [M,14336]x[14336,4096] trunk
        # GEMMs at unaligned M (measured: profiles/PROFILES.md round 2).
        # The pad rows form one trailing fake sequence so attention output
        # stays finite (their grads are exactly zero — the scatter below
        # never reads them).
        T = int(ids_p.shape[0])
        align = _pack_align()
        pad_to = _pack_pad_to(self)
        # fixed-M padding only pays off when the micro is nearly full
        # (dense packing leaves one small tail micro — padding THAT to the
        # whole budget would waste more GEMM work than the shape buys)
        if pad_to and 0.75 * pad_to <= T <= pad_to:
            Tp = pad_to
        else:
            Tp = -(-T // align) * align if align > 1 else T
        if Tp != T:
            padn = Tp - T
            ids_p = torch.cat([ids_p, ids_p.new_zeros(padn)])
            pos_p = torch.cat([pos_p, torch.arange(
                padn, device=pos_p.device, dtype=pos_p.dtype)])
            cu = torch.cat([cu, cu.new_full((1,), Tp)])
        cos, sin = self.rotary.get(pos_p)            # (total, D/2)
        x = self.embed_tokens(ids_p)
        # tile tables host-built ONCE per forward, shared by all layers
        if input_ids.is_cuda:
            import polyrl_amd.ops as pops
            tiles = pops.build_varlen_tiles(cu.cpu(), input_ids.device)
        else:
            tiles = None
        cu_pack = (cu, tiles)
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, cos, sin, None, None, cu_pack,
                    use_reentrant=False)
            else:
                x = layer(x, cos, sin, None, None, cu_pack)
        x = _norm(self.norm, x)
        out = torch.zeros(B, L, x.shape[-1], dtype=x.dtype, device=x.device)
        out[valid] = x[:T]
        return out

    def forward(self, input_ids, attention_mask=None, position_ids=None):
        """With Ulysses SP enabled, every SP rank passes the SAME full
        (B, L) inputs (L divisible by sp); the returned hidden states are
        this rank's (B, L/sp, H) sequence shard."""
        B, L = input_ids.shape
        sp_on = self.ulysses is not None and self.ulysses.enabled
        self.pack_sp_meta = None
        if self.use_remove_padding and attention_mask is not None:
            if position_ids is None:
                position_ids = torch.arange(
                    L, device=input_ids.device).expand(B, L)
            if sp_on:
                cfgh = self.config
                if cfgh.num_attention_heads % self.ulysses.size == 0 and \
                        cfgh.num_key_value_heads % self.ulysses.size == 0:
                    return self._forward_packed_sp(input_ids, attention_mask,
                                                   position_ids)
                # head count not divisible: padded SDPA fallback below
            else:
                return self._forward_packed(input_ids, attention_mask,
                                            position_ids)
        if position_ids is None:
            position_ids = torch.arange(L, device=input_ids.device).expand(B, L)
        # per-row rope tables shaped for broadcast: (B, 1, L, D/2)
        cos, sin = self.rotary.get(position_ids.reshape(-1))
        cos = cos.view(B, 1, L, -1)
        sin = sin.view(B, 1, L, -1)

        attn_bias_mask = None
        if attention_mask is not None and not bool(attention_mask.all()):
            # build additive mask: causal + padding
            causal = torch.ones(L, L, dtype=torch.bool,
                                device=input_ids.device).tril()
            pad = attention_mask.bool().view(B, 1, 1, L)
            attn_bias_mask = (causal.view(1, 1, L, L) & pad)

        sp_group = None
        if self.ulysses is not None and self.ulysses.enabled:
            from ..parallel.ulysses import slice_for_rank
            sp_group = self.ulysses.group
            assert L % self.ulysses.size == 0, \
                f"seq len {L} % sp {self.ulysses.size} != 0 (pad upstream)"
            input_ids = slice_for_rank(input_ids, 1, sp_group)

        x = self.embed_tokens(input_ids)
        for layer in self.layers:
            # NOTE: layers must be invoked via __call__ so FSDP2's
            # unshard/reshard pre/post-forward hooks fire.
            if self.gradient_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, cos, sin, attn_bias_mask, sp_group,
                    use_reentrant=False)
            else:
                x = layer(x, cos, sin, attn_bias_mask, sp_group)
        return self.norm(x)


class CausalLM(nn.Module):
    """Actor / reference policy."""

    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        self.config = cfg
        self.model = DecoderModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.apply(self._init_weights)

    def _init_weights(self, m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def gradient_checkpointing_enable(self):
        self.model.gradient_checkpointing = True

    def forward(self, input_ids, attention_mask=None, position_ids=None,
                logits_slice: Optional[slice] = None):
        """logits_slice: sequence-dim slice applied to the hidden states
        BEFORE lm_head — avoids materializing vocab logits over prompt
        positions when only response logprobs are needed.  Ignored under
        Ulysses SP (hidden is a sequence shard; the worker gathers)."""
        hidden = self.model(input_ids, attention_mask, position_ids)
        sp_on = self.model.ulysses is not None and self.model.ulysses.enabled
        if logits_slice is not None and not sp_on:
            hidden = hidden[:, logits_slice]
        return _lin(self.lm_head, hidden.contiguous())

    @torch.no_grad()
    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())


class CausalLMWithValueHead(nn.Module):
    """Critic: decoder trunk + scalar value head (token-level values)."""

    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        self.config = cfg
        self.model = DecoderModel(cfg)
        self.value_head = nn.Linear(cfg.hidden_size, 1, bias=False)
        nn.init.normal_(self.value_head.weight, std=0.02 / math.sqrt(cfg.hidden_size))

    def gradient_checkpointing_enable(self):
        self.model.gradient_checkpointing = True

    def forward(self, input_ids, attention_mask=None, position_ids=None):
        hidden = self.model(input_ids, attention_mask, position_ids)
        return self.value_head(hidden).squeeze(-1)  # (B, L)
