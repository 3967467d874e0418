"""Pure-PyTorch reference implementations of every HIP kernel.

These are the numerics oracles: each HIP kernel in csrc/ is tested against the
fp32 form here (tests/test_ops_gpu.py), and they serve as the CPU execution
path for the no-GPU test tier.  Keep them simple and obviously correct.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float = 1e-6
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """returns (rmsnorm(x + residual), x + residual)"""
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps), new_residual


def rope_cos_sin(positions: torch.Tensor, head_dim: int, theta: float = 10000.0,
                 dtype: torch.dtype = torch.float32) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables for given positions. (N,) -> (N, head_dim//2)."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2,
                                             device=positions.device).float() / head_dim))
    freqs = positions.float().unsqueeze(-1) * inv_freq  # (N, D/2)
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def apply_rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
               sin: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """NEOX-style (rotate-half) RoPE.

    q: (N, Hq, D), k: (N, Hk, D); cos/sin: (N, D/2).
    """
    def rot(x):
        d = x.shape[-1] // 2
        x1, x2 = x[..., :d], x[..., d:]
        c = cos.unsqueeze(1)  # (N,1,D/2)
        s = sin.unsqueeze(1)
        o1 = x1.float() * c - x2.float() * s
        o2 = x2.float() * c + x1.float() * s
        return torch.cat([o1, o2], dim=-1).to(x.dtype)
    return rot(q), rot(k)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


def gather_logprobs(logits: torch.Tensor, labels: torch.Tensor
                    ) -> torch.Tensor:
    """(N, V), (N,) -> (N,) log softmax at labels, fp32."""
    logp = torch.log_softmax(logits.float(), dim=-1)
    return torch.gather(logp, -1, labels.unsqueeze(-1)).squeeze(-1)


def gather_logprobs_entropy(logits: torch.Tensor, labels: torch.Tensor
                            ) -> Tuple[torch.Tensor, torch.Tensor]:
    logp = torch.log_softmax(logits.float(), dim=-1)
    lp = torch.gather(logp, -1, labels.unsqueeze(-1)).squeeze(-1)
    ent = -(logp.exp() * logp).sum(-1)
    return lp, ent


def top_k_top_p_sample(logits: torch.Tensor, temperature: torch.Tensor,
                       top_k: torch.Tensor, top_p: torch.Tensor,
                       generator: Optional[torch.Generator] = None
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-row temperature + top-k + top-p filtering and categorical draw.

    logits (N, V) fp32; temperature/top_k/top_p per-row (N,).
    Returns (token_ids (N,) int64, logprobs (N,) fp32 of the sampled token
    under the *filtered, renormalized* distribution's log-softmax of the
    ORIGINAL distribution — we report logprob under the unfiltered softmax,
    matching what rollout clients consume as output_token_logprobs).
    greedy when temperature == 0.
    """
    N, V = logits.shape
    logits = logits.float()
    out_tokens = torch.empty(N, dtype=torch.long, device=logits.device)
    logp_full = torch.log_softmax(logits, dim=-1)
    for i in range(N):
        t = float(temperature[i])
        if t == 0.0:
            tok = int(torch.argmax(logits[i]).item())
            out_tokens[i] = tok
            continue
        row = logits[i] / t
        k = int(top_k[i])
        if 0 < k < V:
            kth = torch.topk(row, k).values[-1]
            row = torch.where(row < kth, torch.full_like(row, float("-inf")), row)
        p = float(top_p[i])
        if p < 1.0:
            sorted_row, idx = torch.sort(row, descending=True)
            probs = torch.softmax(sorted_row, dim=-1)
            cum = torch.cumsum(probs, dim=-1)
            # keep tokens until cumulative prob exceeds p (always keep first)
            cutoff = cum - probs >= p
            sorted_row[cutoff] = float("-inf")
            row = torch.full_like(row, float("-inf"))
            row[idx] = sorted_row
        probs = torch.softmax(row, dim=-1)
        tok = int(torch.multinomial(probs, 1, generator=generator).item())
        out_tokens[i] = tok
    logps = logp_full.gather(-1, out_tokens.unsqueeze(-1)).squeeze(-1)
    return out_tokens, logps


def paged_attention_decode(
    q: torch.Tensor,            # (B, Hq, D)
    kv_cache_k: torch.Tensor,   # (num_pages, page_size, Hk, D)
    kv_cache_v: torch.Tensor,   # (num_pages, page_size, Hk, D)
    page_table: torch.Tensor,   # (B, max_pages) int32
    context_lens: torch.Tensor, # (B,) int32  (length INCLUDING current token)
    scale: float,
) -> torch.Tensor:
    """One new query token per sequence attending to its paged KV history."""
    B, Hq, D = q.shape
    Hk = kv_cache_k.shape[2]
    page_size = kv_cache_k.shape[1]
    g = Hq // Hk
    out = torch.empty_like(q, dtype=torch.float32)
    for b in range(B):
        L = int(context_lens[b])
        n_pages = -(-L // page_size)
        pages = page_table[b, :n_pages].long()
        k = kv_cache_k[pages].reshape(-1, Hk, D)[:L].float()  # (L, Hk, D)
        v = kv_cache_v[pages].reshape(-1, Hk, D)[:L].float()
        for h in range(Hq):
            hk = h // g
            s = (q[b, h].float() @ k[:, hk].T) * scale       # (L,)
            p = torch.softmax(s, dim=-1)
            out[b, h] = p @ v[:, hk]
    return out.to(q.dtype)


def varlen_prefill_attention(
    q: torch.Tensor,            # (total_q, Hq, D)
    k: torch.Tensor,            # (total_k, Hk, D)
    v: torch.Tensor,            # (total_k, Hk, D)
    cu_seqlens_q: torch.Tensor, # (B+1,)
    cu_seqlens_k: torch.Tensor, # (B+1,)
    scale: float,
    causal: bool = True,
) -> torch.Tensor:
    """Varlen (packed) causal attention.  When seqlen_k > seqlen_q the query
    block is aligned to the END of the keys (chunked-prefill semantics)."""
    Hq, D = q.shape[1], q.shape[2]
    Hk = k.shape[1]
    g = Hq // Hk
    out = torch.empty_like(q, dtype=torch.float32)
    B = cu_seqlens_q.shape[0] - 1
    for b in range(B):
        q0, q1 = int(cu_seqlens_q[b]), int(cu_seqlens_q[b + 1])
        k0, k1 = int(cu_seqlens_k[b]), int(cu_seqlens_k[b + 1])
        Lq, Lk = q1 - q0, k1 - k0
        qi = q[q0:q1].float()
        ki = k[k0:k1].float()
        vi = v[k0:k1].float()
        for h in range(Hq):
            hk = h // g
            s = (qi[:, h] @ ki[:, hk].T) * scale  # (Lq, Lk)
            if causal:
                qpos = torch.arange(Lq, device=q.device).unsqueeze(1) + (Lk - Lq)
                kpos = torch.arange(Lk, device=q.device).unsqueeze(0)
                s = s.masked_fill(kpos > qpos, float("-inf"))
            p = torch.softmax(s, dim=-1)
            out[q0:q1, h] = p @ vi[:, hk]
    return out.to(q.dtype)


def varlen_lse(q, k, cu_seqlens_q, cu_seqlens_k, scale, causal=True):
    """Log-sum-exp of the (masked) attention scores per (token, q-head) —
    reference for the prefill kernel's LSE output."""
    Hq = q.shape[1]
    Hk = k.shape[1]
    g = Hq // Hk
    out = torch.empty(q.shape[0], Hq, dtype=torch.float32)
    B = cu_seqlens_q.shape[0] - 1
    for b in range(B):
        q0, q1 = int(cu_seqlens_q[b]), int(cu_seqlens_q[b + 1])
        k0, k1 = int(cu_seqlens_k[b]), int(cu_seqlens_k[b + 1])
        Lq, Lk = q1 - q0, k1 - k0
        qi = q[q0:q1].float()
        ki = k[k0:k1].float()
        for h in range(Hq):
            s = (qi[:, h] @ ki[:, h // g].T) * scale
            if causal:
                qpos = torch.arange(Lq).unsqueeze(1) + (Lk - Lq)
                kpos = torch.arange(Lk).unsqueeze(0)
                s = s.masked_fill(kpos > qpos, float("-inf"))
            out[q0:q1, h] = torch.logsumexp(s, dim=-1)
    return out
