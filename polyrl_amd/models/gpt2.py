"""GPT-2-small — the CPU/gloo plumbing-tier model (BASELINE config #1).

Learned position embeddings, pre-LN, GELU MLP, MHA.  Runs the full streamed
GRPO loop on CPU with world_size=1..2 so the trainer, scheduler and
checkpoint paths are testable without a GPU.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .registry import DecoderConfig


class GPT2Block(nn.Module):
    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        h = cfg.hidden_size
        self.ln_1 = nn.LayerNorm(h, eps=cfg.layer_norm_eps)
        self.attn_qkv = nn.Linear(h, 3 * h)
        self.attn_out = nn.Linear(h, h)
        self.ln_2 = nn.LayerNorm(h, eps=cfg.layer_norm_eps)
        self.mlp_fc = nn.Linear(h, cfg.intermediate_size)
        self.mlp_proj = nn.Linear(cfg.intermediate_size, h)
        self.n_head = cfg.num_attention_heads
        self.head_dim = h // cfg.num_attention_heads

    def forward(self, x, attn_mask: Optional[torch.Tensor]):
        B, L, H = x.shape
        qkv = self.attn_qkv(self.ln_1(x))
        q, k, v = qkv.split(H, dim=-1)
        q = q.view(B, L, self.n_head, self.head_dim).transpose(1, 2)
        k = k.view(B, L, self.n_head, self.head_dim).transpose(1, 2)
        v = v.view(B, L, self.n_head, self.head_dim).transpose(1, 2)
        if attn_mask is not None:
            o = F.scaled_dot_product_attention(q, k, v, attn_mask=attn_mask)
        else:
            o = F.scaled_dot_product_attention(q, k, v, is_causal=True)
        o = o.transpose(1, 2).reshape(B, L, H)
        x = x + self.attn_out(o)
        x = x + self.mlp_proj(F.gelu(self.mlp_fc(self.ln_2(x))))
        return x


class GPT2LMModel(nn.Module):
    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        self.config = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.wpe = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.h = nn.ModuleList(GPT2Block(cfg) for _ in range(cfg.num_hidden_layers))
        self.ln_f = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.wte.weight
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def gradient_checkpointing_enable(self):
        pass

    def forward(self, input_ids, attention_mask=None, position_ids=None,
                logits_slice=None):
        B, L = input_ids.shape
        if position_ids is None:
            position_ids = torch.arange(L, device=input_ids.device).expand(B, L)
        x = self.wte(input_ids) + self.wpe(position_ids.clamp(min=0))
        attn_mask = None
        if attention_mask is not None and not bool(attention_mask.all()):
            causal = torch.ones(L, L, dtype=torch.bool,
                                device=input_ids.device).tril()
            attn_mask = causal.view(1, 1, L, L) & attention_mask.bool().view(B, 1, 1, L)
        for blk in self.h:
            x = blk(x, attn_mask)
        x = self.ln_f(x)
        if logits_slice is not None:
            x = x[:, logits_slice]
        return self.lm_head(x)

    def num_params(self):
        return sum(p.numel() for p in self.parameters())


class GPT2WithValueHead(nn.Module):
    def __init__(self, cfg: DecoderConfig):
        super().__init__()
        self.config = cfg
        self.trunk = GPT2LMModel(cfg)
        self.value_head = nn.Linear(cfg.hidden_size, 1, bias=False)
        nn.init.normal_(self.value_head.weight, std=0.001)

    def gradient_checkpointing_enable(self):
        pass

    def forward(self, input_ids, attention_mask=None, position_ids=None):
        B, L = input_ids.shape
        if position_ids is None:
            position_ids = torch.arange(L, device=input_ids.device).expand(B, L)
        t = self.trunk
        x = t.wte(input_ids) + t.wpe(position_ids.clamp(min=0))
        attn_mask = None
        if attention_mask is not None and not bool(attention_mask.all()):
            causal = torch.ones(L, L, dtype=torch.bool,
                                device=input_ids.device).tril()
            attn_mask = causal.view(1, 1, L, L) & attention_mask.bool().view(B, 1, 1, L)
        for blk in t.h:
            x = blk(x, attn_mask)
        return self.value_head(t.ln_f(x)).squeeze(-1)
