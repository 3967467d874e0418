"""Model registry: named configs for the BASELINE.json model set.

All models are random-init (no network for checkpoints); parameter names are
HF-compatible so checkpoints interchange with the wider ecosystem and with
the verl-style sharded checkpoint layout.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class DecoderConfig:
    """Config for the Llama/Qwen2 dense decoder family."""
    arch: str = "llama"            # llama | qwen2 | gpt2
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    head_dim: Optional[int] = None
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    tie_word_embeddings: bool = False
    attention_bias: bool = False   # qwen2: True (QKV bias)
    mlp_bias: bool = False
    # gpt2 only
    layer_norm_eps: float = 1e-5

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_attention_heads

    @property
    def num_params_estimate(self) -> int:
        h, i, L, v = (self.hidden_size, self.intermediate_size,
                      self.num_hidden_layers, self.vocab_size)
        hq = self.num_attention_heads * self.head_dim
        hk = self.num_key_value_heads * self.head_dim
        attn = h * hq + 2 * h * hk + hq * h
        mlp = 3 * h * i
        emb = v * h * (1 if self.tie_word_embeddings else 2)
        return L * (attn + mlp + 2 * h) + emb + h


# Named configs — the BASELINE.json model set plus small test models.
MODEL_CONFIGS = {
    # config #1 (CPU tier): GPT-2 small
    "gpt2-small": DecoderConfig(
        arch="gpt2", vocab_size=50257, hidden_size=768, intermediate_size=3072,
        num_hidden_layers=12, num_attention_heads=12, num_key_value_heads=12,
        max_position_embeddings=1024, tie_word_embeddings=True),
    # tiny debug models (CPU unit tests / fast GPU smoke)
    "gpt2-debug": DecoderConfig(
        arch="gpt2", vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=4,
        max_position_embeddings=512, tie_word_embeddings=True),
    "llama-tiny": DecoderConfig(
        arch="llama", vocab_size=1024, hidden_size=256, intermediate_size=688,
        num_hidden_layers=2, num_attention_heads=8, num_key_value_heads=2,
        head_dim=128, max_position_embeddings=2048, rope_theta=10000.0),
    "llama-debug-cpu": DecoderConfig(
        arch="llama", vocab_size=512, hidden_size=128, intermediate_size=344,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=32, max_position_embeddings=1024, rope_theta=10000.0),
    # config #2: Qwen2.5-1.5B
    "qwen2.5-1.5b": DecoderConfig(
        arch="qwen2", vocab_size=151936, hidden_size=1536, intermediate_size=8960,
        num_hidden_layers=28, num_attention_heads=12, num_key_value_heads=2,
        max_position_embeddings=32768, rope_theta=1000000.0, rms_norm_eps=1e-6,
        attention_bias=True, tie_word_embeddings=True),
    # config #4: Qwen2.5-7B
    "qwen2.5-7b": DecoderConfig(
        arch="qwen2", vocab_size=152064, hidden_size=3584, intermediate_size=18944,
        num_hidden_layers=28, num_attention_heads=28, num_key_value_heads=4,
        max_position_embeddings=32768, rope_theta=1000000.0, rms_norm_eps=1e-6,
        attention_bias=True),
    # configs #3 / headline metric: Llama-3-8B
    "llama3-8b": DecoderConfig(
        arch="llama", vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
        max_position_embeddings=8192, rope_theta=500000.0, rms_norm_eps=1e-5),
    # config #5: Llama-3-70B
    "llama3-70b": DecoderConfig(
        arch="llama", vocab_size=128256, hidden_size=8192, intermediate_size=28672,
        num_hidden_layers=80, num_attention_heads=64, num_key_value_heads=8,
        max_position_embeddings=8192, rope_theta=500000.0, rms_norm_eps=1e-5),
    # 1B-class model for single-GPU sanity benches
    "llama3-1b": DecoderConfig(
        arch="llama", vocab_size=128256, hidden_size=2048, intermediate_size=8192,
        num_hidden_layers=16, num_attention_heads=32, num_key_value_heads=8,
        head_dim=64, max_position_embeddings=8192, rope_theta=500000.0),
}


def get_model_config(name: str, override: dict = None) -> DecoderConfig:
    if name not in MODEL_CONFIGS:
        raise KeyError(f"unknown model {name!r}; known: {sorted(MODEL_CONFIGS)}")
    cfg = MODEL_CONFIGS[name]
    if override:
        import dataclasses
        cfg = dataclasses.replace(cfg, **override)   # registry stays pristine
    return cfg
