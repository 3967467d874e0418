"""polyrl_amd — an MI355X-native RL-for-LLM framework.

A from-scratch reinforcement-learning-for-LLMs stack (PPO / GRPO) designed for
AMD Instinct MI355X (gfx950, CDNA4):

- trainer: FSDP2 (``fully_shard``) on PyTorch-ROCm, collectives on RCCL over xGMI
- rollout: in-process continuous-batching decoder with paged KV cache whose hot
  kernels (paged decode attention, prefill attention, RMSNorm, RoPE, SiLU-mul,
  sampling, fused logprobs) are hand-written CDNA4 HIP (MFMA + LDS tiles)
- scheduler: in-process rollout scheduler with streamed result delivery,
  token-level continuation on instance failure and an adaptive local time-box
- weight plane: trainer -> rollout weight publication over RCCL / direct device
  copies (co-located) with version gating

Capability reference: Terra-Flux/PolyRL.  This is not a port —
the architecture is MI355X-first.
"""

__version__ = "0.1.0"
