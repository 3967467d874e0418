"""Op-category breakdown of one bench step (torch.profiler key_averages).

Buckets every CUDA op into {gemm, attention, elementwise, norm, optimizer,
reduce, copy, sample/engine, other} so the update-phase fat is attributed
by CATEGORY (kernel-name stats can't separate trainer vs engine hipBLASLt
calls).  Run on MI355X:

    python profiles/profile_step_ops.py > gpurun_out/step_ops.txt
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bucket(name: str) -> str:
    n = name.lower()
    if "attn" in n or "attention" in n or "sdpa" in n:
        return "attention"
    if any(k in n for k in ("mm", "matmul", "linear", "bmm", "addmm",
                            "tunedlinear")):
        return "gemm"
    if any(k in n for k in ("adam", "_foreach", "multi_tensor",
                            "optimizer")):
        return "optimizer"
    if "norm" in n:
        return "norm"
    if any(k in n for k in ("copy", "contiguous", "clone", "cat", "pad",
                            "index", "masked", "scatter", "gather",
                            "embedding")):
        return "copy/index"
    if any(k in n for k in ("add", "mul", "div", "silu", "exp", "neg",
                            "rsub", "sub", "clamp", "where", "fill",
                            "zero", "sum", "mean", "softmax", "cumsum")):
        return "elementwise/reduce"
    if any(k in n for k in ("sample", "rope", "kv_cache")):
        return "engine-op"
    return "other"


def main():
    assert torch.cuda.is_available()
    from polyrl_amd.config import PPOConfig
    from polyrl_amd.reward import load_reward_manager
    from polyrl_amd.trainer.stream_trainer import StreamPPOTrainer

    cfg = PPOConfig()
    cfg.actor_rollout_ref.model.path = "llama3-8b"
    cfg.actor_rollout_ref.model.enable_gradient_checkpointing = False
    cfg.critic.model.path = "llama3-8b"
    cfg.critic.model.enable_gradient_checkpointing = False
    cfg.actor_rollout_ref.actor.ppo_mini_batch_size = 64
    cfg.actor_rollout_ref.actor.ppo_max_token_len_per_gpu = 8192
    cfg.actor_rollout_ref.actor.fsdp.reshard_after_forward = False
    cfg.critic.fsdp.reshard_after_forward = False
    cfg.critic.ppo_mini_batch_size = 64
    cfg.critic.ppo_max_token_len_per_gpu = 8192
    cfg.actor_rollout_ref.rollout.sampling.n = 8
    cfg.actor_rollout_ref.rollout.prompt_length = 256
    cfg.actor_rollout_ref.rollout.response_length = 256
    cfg.actor_rollout_ref.rollout.min_stream_batch_size = 32
    cfg.data.train_batch_size = 16
    cfg.data.max_prompt_length = 256
    cfg.data.synthetic_num_prompts = 64
    cfg.trainer.device = "cuda"
    cfg.trainer.logger = []
    cfg.trainer.resume_mode = "disable"
    cfg.trainer.default_local_dir = "/tmp/polyrl_prof_ckpt"

    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=1)          # warmup
    torch.cuda.synchronize()

    with torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA]) as prof:
        trainer.fit(max_steps=1)
        torch.cuda.synchronize()

    buckets = {}
    rows = []
    for e in prof.key_averages():
        t = e.self_device_time_total
        if t <= 0:
            continue
        b = bucket(e.key)
        buckets[b] = buckets.get(b, 0.0) + t
        rows.append((t, e.count, e.key, b))
    total = sum(buckets.values())
    print(f"total self-CUDA time {total/1e6:.2f}s")
    for b, t in sorted(buckets.items(), key=lambda kv: -kv[1]):
        print(f"  {t/1e3:9.1f} ms {100*t/total:5.1f}%  {b}")
    print("\ntop 40 ops by self CUDA time:")
    rows.sort(key=lambda r: -r[0])
    for t, c, k, b in rows[:40]:
        print(f"{t/1e3:9.1f} ms {c:7d}x  [{b}] {k[:80]}")


if __name__ == "__main__":
    main()
