"""Per-call GEMM shape/layout profile of the trainer step — round-2 step (a)
of the GEMM plan (profiles/PROFILES.md).  Run on an MI355X:

    timeout 900 python profiles/profile_trainer_gemms.py > gpurun_out/gemms.txt

Runs ONE bench-shaped PPO step under torch.profiler with input shapes and
aggregates every GEMM-family kernel call by (op, input_shapes), sorted by
total CUDA time — the table that says WHICH matmul slots are slow in situ
(vs hipBLASLt's 1.5-1.6 PF on isolated contiguous shapes) and whether the
slow ones are transposed-layout dgrad/wgrad, the vocab GEMM, or skinny
decode projections.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

GEMM_OPS = ("aten::mm", "aten::addmm", "aten::bmm", "aten::matmul",
            "aten::linear", "aten::baddbmm")


def main():
    assert torch.cuda.is_available(), "GPU required"
    from polyrl_amd.config import PPOConfig
    from polyrl_amd.reward import load_reward_manager
    from polyrl_amd.trainer.stream_trainer import StreamPPOTrainer

    cfg = PPOConfig()
    cfg.actor_rollout_ref.model.path = "llama3-8b"
    cfg.actor_rollout_ref.model.dtype = "bfloat16"
    cfg.actor_rollout_ref.model.enable_gradient_checkpointing = False
    cfg.critic.model.path = "llama3-8b"
    cfg.critic.model.dtype = "bfloat16"
    cfg.critic.model.enable_gradient_checkpointing = False
    cfg.actor_rollout_ref.actor.ppo_mini_batch_size = 64
    cfg.actor_rollout_ref.actor.ppo_max_token_len_per_gpu = 8192
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
                        torch.profiler.ProfilerActivity.CUDA],
            record_shapes=True) as prof:
        trainer.fit(max_steps=1)
        torch.cuda.synchronize()

    # aggregate GEMM ops by input shape
    rows = {}
    for e in prof.key_averages(group_by_input_shape=True):
        if e.key in GEMM_OPS:
            t = e.device_time_total
            rows.setdefault((e.key, str(e.input_shapes)),
                            [0.0, 0])
            rows[(e.key, str(e.input_shapes))][0] += t
            rows[(e.key, str(e.input_shapes))][1] += e.count
    total = sum(v[0] for v in rows.values())
    print(f"total GEMM-op CUDA time: {total/1e3:.1f} ms over "
          f"{sum(v[1] for v in rows.values())} calls")
    print(f"{'us_total':>10} {'calls':>6} {'us/call':>9}  op  input_shapes")
    for (op, shapes), (t, n) in sorted(rows.items(),
                                       key=lambda kv: -kv[1][0])[:40]:
        print(f"{t:10.0f} {n:6d} {t/max(n,1):9.1f}  {op}  {shapes[:120]}")


if __name__ == "__main__":
    main()
