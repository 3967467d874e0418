"""Flagship benchmark — BASELINE.json metric:
end-to-end PPO samples/sec (rollout + update), Llama-3-8B, on N MI355X GPUs.

    python bench.py --gpus N --steps K --warmup W
    # N>1 is launched by the driver as
    # python -m torch.distributed.run --nnodes=1 --nproc-per-node N ... bench.py

Synthetic prompts, random-init weights (no network), bf16 compute.
Weak scaling: per-GPU batch fixed; `value` is the WHOLE-JOB samples/sec.
Each timed step = rollout generation (prompt prefill + response decode on the
in-process HIP-kernel engine) + reward + old logprobs + values + GAE +
streamed actor & critic updates (optimizer steps included) + weight
publication to the engine (every iteration, per BASELINE config #3).
"""
from __future__ import annotations

import argparse
import json
import os
import time

# hipBLASLt algo selections tuned offline on MI355X (tunableop/*.csv —
# PyTorch TunableOp).  OPT-IN via POLYRL_TUNABLEOP=1: a table tuned on a
# different box measured SLOWER on a fresh one (old_log_prob 0.94->3.2 s),
# so selections do not transfer reliably across boxes/clock states.
_TUNDIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "tunableop")
if os.path.isdir(_TUNDIR) and os.environ.get("POLYRL_TUNABLEOP") == "1":
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                          os.path.join(_TUNDIR, "tunableop_llama8b.csv"))
elif os.environ.get("POLYRL_TUNABLEOP_TUNE") == "1":
    # tune hipBLASLt algo selection ON THIS BOX during init/warmup (tight
    # per-shape budget); selections stay in-process for the timed region
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                          "/tmp/polyrl_tunableop.csv")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "10")
    os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_WARMUP_DURATION_MS", "5")
os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--model", type=str, default="llama3-8b")
    p.add_argument("--batch-per-gpu", type=int, default=16,
                   help="prompts per GPU per step (weak scaling)")
    p.add_argument("--n-samples", type=int, default=8,
                   help="rollout samples per prompt")
    p.add_argument("--prompt-len", type=int, default=256)
    p.add_argument("--response-len", type=int, default=256)
    p.add_argument("--adv", type=str, default="gae", choices=["gae", "grpo"])
    args = p.parse_args()

    # distributed init (driver launches via torch.distributed.run for N>1)
    world, rank = 1, 0
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        dist.init_process_group(backend="nccl" if torch.cuda.is_available()
                                else "gloo")
        world, rank = dist.get_world_size(), dist.get_rank()
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    assert world == args.gpus or "WORLD_SIZE" not in os.environ, \
        f"--gpus {args.gpus} != WORLD_SIZE {world}"
    device = "cuda" if torch.cuda.is_available() else "cpu"

    from polyrl_amd.config import PPOConfig
    from polyrl_amd.reward import load_reward_manager
    from polyrl_amd.trainer.stream_trainer import StreamPPOTrainer

    use_critic = args.adv == "gae"
    global_batch = args.batch_per_gpu * world
    total_samples = global_batch * args.n_samples

    cfg = PPOConfig()
    cfg.actor_rollout_ref.model.path = args.model
    cfg.actor_rollout_ref.model.dtype = "bfloat16" if device == "cuda" else "float32"
    # long-response configs may need activation headroom for KV
    ckpt = os.environ.get("POLYRL_BENCH_GRAD_CKPT") == "1"
    cfg.actor_rollout_ref.model.enable_gradient_checkpointing = ckpt
    cfg.critic.model.enable_gradient_checkpointing = ckpt
    # keep gathered params resident across the step's micro passes (one
    # all-gather per step; +full-param memory, fine at 288 GB)
    cfg.actor_rollout_ref.actor.fsdp.reshard_after_forward = False
    cfg.critic.fsdp.reshard_after_forward = False
    cfg.actor_rollout_ref.actor.ppo_mini_batch_size = total_samples // 2
    # micro token budget: 8192 fills the CU grid exactly (2 waves of 256
    # blocks at MT256); long-response configs need at least one sequence
    seq_len = args.prompt_len + args.response_len
    budget = max(8192, seq_len)
    cfg.actor_rollout_ref.actor.ppo_max_token_len_per_gpu = budget
    cfg.actor_rollout_ref.rollout.sampling.n = args.n_samples
    cfg.actor_rollout_ref.rollout.sampling.temperature = 1.0
    cfg.actor_rollout_ref.rollout.prompt_length = args.prompt_len
    cfg.actor_rollout_ref.rollout.response_length = args.response_len
    cfg.actor_rollout_ref.rollout.min_stream_batch_size = \
        max(args.n_samples * world, total_samples // 4)
    cfg.actor_rollout_ref.rollout.max_num_batched_tokens = 16384
    cfg.algorithm.adv_estimator = args.adv
    cfg.critic.model.path = args.model
    cfg.critic.model.dtype = cfg.actor_rollout_ref.model.dtype
    cfg.critic.ppo_mini_batch_size = total_samples // 2
    cfg.critic.ppo_max_token_len_per_gpu = budget
    cfg.data.train_batch_size = global_batch
    cfg.data.max_prompt_length = args.prompt_len
    cfg.data.synthetic_num_prompts = max(
        global_batch * (args.steps + args.warmup + 1), 64)
    cfg.data.shuffle = False
    cfg.trainer.device = device
    # POLYRL_BENCH_LOG=1 prints per-step phase timings (gen/prep/update/...)
    cfg.trainer.logger = (["console"]
                          if os.environ.get("POLYRL_BENCH_LOG") else [])
    # POLYRL_BENCH_PROFILE=1 wraps the LAST warmup step in torch.profiler
    # (chrome trace) without touching the timed region
    if os.environ.get("POLYRL_BENCH_PROFILE") and rank == 0:
        cfg.trainer.profile_steps = [args.warmup]
        cfg.trainer.profile_dir = "gpurun_out/torch_prof"
    cfg.trainer.resume_mode = "disable"
    cfg.trainer.default_local_dir = "/tmp/polyrl_bench_ckpt"

    if rank == 0:
        print(f"[bench] model={args.model} world={world} "
              f"global_batch={global_batch} n={args.n_samples} "
              f"prompt={args.prompt_len} resp={args.response_len} "
              f"adv={args.adv}", flush=True)

    t0 = time.perf_counter()
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    if rank == 0:
        print(f"[bench] init done in {time.perf_counter()-t0:.1f}s", flush=True)

    # warmup
    trainer.fit(max_steps=args.warmup)

    # timed region
    if dist.is_initialized():
        dist.barrier()
    if device == "cuda":
        torch.cuda.synchronize()
    t_start = time.perf_counter()
    trainer.fit(max_steps=args.steps)
    if device == "cuda":
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    elapsed = time.perf_counter() - t_start

    # max over ranks
    if dist.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    samples_per_s = total_samples * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "end-to-end PPO samples/sec (rollout+update), "
                      + args.model,
            "value": samples_per_s,
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if device == "cuda" else "fp32",
            "data": "synthetic prompts, random-init weights",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "samples_per_step": total_samples,
                "n_samples_per_prompt": args.n_samples,
                "prompt_len": args.prompt_len,
                "response_len": args.response_len,
                "seq_len": args.prompt_len + args.response_len,
                "adv_estimator": args.adv,
                "critic": use_critic,
                "parallelism": f"dp{world}",
                "weight_transfer": "every iter",
            },
        }), flush=True)

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
