"""Entry point: streamed PPO/GRPO training, one process per GPU.

    # single process (CPU or 1 GPU)
    python -m polyrl_amd.trainer.main_stream data.train_batch_size=32 ...

    # 8 GPUs, RCCL over xGMI
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 -m polyrl_amd.trainer.main_stream ...

Reference capability: rlboost/verl_stream/trainer/main_stream.py (Hydra entry
+ TaskRunner) — here a plain SPMD launcher with dot-path overrides.
"""
from __future__ import annotations

import os
import sys

import torch
import torch.distributed as dist

from ..config import load_config
from ..reward import load_reward_manager
from .stream_trainer import StreamPPOTrainer


def init_distributed(device_type: str):
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        backend = "nccl" if device_type == "cuda" else "gloo"
        dist.init_process_group(backend=backend)
        if device_type == "cuda":
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        return dist.get_world_size(), dist.get_rank()
    return 1, 0


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    yaml_path = None
    overrides = []
    reward_name = "constant"
    max_steps = None
    for a in argv:
        if a.startswith("--config="):
            yaml_path = a.split("=", 1)[1]
        elif a.startswith("reward="):
            reward_name = a.split("=", 1)[1]
        elif a.startswith("max_steps="):
            max_steps = int(a.split("=", 1)[1])
        else:
            overrides.append(a)
    cfg = load_config(yaml_path, overrides)
    device_type = "cuda" if torch.cuda.is_available() else "cpu"
    cfg.trainer.device = device_type
    world, rank = init_distributed(device_type)
    if rank == 0:
        print(f"[main_stream] world={world} device={device_type} "
              f"model={cfg.actor_rollout_ref.model.path} "
              f"adv={cfg.algorithm.adv_estimator}")
    reward_fn = load_reward_manager(reward_name)
    trainer = StreamPPOTrainer(cfg, reward_fn=reward_fn)
    trainer.fit(max_steps=max_steps)
    if max_steps is None or cfg.trainer.save_freq == -2:
        pass
    if dist.is_available() and dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
