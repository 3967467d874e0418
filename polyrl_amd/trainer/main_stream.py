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
              f"adv={cfg.algorithm.adv_estimator}", flush=True)
    if reward_name == "config":
        # reference precedence: custom fn file > sandbox-wrapped dispatch,
        # manager style from reward_model.reward_manager
        from ..reward import load_reward_manager_from_config
        from ..utils.tokenizer import get_tokenizer
        tok = get_tokenizer(cfg.actor_rollout_ref.model.path)
        if tok is None:
            # registry-name models carry no HF tokenizer assets: decode to
            # the space-joined token ids so rule scorers still run
            # (useful for synthetic plumbing; real runs pass an HF dir)
            class _IdsTok:
                def decode(self, ids):
                    return " ".join(str(int(i)) for i in ids)
            tok = _IdsTok()
            if rank == 0:
                print("[main_stream] no HF tokenizer for "
                      f"{cfg.actor_rollout_ref.model.path!r}; rule scorers "
                      "will see space-joined token ids", flush=True)
        reward_fn = load_reward_manager_from_config(cfg, tokenizer=tok)
    else:
        reward_fn = load_reward_manager(reward_name)

    ro = cfg.actor_rollout_ref.rollout
    if ro.num_rollout_ranks > 0:
        # disaggregated split (BASELINE config #4): trailing ranks serve
        # rollout, the rest train.  The trainer subgroup must be created on
        # EVERY rank (collective), before the roles branch.
        from ..models import get_model_config
        from .disagg import advertise_addr, rollout_serve_loop, split_roles
        assert world > 1, "disaggregated mode needs a multi-rank world"
        trainer_ranks, rollout_ranks = split_roles(world,
                                                   ro.num_rollout_ranks)
        tg = dist.new_group(trainer_ranks)
        # address map: each rollout rank advertises where its HTTP engine
        # server is reachable (multi-node: its xGMI-node IP; single node:
        # 127.0.0.1).  Exchanged over the world group before the branch.
        box = [None] * world
        dist.all_gather_object(
            box, advertise_addr() if rank in rollout_ranks else None)
        rollout_addrs = {r: box[r] for r in rollout_ranks}
        sp = cfg.actor_rollout_ref.actor.ulysses_sequence_parallel_size
        if sp > 1:
            # SP groups over the TRAINER subgroup — created on every rank
            # (world collective), used only by trainer ranks
            from ..parallel.ulysses import build_sp_groups
            from .workers import register_sp_groups
            spg, dpg = build_sp_groups(sp, ranks=trainer_ranks)
            if rank in trainer_ranks:
                register_sp_groups(sp, spg, dpg)
        model_cfg = get_model_config(cfg.actor_rollout_ref.model.path)
        print(f"[main_stream] rank {rank}: role="
              f"{'rollout' if rank in rollout_ranks else 'trainer'}",
              flush=True)
        if rank in rollout_ranks:
            dtype = getattr(torch, ro.dtype) if device_type == "cuda" \
                else torch.float32
            rollout_serve_loop(cfg, model_cfg, rank, device_type, dtype,
                               port_base=ro.rollout_port_base)
        else:
            trainer = StreamPPOTrainer(cfg, reward_fn=reward_fn,
                                       process_group=tg,
                                       rollout_addrs=rollout_addrs)
            print(f"[main_stream] rank {rank}: trainer ready", flush=True)
            trainer.fit(max_steps=max_steps)
            print(f"[main_stream] rank {rank}: fit done", flush=True)
            trainer.publisher.shutdown()   # releases the rollout ranks
    else:
        trainer = StreamPPOTrainer(cfg, reward_fn=reward_fn)
        trainer.fit(max_steps=max_steps)
    if dist.is_available() and dist.is_initialized():
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
