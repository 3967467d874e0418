#!/usr/bin/env bash
# Synchronous colocated GRPO baseline (the reference's
# run_sync_grpo_default.sh A/B arm): min_stream_batch == full batch, so the
# trainer waits for the whole generation before updating.
set -ex
python -m torch.distributed.run --nnodes=1 --nproc-per-node=8 \
  --master-addr 127.0.0.1 \
  -m polyrl_amd.trainer.main_ppo \
  actor_rollout_ref.model.path=qwen2.5-1.5b \
  actor_rollout_ref.actor.ppo_mini_batch_size=256 \
  actor_rollout_ref.rollout.sampling.n=8 \
  algorithm.adv_estimator=grpo \
  data.train_batch_size=256 \
  reward=naive \
  "$@"
