#!/usr/bin/env bash
# Streamed GRPO on 8x MI355X (the reference's run_async_grpo_pipeline.sh
# workload shape: train_batch 128, n=8, response up to 14336, dynamic token
# budget 16384/GPU, min_stream_batch 16 — SURVEY.md §6).
set -ex
python -m torch.distributed.run --nnodes=1 --nproc-per-node=8 \
  --master-addr 127.0.0.1 \
  -m polyrl_amd.trainer.main_stream \
  actor_rollout_ref.model.path=qwen2.5-1.5b \
  actor_rollout_ref.model.dtype=bfloat16 \
  actor_rollout_ref.actor.ppo_mini_batch_size=256 \
  actor_rollout_ref.actor.ppo_max_token_len_per_gpu=16384 \
  actor_rollout_ref.actor.use_kl_loss=true \
  actor_rollout_ref.rollout.sampling.n=8 \
  actor_rollout_ref.rollout.prompt_length=512 \
  actor_rollout_ref.rollout.response_length=14336 \
  actor_rollout_ref.rollout.min_stream_batch_size=16 \
  algorithm.adv_estimator=grpo \
  data.train_batch_size=128 \
  data.max_prompt_length=512 \
  trainer.save_freq=20 trainer.test_freq=20 \
  reward=config \
  "$@"
