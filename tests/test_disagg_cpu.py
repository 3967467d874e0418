"""Disaggregated split mode (gloo, world_size=3: 2 trainer + 1 rollout rank)
— BASELINE config #4 path: scheduler-driven generation over the HTTP facade,
bucketed weight broadcast over the world group, streamed updates
(SURVEY.md §3.2-§3.4)."""
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(600)
def test_disagg_2train_1rollout(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=3", "--master-addr", "127.0.0.1",
         "--master-port", "29661",
         "-m", "polyrl_amd.trainer.main_stream",
         "actor_rollout_ref.model.path=llama-debug-cpu",
         "actor_rollout_ref.model.dtype=float32",
         "actor_rollout_ref.model.enable_gradient_checkpointing=false",
         "actor_rollout_ref.actor.ppo_mini_batch_size=8",
         "actor_rollout_ref.actor.ppo_max_token_len_per_gpu=512",
         "actor_rollout_ref.rollout.sampling.n=2",
         "actor_rollout_ref.rollout.response_length=8",
         "actor_rollout_ref.rollout.min_stream_batch_size=4",
         "actor_rollout_ref.rollout.num_rollout_ranks=1",
         "actor_rollout_ref.rollout.rollout_port_base=31800",
         "data.train_batch_size=8",
         "data.max_prompt_length=16",
         "data.synthetic_num_prompts=32",
         f"trainer.default_local_dir={tmp_path}/ckpt",
         "trainer.resume_mode=disable",
         "reward=random",
         "max_steps=2",
         ],
        capture_output=True, text=True, timeout=540, env=env)
    assert r.returncode == 0, \
        f"stdout:\n{r.stdout[-4000:]}\nstderr:\n{r.stderr[-4000:]}"


@pytest.mark.timeout(600)
def test_elastic_colocated_world2(tmp_path):
    """Elastic co-located mode: each rank serves its engine over HTTP,
    rank 0's scheduler drives the pool as local instances, manager facade
    serving on rank 0 (the reference's primary operating shape, §3.4)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", "29674",
         "-m", "polyrl_amd.trainer.main_stream",
         "actor_rollout_ref.model.path=llama-debug-cpu",
         "actor_rollout_ref.model.dtype=float32",
         "actor_rollout_ref.model.enable_gradient_checkpointing=false",
         "actor_rollout_ref.actor.ppo_mini_batch_size=8",
         "actor_rollout_ref.actor.ppo_max_token_len_per_gpu=512",
         "actor_rollout_ref.rollout.name=elastic",
         "actor_rollout_ref.rollout.sampling.n=2",
         "actor_rollout_ref.rollout.response_length=8",
         "actor_rollout_ref.rollout.min_stream_batch_size=4",
         "actor_rollout_ref.rollout.rollout_port_base=31860",
         "actor_rollout_ref.rollout.rollout_manager_port=31859",
         "data.train_batch_size=8",
         "data.max_prompt_length=16",
         "data.synthetic_num_prompts=32",
         f"trainer.default_local_dir={tmp_path}/ckpt",
         "trainer.resume_mode=disable",
         "reward=random",
         "max_steps=2",
         ],
        capture_output=True, text=True, timeout=540, env=env)
    assert r.returncode == 0, \
        f"stdout:\n{r.stdout[-4000:]}\nstderr:\n{r.stderr[-4000:]}"
