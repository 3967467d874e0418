"""Full streamed GRPO/PPO loop on CPU — BASELINE config #1 tier.

Covers: stream batching with exact sizes, minibatch-boundary opt steps,
GRPO group advantage over streamed groups, weight publication to the engine,
checkpoint save/resume round-trip, GAE/critic path.
"""
import os

import pytest
import torch

from polyrl_amd.config import PPOConfig, apply_overrides
from polyrl_amd.reward import load_reward_manager
from polyrl_amd.trainer.stream_trainer import StreamPPOTrainer


def tiny_config(tmp_path, model="llama-debug-cpu", adv="grpo",
                **over) -> PPOConfig:
    cfg = PPOConfig()
    cfg.actor_rollout_ref.model.path = model
    cfg.actor_rollout_ref.model.dtype = "float32"
    cfg.actor_rollout_ref.model.enable_gradient_checkpointing = False
    cfg.actor_rollout_ref.actor.ppo_mini_batch_size = 8
    cfg.actor_rollout_ref.actor.use_dynamic_bsz = True
    cfg.actor_rollout_ref.actor.ppo_max_token_len_per_gpu = 512
    cfg.actor_rollout_ref.rollout.sampling.n = 2
    cfg.actor_rollout_ref.rollout.prompt_length = 16
    cfg.actor_rollout_ref.rollout.response_length = 8
    cfg.actor_rollout_ref.rollout.min_stream_batch_size = 4
    cfg.algorithm.adv_estimator = adv
    cfg.critic.model.path = model
    cfg.critic.model.dtype = "float32"
    cfg.critic.model.enable_gradient_checkpointing = False
    cfg.critic.ppo_mini_batch_size = 8
    cfg.critic.ppo_max_token_len_per_gpu = 512
    cfg.data.train_batch_size = 8
    cfg.data.max_prompt_length = 16
    cfg.data.synthetic_num_prompts = 64
    cfg.trainer.device = "cpu"
    cfg.trainer.default_local_dir = str(tmp_path / "ckpt")
    cfg.trainer.logger = []
    cfg.trainer.total_epochs = 10
    cfg.trainer.resume_mode = "disable"
    for k, v in over.items():
        apply_overrides(cfg, [f"{k}={v}"])
    return cfg


def snapshot(model):
    return {k: v.detach().clone() for k, v in model.state_dict().items()}


def test_grpo_stream_loop_runs_and_updates(tmp_path):
    cfg = tiny_config(tmp_path)
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    before = snapshot(trainer.actor.model)
    trainer.fit(max_steps=2)
    after = snapshot(trainer.actor.model)
    changed = any(not torch.equal(before[k], after[k]) for k in before)
    assert changed, "actor params unchanged after 2 GRPO steps"
    assert trainer.global_step == 2
    # engine got the published weights (version bumped each step)
    assert trainer.publisher.version == 2
    # engine weight == actor weight after final publish? publish happens at
    # step start, so engine holds the post-step-1 weights
    assert not trainer.coordinator.has_work()


def test_ppo_gae_critic_path(tmp_path):
    cfg = tiny_config(tmp_path, adv="gae")
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    assert trainer.use_critic
    c_before = snapshot(trainer.critic.model)
    trainer.fit(max_steps=1)
    c_after = snapshot(trainer.critic.model)
    assert any(not torch.equal(c_before[k], c_after[k]) for k in c_before)


def test_gpt2_plumbing_config1(tmp_path):
    """BASELINE config #1: GPT-2 GRPO, constant reward, world_size=1 CPU."""
    cfg = tiny_config(tmp_path, model="gpt2-debug")
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("constant"))
    trainer.fit(max_steps=1)
    assert trainer.global_step == 1


def test_kl_in_reward_and_ref_worker(tmp_path):
    cfg = tiny_config(tmp_path)
    cfg.algorithm.use_kl_in_reward = True
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    assert trainer.ref is not None
    trainer.fit(max_steps=1)


def test_checkpoint_save_resume(tmp_path):
    cfg = tiny_config(tmp_path)
    cfg.trainer.save_freq = 2
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=2)
    w = snapshot(trainer.actor.model)
    del trainer
    # resume into a fresh trainer
    cfg2 = tiny_config(tmp_path)
    cfg2.trainer.resume_mode = "auto"
    trainer2 = StreamPPOTrainer(cfg2, reward_fn=load_reward_manager("random"))
    assert trainer2.global_step == 2
    w2 = snapshot(trainer2.actor.model)
    for k in w:
        assert torch.equal(w[k], w2[k]), f"mismatch after resume: {k}"


def test_minibatch_boundary_math(tmp_path):
    """opt steps happen exactly total/mini times per global batch"""
    cfg = tiny_config(tmp_path)
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    steps = {"n": 0}
    orig = trainer.actor._optimizer_step

    def counting():
        steps["n"] += 1
        return orig()

    trainer.actor._optimizer_step = counting
    trainer.fit(max_steps=1)
    total = cfg.data.train_batch_size * cfg.actor_rollout_ref.rollout.sampling.n
    expect = total // cfg.actor_rollout_ref.actor.ppo_mini_batch_size
    assert steps["n"] == expect, f"{steps['n']} opt steps, expected {expect}"


def test_validate_greedy_rollouts(tmp_path):
    cfg = tiny_config(tmp_path)
    cfg.trainer.test_freq = 1
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=1)          # runs validate() at step 1 via test_freq
    val = trainer.validate(num_prompts=4)
    assert "val/score/mean" in val
    assert val["val/n"] == 4.0
    assert 0.0 <= val["val/score/mean"] <= 1.0


def test_gpt2_constant_reward_config1(tmp_path):
    """BASELINE config #1: GPT-2 family, GRPO, constant reward, world 1 on
    CPU — the plumbing tier the reference defines for no-GPU runs."""
    cfg = tiny_config(tmp_path, model="gpt2-debug")
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("constant"))
    before = snapshot(trainer.actor.model)
    trainer.fit(max_steps=2)
    after = snapshot(trainer.actor.model)
    # constant reward => zero advantage => pg_loss 0; params may still move
    # only via optimizer side effects, so just assert the loop is stable
    for k in before:
        assert torch.isfinite(after[k]).all()
    val = trainer.validate(num_prompts=4)
    assert val["val/score/mean"] == 1.0       # constant reward


def test_remax_greedy_baseline_path(tmp_path):
    """ReMax: every step runs an extra greedy rollout whose sequence reward
    becomes the per-prompt baseline; advantage = score - baseline on every
    response token (core/algos.py::compute_remax_outcome_advantage)."""
    cfg = tiny_config(tmp_path, adv="remax")
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    before = snapshot(trainer.actor.model)
    trainer.fit(max_steps=2)
    after = snapshot(trainer.actor.model)
    assert any(not torch.equal(before[k], after[k]) for k in before)
    # every trained sample's uid had a baseline from the greedy pass
    full = trainer._last_full_batch
    assert all(str(u) in trainer._remax_baselines for u in full["uid"])
    # advantage really is (seq score - baseline) broadcast over the mask
    scores = full["token_level_rewards"].sum(-1)
    base = torch.tensor([trainer._remax_baselines[str(u)]
                         for u in full["uid"]])
    mask = full["response_mask"].float()
    expect = (scores - base).unsqueeze(-1) * mask
    assert torch.allclose(full["advantages"], expect, atol=1e-6)


def test_profile_steps_emit_chrome_trace(tmp_path):
    """trainer.profile_steps wraps those steps in a torch.profiler window
    and writes a chrome trace (reference: per-step nsys start/stop hooks,
    main_stream.py:79-93 capability)."""
    cfg = tiny_config(tmp_path)
    cfg.trainer.profile_steps = [2]
    cfg.trainer.profile_dir = str(tmp_path / "prof")
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=2)
    assert (tmp_path / "prof" / "step_2.json").exists()
    assert not (tmp_path / "prof" / "step_1.json").exists()


def test_esi_expiry_forces_checkpoint(tmp_path, monkeypatch):
    """POLYRL_ESI_EXPIRE_AT near in the future forces a save even with
    save_freq disabled (should_save_ckpt_esi capability,
    stream_ray_trainer.py:604-623)."""
    import time

    cfg = tiny_config(tmp_path)
    cfg.trainer.save_freq = -1
    monkeypatch.setenv("POLYRL_ESI_EXPIRE_AT", str(time.time() + 5))
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=1)
    ckpts = list((tmp_path / "ckpt").glob("global_step_*"))
    assert ckpts, "ESI expiry did not trigger a checkpoint"


def test_reference_fences(tmp_path):
    """Config fences the reference also has: ppo_epochs != 1 raises at
    construction; multi-turn without an interaction file is a config
    error."""
    cfg = tiny_config(tmp_path)
    cfg.actor_rollout_ref.actor.ppo_epochs = 2
    with pytest.raises(NotImplementedError):
        StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    cfg2 = tiny_config(tmp_path)
    cfg2.actor_rollout_ref.rollout.multi_turn.enable = True
    with pytest.raises(ValueError):
        StreamPPOTrainer(cfg2, reward_fn=load_reward_manager("random"))


def test_tis_importance_weight_path(tmp_path):
    """actor.tis_imp_ratio_cap reweights tokens by pi_old/pi_rollout
    (capped): the run completes and trains."""
    cfg = tiny_config(tmp_path)
    cfg.actor_rollout_ref.actor.tis_imp_ratio_cap = 2.0
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    before = snapshot(trainer.actor.model)
    trainer.fit(max_steps=1)
    after = snapshot(trainer.actor.model)
    assert any(not torch.equal(before[k], after[k]) for k in before)


def test_parquet_train_files_and_val_before_train(tmp_path):
    """data.train_files/val_files load tokenized parquet prompts;
    trainer.val_before_train runs a validation pass at step 0."""
    import numpy as np
    import pandas as pd

    rows = [{"prompt": f"q{i}", "input_ids": list(range(1, 6 + i % 3)),
             "data_source": "openai/gsm8k", "ground_truth": str(i)}
            for i in range(16)]
    f = tmp_path / "train.parquet"
    pd.DataFrame(rows).to_parquet(f)
    cfg = tiny_config(tmp_path)
    cfg.data.train_files = [str(f)]
    cfg.data.val_files = [str(f)]
    cfg.data.train_batch_size = 8
    cfg.trainer.val_before_train = True
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    assert len(trainer.dataset) == 16
    assert trainer.val_dataset is not None
    trainer.fit(max_steps=1)
    assert trainer.global_step == 1
    full = trainer._last_full_batch
    assert "ground_truth" in full.non_tensors       # passthrough for rewards


def test_resume_from_explicit_path(tmp_path):
    """trainer.resume_mode=resume_path restores the named global_step_N."""
    cfg = tiny_config(tmp_path)
    cfg.trainer.save_freq = 1
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=2)               # saves steps 1 and 2
    cfg2 = tiny_config(tmp_path)
    cfg2.trainer.resume_mode = "resume_path"
    cfg2.trainer.resume_from_path = str(
        tmp_path / "ckpt" / "global_step_1")
    t2 = StreamPPOTrainer(cfg2, reward_fn=load_reward_manager("random"))
    assert t2.global_step == 1             # the EARLIER step, not latest


def test_checkpoint_contents_model_only(tmp_path):
    """actor.checkpoint_contents=[model] writes no optimizer shards
    (configurable save contents, stream_fsdp_workers.py:357-376)."""
    import os

    cfg = tiny_config(tmp_path)
    cfg.actor_rollout_ref.actor.checkpoint_contents = ["model"]
    cfg.trainer.save_freq = 1
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=1)
    d = tmp_path / "ckpt" / "global_step_1" / "actor"
    files = os.listdir(d)
    assert any(f.startswith("model_") for f in files)
    assert not any(f.startswith("optim_") for f in files)


def test_free_cache_engine_and_override_config(tmp_path):
    """rollout.free_cache_engine releases the KV pool between steps
    (rollout_mode/trainer_mode dance); model.override_config patches the
    registry geometry without mutating it."""
    from polyrl_amd.models import get_model_config

    cfg = tiny_config(tmp_path)
    cfg.actor_rollout_ref.rollout.free_cache_engine = True
    cfg.actor_rollout_ref.model.override_config = {"num_hidden_layers": 1}
    cfg.trainer.test_freq = 2            # validate() on a released engine
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    assert len(trainer.actor.model.model.layers) == 1
    assert get_model_config("llama-debug-cpu").num_hidden_layers != 1
    trainer.fit(max_steps=2)
    # KV tensors are released after the step's stream drained
    assert trainer.engine.kv.k_cache == []


def test_rollout_data_dump(tmp_path):
    """trainer.rollout_data_dir dumps each step's samples as jsonl
    (stream_ray_trainer.py:585-587 capability)."""
    import json

    cfg = tiny_config(tmp_path)
    cfg.trainer.rollout_data_dir = str(tmp_path / "dump")
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    trainer.fit(max_steps=1)
    f = tmp_path / "dump" / "step_1.jsonl"
    rows = [json.loads(line) for line in f.read_text().splitlines()]
    assert len(rows) == 16                # 8 prompts x n=2
    assert all("uid" in r and "score" in r and r["response_ids"]
               for r in rows)


def test_multi_turn_rollout(tmp_path):
    """Multi-turn rollouts (reference MultiTurnConfig capability): an
    interaction file generates the next user turn; user tokens attend but
    carry no loss; training runs end-to-end."""
    inter = tmp_path / "interaction.py"
    inter.write_text(
        "def generate_turn(prompt_ids, response_ids):\n"
        "    # one user follow-up of 3 tokens, then done\n"
        "    if sum(1 for _ in response_ids) >= 10:\n"
        "        return None, True\n"
        "    return [7, 8, 9], False\n")
    cfg = tiny_config(tmp_path)
    cfg.actor_rollout_ref.rollout.response_length = 16   # room for 2 turns
    mt = cfg.actor_rollout_ref.rollout.multi_turn
    mt.enable = True
    mt.interaction_path = str(inter)
    mt.max_assistant_turns = 2
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))
    assert trainer.coordinator.multi_turn is not None
    trainer.fit(max_steps=1)

    # unit-level: the loss mask marks assistant tokens only
    import torch

    from polyrl_amd.protocol import TensorBatch
    from polyrl_amd.rollout.engine import SamplingParams
    coord = trainer.coordinator
    prompts = TensorBatch(
        tensors={"input_ids": torch.tensor([[11, 12, 13]]),
                 "attention_mask": torch.ones(1, 3, dtype=torch.long)},
        non_tensors={"uid": __import__("numpy").array(["u0"], dtype=object)})
    coord.submit(prompts, SamplingParams(temperature=1.0, max_new_tokens=5),
                 n=1)
    batches = list(coord.stream_batches(1))
    assert len(batches) == 1
    b = batches[0]
    resp_mask = b["response_mask"][0]
    attn = b["attention_mask"][0][3:]      # response presence
    # assistant turn 1 (5 toks) + user (3 toks, masked) + assistant turn 2
    n_present = int(attn.sum())
    n_loss = int(resp_mask.sum())
    assert n_present > n_loss >= 5, (n_present, n_loss)
    # the user tokens [7,8,9] sit in the response with loss 0
    resp = b["responses"][0][:n_present].tolist()
    assert [7, 8, 9] == resp[5:8]
    assert resp_mask[5:8].sum() == 0


def test_async_decode_matches_serial(tmp_path, monkeypatch):
    """The async decode pump must produce EXACTLY the serial rollouts (the
    engine's counter-based RNG is deterministic; the pump only moves the
    same engine.step() calls to a side thread)."""
    def run(async_on):
        monkeypatch.setenv("POLYRL_ASYNC_DECODE", "1" if async_on else "0")
        cfg = tiny_config(tmp_path / ("a" if async_on else "s"))
        trainer = StreamPPOTrainer(cfg,
                                   reward_fn=load_reward_manager("random"))
        trainer.fit(max_steps=1)
        full = trainer._last_full_batch
        return (full["responses"].clone(), full["rollout_log_probs"].clone(),
                snapshot(trainer.actor.model))

    r_async, lp_async, w_async = run(True)
    r_sync, lp_sync, w_sync = run(False)
    assert torch.equal(r_async, r_sync)
    assert torch.equal(lp_async, lp_sync)
    for k in w_sync:
        assert torch.allclose(w_async[k], w_sync[k], atol=1e-6), k


def test_async_decode_with_multi_turn(tmp_path, monkeypatch):
    """The decode pump thread performs the multi-turn interaction calls and
    resubmissions; results must match the serial path exactly."""
    inter = tmp_path / "interaction.py"
    inter.write_text(
        "def generate_turn(prompt_ids, response_ids):\n"
        "    if len(response_ids) >= 9:\n"
        "        return None, True\n"
        "    return [5, 6], False\n")

    def run(async_on):
        monkeypatch.setenv("POLYRL_ASYNC_DECODE", "1" if async_on else "0")
        cfg = tiny_config(tmp_path / ("a2" if async_on else "s2"))
        cfg.actor_rollout_ref.rollout.response_length = 16
        mt = cfg.actor_rollout_ref.rollout.multi_turn
        mt.enable = True
        mt.interaction_path = str(inter)
        mt.max_assistant_turns = 3
        mt.per_turn_max_tokens = 5
        trainer = StreamPPOTrainer(cfg,
                                   reward_fn=load_reward_manager("random"))
        trainer.fit(max_steps=1)
        f = trainer._last_full_batch
        return (f["responses"].clone(), f["response_mask"].clone(),
                f["attention_mask"].clone())

    r_a, m_a, am_a = run(True)
    r_s, m_s, am_s = run(False)
    assert torch.equal(r_a, r_s)
    assert torch.equal(m_a, m_s)
    assert torch.equal(am_a, am_s)
    # multi-turn really engaged: some present tokens carry no loss
    assert int(am_a[:, -16:].sum()) > int(m_a.sum())


def test_main_stream_reward_config_dapo(tmp_path):
    """reward=config entry: reward_model.reward_manager=dapo + parquet
    ground truth, through the real main_stream entry point (the
    reference's load_reward_manager precedence, reward.py:95-150)."""
    import subprocess
    import sys

    import pandas as pd
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    rows = [{"prompt": f"what is {i}+1?",
             "input_ids": [7 + i % 11, 3, 5, (i * 13) % 500],
             "data_source": "openai/gsm8k",
             "ground_truth": str(i + 1)} for i in range(16)]
    pd.DataFrame(rows).to_parquet(tmp_path / "g.parquet")
    env = dict(os.environ)
    env["PYTHONPATH"] = repo
    r = subprocess.run(
        [sys.executable, "-m", "polyrl_amd.trainer.main_stream",
         "actor_rollout_ref.model.path=llama-debug-cpu",
         "actor_rollout_ref.model.dtype=float32",
         "actor_rollout_ref.model.enable_gradient_checkpointing=false",
         "actor_rollout_ref.actor.ppo_mini_batch_size=8",
         "actor_rollout_ref.rollout.sampling.n=2",
         "actor_rollout_ref.rollout.response_length=8",
         "actor_rollout_ref.rollout.min_stream_batch_size=4",
         "data.train_batch_size=8",
         "data.max_prompt_length=24",
         f"data.train_files=[{tmp_path}/g.parquet]",
         "reward_model.reward_manager=dapo",
         "reward_model.overlong_buffer_len=2",
         f"trainer.default_local_dir={tmp_path}/ckpt",
         "trainer.resume_mode=disable",
         "reward=config",
         "max_steps=1"],
        capture_output=True, text=True, timeout=420, env=env)
    assert r.returncode == 0, f"{r.stdout[-2500:]}\n{r.stderr[-2500:]}"


def test_lora_end_to_end_freezes_base_and_publishes_merged(tmp_path):
    """LoRA path through the FULL streamed loop (model.lora_rank>0,
    reference capability: LoRA collect/summon in the FSDP machinery,
    stream_fsdp_workers.py:221-233): base weights stay frozen, adapters
    train, and the engine receives MERGED weights on publication."""
    cfg = tiny_config(tmp_path, adv="grpo")
    cfg.actor_rollout_ref.model.lora_rank = 4
    cfg.actor_rollout_ref.model.lora_alpha = 8.0
    trainer = StreamPPOTrainer(cfg, reward_fn=load_reward_manager("random"))

    actor = trainer.actor.model
    base_before = {k: v.detach().clone()
                   for k, v in actor.state_dict().items()
                   if ".base." in k or "embed" in k}
    lora_before = {k: v.detach().clone()
                   for k, v in actor.state_dict().items() if ".lora_" in k}
    assert lora_before, "apply_lora did not wrap any linears"

    trainer.fit(max_steps=2)

    sd = actor.state_dict()
    for k, v in base_before.items():
        assert torch.equal(sd[k], v), f"frozen base weight changed: {k}"
    assert any(not torch.equal(sd[k], v) for k, v in lora_before.items()), \
        "no LoRA adapter weight changed after 2 steps"

    # publication folded the deltas: engine weights == merged_state_dict
    from polyrl_amd.models.lora import merged_state_dict
    merged = merged_state_dict(actor)
    eng_sd = dict(trainer.engine.model._name_map)   # fused-buffer views
    common = [k for k in merged if k in eng_sd]
    assert common, "no overlapping names between merged sd and engine"
    for k in common:
        assert torch.allclose(eng_sd[k].float(), merged[k].float(),
                              atol=1e-5), f"engine got unmerged {k}"
