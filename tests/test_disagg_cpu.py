"""Disaggregated split mode (gloo, world_size=3: 2 trainer + 1 rollout rank)
— BASELINE config #4 path: scheduler-driven generation over the HTTP facade,
bucketed weight broadcast over the world group, streamed updates
(SURVEY.md §3.2-§3.4)."""
import os
import subprocess
import sys

import pytest

from conftest import free_port, retry_run


@pytest.mark.timeout(600)
def test_disagg_2train_1rollout(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = retry_run(lambda: subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=3", "--master-addr", "127.0.0.1",
         "--master-port", free_port(),
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
        capture_output=True, text=True, timeout=540, env=env))
    assert r.returncode == 0, \
        f"stdout:\n{r.stdout[-4000:]}\nstderr:\n{r.stderr[-4000:]}"


@pytest.mark.timeout(600)
def test_elastic_colocated_world2(tmp_path):
    """Elastic co-located mode: each rank serves its engine over HTTP,
    rank 0's scheduler drives the pool as local instances, manager facade
    serving on rank 0 (the reference's primary operating shape, §3.4)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = retry_run(lambda: subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", free_port(),
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
        capture_output=True, text=True, timeout=540, env=env))
    assert r.returncode == 0, \
        f"stdout:\n{r.stdout[-4000:]}\nstderr:\n{r.stderr[-4000:]}"


@pytest.mark.timeout(900)
def test_elastic_runtime_join(tmp_path):
    """A remote instance joins a RUNNING elastic training job through the
    manager facade (reference §3.4 lifecycle: register -> health gate ->
    version gate -> TCP weight push -> active pool) and the run completes."""
    import threading
    import time as _time

    import httpx
    import torch
    import uvicorn

    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.rollout.engine import Engine
    from polyrl_amd.server import create_app

    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.Popen(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", free_port(),
         "-m", "polyrl_amd.trainer.main_stream",
         "actor_rollout_ref.model.path=llama-debug-cpu",
         "actor_rollout_ref.model.dtype=float32",
         "actor_rollout_ref.model.enable_gradient_checkpointing=false",
         "actor_rollout_ref.actor.ppo_mini_batch_size=8",
         "actor_rollout_ref.actor.ppo_max_token_len_per_gpu=512",
         "actor_rollout_ref.rollout.name=elastic",
         "actor_rollout_ref.rollout.sampling.n=2",
         "actor_rollout_ref.rollout.response_length=16",
         "actor_rollout_ref.rollout.min_stream_batch_size=4",
         "actor_rollout_ref.rollout.rollout_port_base=31880",
         "actor_rollout_ref.rollout.rollout_manager_port=31879",
         "data.train_batch_size=8",
         "data.max_prompt_length=16",
         "data.synthetic_num_prompts=64",
         f"trainer.default_local_dir={tmp_path}/ckpt",
         "trainer.resume_mode=disable",
         "reward=random",
         "max_steps=6",
         ], env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)

    joined = {"ok": False, "active": False, "version": -1}
    try:
        # our "spot node": a served engine on this test process
        cfg = get_model_config("llama-debug-cpu")
        torch.manual_seed(1)
        eng = Engine(cfg, device="cpu", dtype=torch.float32,
                     kv_bytes_budget=16 << 20)
        for t in eng.model._name_map.values():
            t.normal_(0, 0.02)
        embed_before = eng.model.embed.clone()
        app = create_app(eng)
        server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1",
                                               port=31885,
                                               log_level="error"))
        threading.Thread(target=server.run, daemon=True).start()

        mgr = "http://127.0.0.1:31879"
        deadline = _time.monotonic() + 120
        registered = False
        while _time.monotonic() < deadline and proc.poll() is None:
            try:
                if not registered:
                    r = httpx.post(f"{mgr}/register_rollout_instance",
                                   json={"addr": "http://127.0.0.1:31885"},
                                   timeout=30.0)
                    registered = r.status_code == 200
                    continue
                d = httpx.get(f"{mgr}/get_instances_status",
                              timeout=5.0).json()
                me = [i for i in d["instances"]
                      if i["id"].endswith(":31885")]
                if me:
                    joined["ok"] = True
                    joined["sender"] = me[0].get("weight_sender_endpoint")
                    if me[0]["active"]:
                        joined["active"] = True
                        joined["version"] = me[0]["weight_version"]
                        break
            except Exception:
                pass
            _time.sleep(0.3)
        out, _ = proc.communicate(timeout=600)
        server.should_exit = True
        assert proc.returncode == 0, out[-4000:]
        assert joined["ok"], "instance never appeared in the manager roster"
        assert joined["active"], "joined instance never became active"
        assert joined["version"] >= 1   # received a weight version via TCP
        # the trainer registered itself as the weight sender and the
        # scheduler assigned it to the joining instance (PUT
        # /update_weight_senders -> round-robin at registration)
        assert joined.get("sender"), "no weight_sender_endpoint assigned"
        # TCP push actually replaced our random init with trainer weights
        assert not torch.equal(eng.model.embed, embed_before)
    finally:
        if proc.poll() is None:
            proc.kill()


@pytest.mark.timeout(600)
def test_disagg_with_ulysses_sp2(tmp_path):
    """Disagg split + Ulysses SP inside the trainer subgroup (world=3:
    2 trainer ranks forming one SP=2 group + 1 rollout rank).  The SP
    groups are a world collective built before the role branch
    (main_stream.py / workers.register_sp_groups)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = retry_run(lambda: subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=3", "--master-addr", "127.0.0.1",
         "--master-port", free_port(),
         "-m", "polyrl_amd.trainer.main_stream",
         "actor_rollout_ref.model.path=llama-debug-cpu",
         "actor_rollout_ref.model.dtype=float32",
         "actor_rollout_ref.model.enable_gradient_checkpointing=false",
         "actor_rollout_ref.actor.ppo_mini_batch_size=8",
         "actor_rollout_ref.actor.ppo_max_token_len_per_gpu=512",
         "actor_rollout_ref.actor.ulysses_sequence_parallel_size=2",
         "actor_rollout_ref.rollout.sampling.n=2",
         "actor_rollout_ref.rollout.response_length=8",
         "actor_rollout_ref.rollout.min_stream_batch_size=4",
         "actor_rollout_ref.rollout.num_rollout_ranks=1",
         "actor_rollout_ref.rollout.rollout_port_base=31830",
         "trainer.save_freq=1",
         "data.train_batch_size=8",
         "data.max_prompt_length=16",
         "data.synthetic_num_prompts=32",
         f"trainer.default_local_dir={tmp_path}/ckpt",
         "trainer.resume_mode=disable",
         "reward=random",
         "max_steps=2",
         ],
        capture_output=True, text=True, timeout=540, env=env))
    assert r.returncode == 0, \
        f"stdout:\n{r.stdout[-4000:]}\nstderr:\n{r.stderr[-4000:]}"
    assert "fit done" in r.stdout


def test_scheduler_path_multi_turn():
    """Multi-turn through the DisaggCoordinator/scheduler path: the
    interaction's user turns are inserted with loss 0 and the assistant
    continuations run through the scheduler (unit-level, FakeInstance)."""
    import asyncio
    import time

    import numpy as np
    import torch

    from polyrl_amd.protocol import TensorBatch
    from polyrl_amd.rollout.engine import SamplingParams
    from polyrl_amd.scheduler.instances import FakeInstance
    from polyrl_amd.trainer.disagg import DisaggCoordinator

    mt = {"interaction":
          lambda prompt, resp: (None, True) if len(resp) >= 10
          else ([9, 9], False),
          "max_assistant_turns": 3, "max_user_turns": 5,
          "max_tool_response_length": 8, "per_turn_max_tokens": 4}
    coord = DisaggCoordinator(response_length=16, trainer_group=None,
                              rollout_urls=[], rank=0, n_trainer=1,
                              multi_turn=mt)

    async def _reg():
        await coord.scheduler.register_instance(
            FakeInstance("f0", is_local=True), skip_health_check=True)
    asyncio.run_coroutine_threadsafe(_reg(), coord.loop).result(timeout=30)

    prompts = TensorBatch(
        tensors={"input_ids": torch.tensor([[11, 12, 13]]),
                 "attention_mask": torch.ones(1, 3, dtype=torch.long)},
        non_tensors={"uid": np.array(["u0"], dtype=object)})
    coord.submit(prompts, SamplingParams(temperature=1.0, max_new_tokens=4),
                 n=1)
    shards = coord._next_shards(1)    # rank-0 view (no dist in this test)
    b = shards[0] if not isinstance(shards[0], list) else shards[0][0]
    attn = b["attention_mask"][0][3:]
    loss = b["response_mask"][0]
    n_present = int(attn.sum())
    n_loss = int(loss.sum())
    # turn1 (4 assistant) + user [9,9] + turn2 (4) + user + turn3 => user
    # tokens attend but carry no loss
    assert n_present > n_loss, (n_present, n_loss)
    resp = b["responses"][0][:n_present].tolist()
    assert resp[4:6] == [9, 9]
    assert loss[4:6].sum() == 0


@pytest.mark.timeout(600)
def test_disagg_multi_turn_world3(tmp_path):
    """Full-stack: disaggregated split (2 trainer + 1 rollout) WITH
    multi-turn interactions — user turns inserted by the trainer-side
    coordinator, continuations through the scheduler/HTTP engines."""
    inter = tmp_path / "interaction.py"
    inter.write_text(
        "def generate_turn(prompt_ids, response_ids):\n"
        "    if len(response_ids) >= 8:\n"
        "        return None, True\n"
        "    return [3, 4], False\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = retry_run(lambda: subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=3", "--master-addr", "127.0.0.1",
         "--master-port", free_port(),
         "-m", "polyrl_amd.trainer.main_stream",
         "actor_rollout_ref.model.path=llama-debug-cpu",
         "actor_rollout_ref.model.dtype=float32",
         "actor_rollout_ref.model.enable_gradient_checkpointing=false",
         "actor_rollout_ref.actor.ppo_mini_batch_size=8",
         "actor_rollout_ref.actor.ppo_max_token_len_per_gpu=512",
         "actor_rollout_ref.rollout.sampling.n=2",
         "actor_rollout_ref.rollout.response_length=16",
         "actor_rollout_ref.rollout.min_stream_batch_size=4",
         "actor_rollout_ref.rollout.num_rollout_ranks=1",
         "actor_rollout_ref.rollout.rollout_port_base=31880",
         "actor_rollout_ref.rollout.multi_turn.enable=true",
         f"actor_rollout_ref.rollout.multi_turn.interaction_path={inter}",
         "actor_rollout_ref.rollout.multi_turn.max_assistant_turns=2",
         "actor_rollout_ref.rollout.multi_turn.per_turn_max_tokens=4",
         "data.train_batch_size=8",
         "data.max_prompt_length=16",
         "data.synthetic_num_prompts=32",
         f"trainer.default_local_dir={tmp_path}/ckpt",
         "trainer.resume_mode=disable",
         "reward=random",
         "max_steps=1",
         ],
        capture_output=True, text=True, timeout=540, env=env))
    assert r.returncode == 0, \
        f"stdout:\n{r.stdout[-4000:]}\nstderr:\n{r.stderr[-4000:]}"


def test_multi_turn_survives_instance_failure_token_exact():
    """Multi-turn + fault tolerance composed: an instance that dies
    mid-assistant-turn is evicted and the turn CONTINUES token-exactly on
    the survivor, so the final multi-turn batch (responses, presence mask,
    assistant-only loss mask) is bitwise identical to a run with no
    failure at all (handlers.rs continuation x interaction turns)."""
    import asyncio

    import numpy as np
    import torch

    from polyrl_amd.protocol import TensorBatch
    from polyrl_amd.rollout.engine import SamplingParams
    from polyrl_amd.scheduler.instances import FakeInstance
    from polyrl_amd.trainer.disagg import DisaggCoordinator

    mt = {"interaction":
          lambda prompt, resp: (None, True) if len(resp) >= 10
          else ([9, 9], False),
          "max_assistant_turns": 3, "max_user_turns": 5,
          "max_tool_response_length": 8, "per_turn_max_tokens": 4}

    def run(insts):
        coord = DisaggCoordinator(response_length=16, trainer_group=None,
                                  rollout_urls=[], rank=0, n_trainer=1,
                                  multi_turn=dict(mt))

        async def _reg():
            for i in insts:
                await coord.scheduler.register_instance(
                    i, skip_health_check=True)
        asyncio.run_coroutine_threadsafe(_reg(), coord.loop).result(
            timeout=30)
        prompts = TensorBatch(
            tensors={"input_ids": torch.tensor([[11, 12, 13]]),
                     "attention_mask": torch.ones(1, 3, dtype=torch.long)},
            non_tensors={"uid": np.array(["u0"], dtype=object)})
        coord.submit(prompts,
                     SamplingParams(temperature=1.0, max_new_tokens=4), n=1)
        shards = coord._next_shards(1)
        b = shards[0] if not isinstance(shards[0], list) else shards[0][0]
        return b

    ref = run([FakeInstance("healthy", is_local=True)])
    # dies after 2 tokens of whichever turn it serves first -> evicted,
    # the turn continues on the survivor
    dying = FakeInstance("dies", is_local=True, fail_after_tokens=2)
    survivor = FakeInstance("survivor", is_local=True)
    got = run([dying, survivor])

    assert dying.served_gids, \
        "fault path not exercised: dying instance never picked"
    assert survivor.served_gids, "survivor never picked up the continuation"
    for key in ("responses", "attention_mask", "response_mask"):
        assert torch.equal(ref[key], got[key]), \
            f"{key} diverged under mid-turn failure:\n{ref[key]}\n{got[key]}"
