"""TP rollout decoder (gloo, world_size=2): a TP=2 engine loaded from full
weights must generate greedily exactly what the TP=1 engine generates
(SURVEY.md §2.3 TP row; receiver-side resharding patches.py:196-241)."""
import os
import subprocess
import sys

import pytest

from conftest import free_port, retry_run

WORKER = r"""
import torch, torch.distributed as dist
dist.init_process_group("gloo")
rank = dist.get_rank()

from polyrl_amd.models import create_model, get_model_config
from polyrl_amd.parallel.tp import TPContext
from polyrl_amd.rollout.engine import Engine, SamplingParams


cfg = get_model_config("llama-debug-cpu")
torch.manual_seed(0)
model = create_model(cfg, kind="actor", dtype="float32", device="cpu")
sd = model.state_dict()

prompts = [[5, 9, 2, 7], [11, 3]]

# reference: TP=1 engine
eng1 = Engine(cfg, device="cpu", dtype=torch.float32, kv_bytes_budget=8 << 20)
eng1.model.load_state_dict(sd)
ref = eng1.generate(prompts, SamplingParams(temperature=0.0,
                                            max_new_tokens=6), "r")
ref_ids = [o.output_ids for o in ref]
ref_lps = [o.output_logprobs for o in ref]

# TP=2 engine: same full weights, sharded on ingest
tp = TPContext(dist.group.WORLD)
eng2 = Engine(cfg, device="cpu", dtype=torch.float32, kv_bytes_budget=8 << 20,
              tp_ctx=tp)
eng2.model.load_state_dict(sd)
# shard shapes really are sharded
w = eng2.model.layers[0].wqkv
assert w.shape[0] * 2 == (cfg.num_attention_heads + 2 * cfg.num_key_value_heads) * cfg.head_dim, w.shape
out = eng2.generate(prompts, SamplingParams(temperature=0.0,
                                            max_new_tokens=6), "t")
for i, o in enumerate(out):
    assert o.output_ids == ref_ids[i], (rank, o.output_ids, ref_ids[i])
    for a, b in zip(o.output_logprobs, ref_lps[i]):
        assert abs(a - b) < 1e-4, (a, b)

# stochastic decode agrees ACROSS TP ranks (determinism contract)
out_s = eng2.generate(prompts, SamplingParams(temperature=1.0,
                                              max_new_tokens=8), "s")
ids = [o.output_ids for o in out_s]
box = [None, None]
dist.all_gather_object(box, ids)
assert box[0] == box[1], f"TP ranks diverged: {box}"
print(f"rank {rank}: TP OK")
"""


@pytest.mark.timeout(600)
def test_tp2_engine_matches_tp1(tmp_path):
    script = tmp_path / "tp_worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = retry_run(lambda: subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", free_port(), str(script)],
        capture_output=True, text=True, timeout=540, env=env))
    assert r.returncode == 0, f"stdout:\n{r.stdout[-3000:]}\nstderr:\n{r.stderr[-3000:]}"
    assert "TP OK" in r.stdout


@pytest.mark.timeout(600)
def test_tp2_group_prefix_sharing(tmp_path):
    """TP engine + prefix-sharing group API: greedy equality with separate
    requests and cross-rank agreement (the two features compose)."""
    script = tmp_path / "tp_group_worker.py"
    script.write_text(r"""
import torch, torch.distributed as dist
dist.init_process_group("gloo")
from polyrl_amd.models import create_model, get_model_config
from polyrl_amd.parallel.tp import TPContext
from polyrl_amd.rollout.engine import Engine, SamplingParams

cfg = get_model_config("llama-debug-cpu")
torch.manual_seed(0)
model = create_model(cfg, kind="actor", dtype="float32", device="cpu")
tp = TPContext(dist.group.WORLD)
eng = Engine(cfg, device="cpu", dtype=torch.float32, kv_bytes_budget=8 << 20,
             tp_ctx=tp)
eng.model.load_state_dict(model.state_dict())
prompt = [5, 9, 2, 7, 11]
sp = SamplingParams(temperature=0.0, max_new_tokens=5)
eng.enable_prefix_sharing = False
sep = eng.generate([prompt] * 2, sp, "sep")
eng.enable_prefix_sharing = True
eng.add_request_group("grp", prompt, sp, 2)
outs = {}
while eng.has_work():
    for o in eng.step():
        outs[o.rid] = o
for s in range(2):
    assert outs[f"grp-s{s}"].output_ids == sep[s].output_ids
ids = [outs[f"grp-s{s}"].output_ids for s in range(2)]
box = [None, None]
dist.all_gather_object(box, ids)
assert box[0] == box[1], box
print(f"rank {dist.get_rank()}: TP+group OK")
""")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = retry_run(lambda: subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", free_port(), str(script)],
        capture_output=True, text=True, timeout=540, env=env))
    assert r.returncode == 0, f"{r.stdout[-3000:]}\n{r.stderr[-3000:]}"
    assert "TP+group OK" in r.stdout
