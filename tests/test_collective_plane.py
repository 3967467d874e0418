"""CollectiveWeightPlane (gloo, world 2): bucketed broadcast round-trip,
bitwise equality, receiver-side TP slicing (SURVEY.md §4 tier 3:
multi-process weight-transfer round-trip)."""
import os
import subprocess
import sys

import pytest

from conftest import free_port, retry_run

WORKER = r"""
import torch, torch.distributed as dist
dist.init_process_group("gloo")
rank = dist.get_rank()

from polyrl_amd.transfer.collective import CollectiveWeightPlane, tp_slice


torch.manual_seed(0)
named = {
    "model.layers.0.self_attn.q_proj.weight": torch.randn(8, 16),
    "model.layers.0.self_attn.o_proj.weight": torch.randn(16, 8),
    "model.layers.0.mlp.gate_proj.weight": torch.randn(12, 16),
    "model.norm.weight": torch.randn(16),
    "big.weight": torch.randn(300, 40),   # forces a second bucket
}

plane = CollectiveWeightPlane(src=0, device="cpu", bucket_bytes=16 << 10)
if rank == 0:
    v = plane.publish(named, version=3)
    assert v == 3
else:
    got = {}
    v = plane.receive(lambda n, t: got.__setitem__(n, t.clone()),
                      tp_rank=1, tp_size=2, arch="llama")
    assert v == 3
    # TP slicing applied: q rows halved (rank 1 = second half), o cols halved
    assert got["model.layers.0.self_attn.q_proj.weight"].shape == (4, 16)
    assert torch.equal(got["model.layers.0.self_attn.q_proj.weight"],
                       named["model.layers.0.self_attn.q_proj.weight"][4:])
    assert got["model.layers.0.self_attn.o_proj.weight"].shape == (16, 4)
    assert torch.equal(got["model.layers.0.self_attn.o_proj.weight"],
                       named["model.layers.0.self_attn.o_proj.weight"][:, 4:])
    assert torch.equal(got["model.norm.weight"], named["model.norm.weight"])
    assert torch.equal(got["big.weight"], named["big.weight"])

# second publish reuses the plan (no re-negotiation) and bumps the version
named2 = {k: v + 1.0 for k, v in named.items()}
if rank == 0:
    assert plane.publish(named2, version=4) == 4
else:
    got2 = {}
    assert plane.receive(lambda n, t: got2.__setitem__(n, t.clone())) == 4
    assert torch.equal(got2["big.weight"], named2["big.weight"])
print(f"rank {rank}: plane OK")
"""


@pytest.mark.timeout(300)
def test_collective_plane_world2(tmp_path):
    script = tmp_path / "plane_worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = retry_run(lambda: subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", free_port(), str(script)],
        capture_output=True, text=True, timeout=240, env=env))
    assert r.returncode == 0, f"{r.stdout[-3000:]}\n{r.stderr[-3000:]}"
    assert "plane OK" in r.stdout


def test_tp_slice_qwen_biases_and_dims():
    """tp_slice unit coverage: qwen2 QKV biases shard dim 0 alongside their
    weights; consumers shard dim 1; norms/embeddings replicate; the engine
    vocab-shards lm_head itself (engine.py receiver special-case)."""
    import torch

    from polyrl_amd.transfer.collective import tp_shard_dim, tp_slice

    w = torch.arange(24, dtype=torch.float32).reshape(6, 4)
    b = torch.arange(6, dtype=torch.float32)
    assert tp_shard_dim("model.layers.0.self_attn.k_proj.bias") == 0
    assert torch.equal(tp_slice("l.self_attn.q_proj.bias", b, 1, 2), b[3:])
    assert torch.equal(tp_slice("l.self_attn.v_proj.weight", w, 0, 2), w[:3])
    assert torch.equal(tp_slice("l.mlp.down_proj.weight", w, 1, 2), w[:, 2:])
    # replicated tensors come back untouched for any rank
    assert torch.equal(tp_slice("model.norm.weight", b, 1, 2), b)
    assert torch.equal(tp_slice("model.embed_tokens.weight", w, 1, 2), w)
    assert tp_shard_dim("lm_head.weight") is None   # engine special-case
    # gpt2 tier replicates everything
    assert tp_shard_dim("h.0.attn_qkv.weight", arch="gpt2") is None
