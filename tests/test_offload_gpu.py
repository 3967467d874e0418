"""GPU tier for the 70B memory levers: FSDP2
param+optimizer CPU offload and the FSDP->TP-engine publish path must
execute on real MI355X hardware, not just in docs/memory_70b.md arithmetic.

Single GPU, world-1 nccl process group (the FSDP/DTensor machinery is the
same one the 8-GPU run uses; only the shard count differs)."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda"


def _tiny_cfg():
    from polyrl_amd.models.registry import DecoderConfig
    return DecoderConfig(arch="llama", vocab_size=512, hidden_size=256,
                         intermediate_size=512, num_hidden_layers=2,
                         num_attention_heads=2, num_key_value_heads=1,
                         head_dim=128, max_position_embeddings=256,
                         rope_theta=10000.0, rms_norm_eps=1e-6)


@pytest.fixture()
def world1():
    import torch.distributed as dist
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29731")
        dist.init_process_group("nccl", world_size=1, rank=0)
    yield
    # keep the group for the next test in this module (teardown at exit)


def test_fsdp2_cpu_offload_step_and_publish(world1):
    """Param+optimizer CPU offload (docs/memory_70b.md lever): a training
    step runs with sharded params resident on HOST between uses, and the
    per-param all-gather publish still lands correct bytes in the engine."""
    from polyrl_amd.config import ActorConfig
    from polyrl_amd.models import create_model
    from polyrl_amd.rollout.engine import Engine, SamplingParams
    from polyrl_amd.trainer.workers import ActorWorker
    from polyrl_amd.transfer.weight_transfer import WeightPublisher
    from polyrl_amd.protocol import TensorBatch

    cfg = _tiny_cfg()
    torch.manual_seed(70)
    model = create_model(cfg, kind="actor", dtype="bfloat16", device=DEV)
    ref_sd = {k: v.detach().clone() for k, v in model.state_dict().items()}

    acfg = ActorConfig()
    acfg.fsdp.param_offload = True
    acfg.use_dynamic_bsz = False
    acfg.ppo_micro_batch_size_per_gpu = 8
    worker = ActorWorker(model, acfg, device=DEV)

    # sharded params must live on CPU between uses (CPUOffloadPolicy)
    off_devices = {p.device.type for p in worker.model.parameters()}
    assert off_devices == {"cpu"}, off_devices

    B, Lp, Lr = 4, 8, 8
    ids = torch.randint(0, cfg.vocab_size, (B, Lp + Lr), device=DEV)
    batch = TensorBatch(tensors={
        "input_ids": ids.cpu(),
        "attention_mask": torch.ones(B, Lp + Lr, dtype=torch.long),
        "position_ids": torch.arange(Lp + Lr).expand(B, -1).contiguous(),
        "responses": ids[:, Lp:].cpu(),
        "response_mask": torch.ones(B, Lr),
        "old_log_probs": torch.zeros(B, Lr),
        "advantages": torch.randn(B, Lr),
    })
    m = worker.update_policy_stream(batch, is_opt_step=True, is_lr_step=True,
                                    accum_scale=1.0)
    gn = m["actor/grad_norm"][0]
    assert gn == gn and gn < 1e6, gn
    # optimizer state created by the step must be host-resident too
    for st in worker.optimizer.state.values():
        for v in st.values():
            if torch.is_tensor(v) and v.numel() > 1:
                assert v.device.type == "cpu", v.device

    # publish from the offloaded FSDP model into a GPU engine
    eng = Engine(cfg, device=DEV, dtype=torch.bfloat16,
                 kv_bytes_budget=32 << 20)
    pub = WeightPublisher(worker.model, [eng.model])
    pub.publish()
    sd_now = {k: (v.full_tensor() if hasattr(v, "full_tensor") else v)
              for k, v in worker.model.state_dict().items()}
    name = "model.layers.0.mlp.down_proj.weight"
    got = eng.model._name_map[name]
    assert torch.equal(got, sd_now[name].to(got.dtype).to(got.device))
    # a step actually happened: weights differ from init
    assert not torch.equal(sd_now[name].cpu().float(),
                           ref_sd[name].cpu().float())
    # engine decodes fine on the published weights
    outs = eng.generate([[1, 2, 3]], SamplingParams(temperature=0.0,
                                                    max_new_tokens=4), "p")
    assert len(outs[0].output_ids) == 4


class _StubTP:
    """TPContext stand-in for single-process reshard tests: carries
    size/rank without a process group (no collectives are issued by
    construction or update_named)."""

    def __init__(self, size, rank):
        self.size = size
        self.rank = rank
        self.group = None
        self.enabled = size > 1


def test_fsdp_to_tp_reshard_publish_on_device(world1):
    """On-device FSDP->TP reshard (config #5 lever): full params gathered
    from the FSDP2 (DTensor) model are sliced into each TP rank's engine
    buffers directly on the GPU — row shards for qkv/gate/up, col shards
    for o/down, vocab shards for lm_head — and the shards tile exactly."""
    from polyrl_amd.models import create_model
    from polyrl_amd.models.registry import DecoderConfig
    from polyrl_amd.rollout.engine import InferenceModel
    from polyrl_amd.trainer.workers import _maybe_fully_shard

    cfg = DecoderConfig(arch="llama", vocab_size=512, hidden_size=256,
                        intermediate_size=512, num_hidden_layers=2,
                        num_attention_heads=4, num_key_value_heads=2,
                        head_dim=64, max_position_embeddings=256,
                        rope_theta=10000.0, rms_norm_eps=1e-6)
    torch.manual_seed(71)
    model = create_model(cfg, kind="actor", dtype="bfloat16", device=DEV)
    fsdp_model = _maybe_fully_shard(model)
    shards = [InferenceModel(cfg, device=DEV, dtype=torch.bfloat16,
                             tp_ctx=_StubTP(2, r)) for r in range(2)]
    sd = {}
    for name, p in fsdp_model.state_dict().items():
        sd[name] = p.full_tensor() if hasattr(p, "full_tensor") else p
        assert sd[name].device.type == "cuda"      # no host round-trip
        for s in shards:
            s.update_named(name, sd[name])
    # row-parallel: q rows split across ranks
    qname = "model.layers.0.self_attn.q_proj.weight"
    full_q = sd[qname]
    assert torch.equal(shards[0]._name_map[qname], full_q[:full_q.shape[0] // 2])
    assert torch.equal(shards[1]._name_map[qname], full_q[full_q.shape[0] // 2:])
    # col-parallel: down cols split
    dname = "model.layers.0.mlp.down_proj.weight"
    full_d = sd[dname]
    assert torch.equal(shards[0]._name_map[dname], full_d[:, :full_d.shape[1] // 2])
    assert torch.equal(shards[1]._name_map[dname], full_d[:, full_d.shape[1] // 2:])
    # vocab-parallel lm_head tiles the full vocab
    lm = sd["lm_head.weight"]
    cat = torch.cat([shards[0]._name_map["lm_head.weight"],
                     shards[1]._name_map["lm_head.weight"]], dim=0)
    assert torch.equal(cat, lm)
