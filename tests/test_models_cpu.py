import torch

from polyrl_amd.models import create_model, get_model_config


def test_llama_forward_shapes():
    cfg = get_model_config("llama-debug-cpu")
    m = create_model(cfg, dtype="float32")
    x = torch.randint(0, cfg.vocab_size, (2, 10))
    logits = m(x)
    assert logits.shape == (2, 10, cfg.vocab_size)


def test_llama_padding_mask_invariance():
    """left-padded batch rows must match unpadded single-row forward"""
    torch.manual_seed(0)
    cfg = get_model_config("llama-debug-cpu")
    m = create_model(cfg, dtype="float32")
    ids = torch.randint(0, cfg.vocab_size, (1, 6))
    full = m(ids, attention_mask=torch.ones(1, 6))
    # left-pad by 3
    pad = torch.zeros(1, 3, dtype=torch.long)
    padded_ids = torch.cat([pad, ids], 1)
    am = torch.cat([torch.zeros(1, 3), torch.ones(1, 6)], 1)
    pos = torch.clamp(torch.cumsum(am, 1) - 1, min=0).long()
    out = m(padded_ids, attention_mask=am, position_ids=pos)
    assert torch.allclose(out[0, 3:], full[0], atol=1e-4)


def test_gpt2_forward_and_value_head():
    cfg = get_model_config("gpt2-small")
    # shrink for test speed
    cfg = type(cfg)(**{**cfg.__dict__, "num_hidden_layers": 2,
                       "hidden_size": 64, "intermediate_size": 128,
                       "num_attention_heads": 4, "num_key_value_heads": 4,
                       "head_dim": None, "vocab_size": 128})
    m = create_model(cfg, dtype="float32")
    x = torch.randint(0, cfg.vocab_size, (2, 8))
    assert m(x).shape == (2, 8, cfg.vocab_size)
    critic = create_model(cfg, kind="critic", dtype="float32")
    assert critic(x).shape == (2, 8)


def test_critic_value_head_llama():
    cfg = get_model_config("llama-debug-cpu")
    c = create_model(cfg, kind="critic", dtype="float32")
    x = torch.randint(0, cfg.vocab_size, (3, 5))
    v = c(x)
    assert v.shape == (3, 5)


def test_param_name_compat():
    """HF-style parameter names (checkpoint interchange)."""
    cfg = get_model_config("llama-debug-cpu")
    m = create_model(cfg, dtype="float32")
    names = set(m.state_dict().keys())
    assert "model.embed_tokens.weight" in names
    assert "model.layers.0.self_attn.q_proj.weight" in names
    assert "model.layers.1.mlp.down_proj.weight" in names
    assert "model.norm.weight" in names
    assert "lm_head.weight" in names


def test_grad_flow():
    cfg = get_model_config("llama-debug-cpu")
    m = create_model(cfg, dtype="float32")
    x = torch.randint(0, cfg.vocab_size, (2, 6))
    loss = m(x).float().logsumexp(-1).mean()
    loss.backward()
    g = m.model.layers[0].self_attn.q_proj.weight.grad
    assert g is not None and g.abs().sum() > 0


def test_remove_padding_matches_dense():
    """Packed varlen path == dense masked path at all valid positions."""
    import torch
    from polyrl_amd.models import create_model, get_model_config
    cfg = get_model_config("llama-debug-cpu")
    torch.manual_seed(11)
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    B, L = 3, 14
    ids = torch.randint(0, cfg.vocab_size, (B, L))
    am = torch.ones(B, L, dtype=torch.long)
    am[0, :5] = 0   # left-padded prompt
    am[2, :2] = 0
    pos = torch.cumsum(am, dim=1) - 1
    pos = pos.clamp(min=0)
    with torch.no_grad():
        dense = m(ids, attention_mask=am, position_ids=pos)
        m.model.use_remove_padding = True
        packed = m(ids, attention_mask=am, position_ids=pos)
        m.model.use_remove_padding = False
    valid = am.bool()
    err = (dense[valid] - packed[valid]).abs().max().item()
    assert err < 1e-3, err


def test_remove_padding_backward_cpu():
    """Packed path is differentiable on the CPU reference."""
    import torch
    from polyrl_amd.models import create_model, get_model_config
    cfg = get_model_config("llama-debug-cpu")
    torch.manual_seed(12)
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    m.model.use_remove_padding = True
    B, L = 2, 10
    ids = torch.randint(0, cfg.vocab_size, (B, L))
    am = torch.ones(B, L, dtype=torch.long)
    am[1, :3] = 0
    out = m(ids, attention_mask=am)
    out.float().pow(2).mean().backward()
    g = m.model.layers[0].self_attn.q_proj.weight.grad
    assert g is not None and torch.isfinite(g).all()


def test_lora_adapters_and_merge():
    """LoRA: only adapters train; merged publication matches manual math and
    round-trips into the rollout engine under HF names."""
    import torch
    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.models.lora import LoRALinear, apply_lora, merged_state_dict
    from polyrl_amd.rollout.engine import Engine

    cfg = get_model_config("llama-debug-cpu")
    torch.manual_seed(20)
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    wrapped = apply_lora(m, r=4, alpha=8.0)
    assert any(w.endswith("q_proj") for w in wrapped)
    trainable = [n for n, p in m.named_parameters() if p.requires_grad]
    assert all("lora_" in n or "value_head" in n for n in trainable)

    # nudge an adapter; merged weight = base + scale * B A
    lin = m.model.layers[0].self_attn.q_proj
    assert isinstance(lin, LoRALinear)
    with torch.no_grad():
        lin.lora_B.normal_(0, 0.1)
    merged = merged_state_dict(m)
    name = "model.layers.0.self_attn.q_proj.weight"
    expect = lin.base.weight + lin.lora_B @ lin.lora_A * lin.scaling
    assert torch.allclose(merged[name], expect, atol=1e-5)
    assert not any(".lora_" in k or ".base." in k for k in merged)

    # engine ingests the merged dict under plain HF names
    eng = Engine(cfg, device="cpu", dtype=torch.float32,
                 kv_bytes_budget=8 << 20)
    eng.model.load_state_dict(merged, strict=True)
    assert torch.allclose(
        eng.model.layers[0].wqkv[:lin.base.out_features], expect)


def test_lora_backward_only_adapters():
    import torch
    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.models.lora import apply_lora
    cfg = get_model_config("llama-debug-cpu")
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    apply_lora(m, r=2)
    ids = torch.randint(0, cfg.vocab_size, (2, 8))
    out = m(ids)
    out.float().pow(2).mean().backward()
    for n, p in m.named_parameters():
        if "lora_" in n:
            assert p.grad is not None and torch.isfinite(p.grad).all(), n
        elif "value_head" not in n:
            assert p.grad is None, n


def test_hf_checkpoint_loader_roundtrip(tmp_path):
    """Save a model's state dict as sharded safetensors (HF layout) and
    load it into a fresh model by name; tied lm_head fallback covered."""
    import torch
    from safetensors.torch import save_file

    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.models.hf_loader import load_hf_checkpoint

    cfg = get_model_config("llama-debug-cpu")
    torch.manual_seed(0)
    src = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    sd = {k: v.clone().contiguous() for k, v in src.state_dict().items()}
    # shard into 2 files like HF repos; drop lm_head to force the tied
    # fallback path
    keys = [k for k in sd if k != "lm_head.weight"]
    half = len(keys) // 2
    save_file({k: sd[k] for k in keys[:half]},
              str(tmp_path / "model-00001-of-00002.safetensors"))
    save_file({k: sd[k] for k in keys[half:]},
              str(tmp_path / "model-00002-of-00002.safetensors"))

    torch.manual_seed(1)
    dst = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    missing, unexpected = load_hf_checkpoint(dst, str(tmp_path))
    assert missing == [] and unexpected == []
    for k in keys:
        assert torch.equal(dst.state_dict()[k], sd[k]), k
    assert torch.equal(dst.state_dict()["lm_head.weight"],
                       sd["model.embed_tokens.weight"])


def test_pack_align_padding_equivalence(monkeypatch):
    """_forward_packed pads total tokens to a POLYRL_PACK_ALIGN multiple
    (GEMM M-alignment, measured 35-45% on MI355X trunk GEMMs); outputs and
    grads must equal the unpadded path exactly on the fp32 CPU tier."""
    import torch

    from polyrl_amd.models import create_model, get_model_config

    cfg = get_model_config("llama-tiny")
    torch.manual_seed(7)
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    m.model.use_remove_padding = True
    B, L = 3, 17
    ids = torch.randint(0, cfg.vocab_size, (B, L))
    mask = torch.ones(B, L, dtype=torch.long)
    mask[0, :5] = 0     # left padding
    mask[2, :9] = 0
    pos = torch.arange(L).expand(B, L)

    monkeypatch.setenv("POLYRL_PACK_ALIGN", "1")
    out_ref = m(ids, attention_mask=mask, position_ids=pos)
    loss_ref = (out_ref.float() ** 2).mean()
    loss_ref.backward()
    g_ref = m.model.embed_tokens.weight.grad.clone()
    m.zero_grad()

    monkeypatch.setenv("POLYRL_PACK_ALIGN", "64")
    out_pad = m(ids, attention_mask=mask, position_ids=pos)
    assert torch.allclose(out_ref, out_pad, atol=1e-5), \
        (out_ref - out_pad).abs().max()
    loss_pad = (out_pad.float() ** 2).mean()
    loss_pad.backward()
    assert torch.allclose(g_ref, m.model.embed_tokens.weight.grad, atol=1e-5)


def test_pack_pad_to_guards(monkeypatch):
    """pack_pad_to applies only to >=75%-full micros (the dense-packing
    tail must not blow up to the whole budget); env overrides the model
    attribute; POLYRL_PACK_PAD_TO=0 disables."""
    import torch

    from polyrl_amd.models import create_model, get_model_config

    cfg = get_model_config("llama-tiny")
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    trunk = m.model
    trunk.use_remove_padding = True
    trunk.pack_pad_to = 64
    monkeypatch.setenv("POLYRL_PACK_ALIGN", "8")

    # nearly-full micro (48/64 = 75%): padded to 64
    trunk.pack_pad_to = 64
    B, L = 1, 48
    ids = torch.randint(0, cfg.vocab_size, (B, L))
    am = torch.ones(B, L, dtype=torch.long)
    pos = torch.arange(L).expand(B, L)
    # capture Tp via the embed input size
    sizes = []
    orig_embed = trunk.embed_tokens.forward

    def spy_embed(x):
        sizes.append(x.shape[0])
        return orig_embed(x)
    trunk.embed_tokens.forward = spy_embed
    m(ids, attention_mask=am, position_ids=pos)
    assert sizes[-1] == 64, sizes            # padded to budget
    # small tail micro (16/64 < 75%): align-8 only
    ids2 = ids[:, :16]
    am2 = am[:, :16]
    pos2 = pos[:, :16]
    m(ids2, attention_mask=am2, position_ids=pos2)
    assert sizes[-1] == 16, sizes
    # env disable
    monkeypatch.setenv("POLYRL_PACK_PAD_TO", "0")
    m(ids, attention_mask=am, position_ids=pos)
    assert sizes[-1] == 48, sizes

def test_lora_composes_with_remove_padding():
    """LoRA-wrapped linears through the packed varlen path == dense path
    (the _lin router must leave wrapped modules their own forward)."""
    import torch
    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.models.lora import apply_lora
    cfg = get_model_config("llama-debug-cpu")
    torch.manual_seed(21)
    m = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    apply_lora(m, r=4, alpha=8.0)
    # non-zero B so the adapters actually contribute
    for mod in m.modules():
        if hasattr(mod, "lora_B"):
            with torch.no_grad():
                mod.lora_B.normal_(0, 0.05)
    B, L = 2, 12
    ids = torch.randint(0, cfg.vocab_size, (B, L))
    am = torch.ones(B, L, dtype=torch.long)
    am[1, :4] = 0
    with torch.no_grad():
        dense = m(ids, attention_mask=am)
        m.model.use_remove_padding = True
        packed = m(ids, attention_mask=am)
        m.model.use_remove_padding = False
    valid = am.bool()
    err = (dense[valid] - packed[valid]).abs().max().item()
    assert err < 1e-3, err
    # and adapters really changed the output vs the un-adapted model
    torch.manual_seed(21)   # identical base init to m
    m2 = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    with torch.no_grad():
        base = m2(ids, attention_mask=am)
    assert not torch.allclose(base[valid], dense[valid], atol=1e-4)
