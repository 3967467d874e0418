"""Rollout engine on CPU (ref ops): continuous batching, chunked prefill,
abort, logprob capture, weight updates.  Uses the tiny llama config."""
import math

import pytest
import torch

from polyrl_amd.models import create_model, get_model_config
from polyrl_amd.rollout.engine import Engine, SamplingParams


@pytest.fixture(scope="module")
def setup():
    torch.manual_seed(0)
    cfg = get_model_config("llama-debug-cpu")
    model = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    eng = Engine(cfg, device="cpu", dtype=torch.float32,
                 kv_bytes_budget=8 << 20, max_num_batched_tokens=64,
                 max_running_requests=8)
    eng.model.load_state_dict(model.state_dict())
    return cfg, model, eng


def test_generate_batch(setup):
    cfg, model, eng = setup
    torch.manual_seed(1)
    prompts = [torch.randint(0, cfg.vocab_size, (n,)).tolist()
               for n in (5, 12, 3, 20)]
    outs = eng.generate(prompts, SamplingParams(temperature=1.0, max_new_tokens=6),
                        rid_prefix="g")
    assert len(outs) == 4
    for o in outs:
        assert len(o.output_ids) == 6
        assert len(o.output_logprobs) == 6
        assert o.finish_reason == "length"
        assert all(lp <= 0.0 for lp in o.output_logprobs)
    assert not eng.has_work()
    assert eng.kv.free_pages == eng.kv.num_pages  # all pages returned


def test_greedy_matches_model_forward(setup):
    """Greedy engine decode must match the training model's argmax chain
    (same weights, engine runs the kernel path)."""
    cfg, model, eng = setup
    torch.manual_seed(2)
    prompt = torch.randint(0, cfg.vocab_size, (9,)).tolist()
    outs = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=4), "greedy")
    got = outs[0].output_ids
    # reference: step the training model
    ids = list(prompt)
    expect = []
    with torch.no_grad():
        for _ in range(4):
            x = torch.tensor([ids])
            logits = model(x)
            t = int(logits[0, -1].argmax())
            expect.append(t)
            ids.append(t)
    assert got == expect, f"{got} vs {expect}"


def test_engine_logprobs_match_model(setup):
    cfg, model, eng = setup
    torch.manual_seed(3)
    prompt = torch.randint(0, cfg.vocab_size, (7,)).tolist()
    outs = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=3), "lp")
    ids = prompt + outs[0].output_ids
    with torch.no_grad():
        logits = model(torch.tensor([ids]))
        logp = torch.log_softmax(logits[0].float(), -1)
    for j, t in enumerate(outs[0].output_ids):
        expect = logp[len(prompt) - 1 + j, t].item()
        assert abs(outs[0].output_logprobs[j] - expect) < 5e-3


def test_chunked_prefill(setup):
    """Prompt longer than max_num_batched_tokens prefills over several steps
    and still matches greedy."""
    cfg, model, eng = setup
    torch.manual_seed(4)
    prompt = torch.randint(0, cfg.vocab_size, (150,)).tolist()  # > 64 budget
    outs = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=2), "ck")
    ids = list(prompt)
    expect = []
    with torch.no_grad():
        for _ in range(2):
            logits = model(torch.tensor([ids]))
            t = int(logits[0, -1].argmax())
            expect.append(t)
            ids.append(t)
    assert outs[0].output_ids == expect


def test_abort_returns_partial(setup):
    cfg, model, eng = setup
    torch.manual_seed(5)
    prompt = torch.randint(0, cfg.vocab_size, (4,)).tolist()
    eng.add_request("ab-0", prompt, SamplingParams(max_new_tokens=500))
    for _ in range(5):
        eng.step()
    eng.abort_request("ab-0")
    outs = eng.step()
    assert len(outs) == 1
    assert outs[0].finish_reason == "abort"
    assert 0 < len(outs[0].output_ids) < 500
    assert not eng.has_work()


def test_stop_token(setup):
    cfg, model, eng = setup
    torch.manual_seed(6)
    prompt = torch.randint(0, cfg.vocab_size, (6,)).tolist()
    # find the greedy first token, then use it as the stop token
    outs = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=1), "s1")
    stop = outs[0].output_ids[0]
    outs = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=10,
                                                 stop_token_ids=(stop,)), "s2")
    assert outs[0].finish_reason == "stop"
    assert outs[0].output_ids == [stop]


def test_weight_update_changes_output(setup):
    cfg, model, eng = setup
    torch.manual_seed(7)
    prompt = torch.randint(0, cfg.vocab_size, (8,)).tolist()
    out1 = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=3), "w1")
    # perturb weights, push to engine
    model2 = create_model(cfg, kind="actor", dtype="float32", device="cpu")
    eng.model.load_state_dict(model2.state_dict())
    out2 = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=3), "w2")
    assert out1[0].output_ids != out2[0].output_ids  # overwhelmingly likely
    # restore
    eng.model.load_state_dict(model.state_dict())


def test_many_concurrent_requests(setup):
    """more requests than max_running: continuous batching drains the queue"""
    cfg, model, eng = setup
    torch.manual_seed(8)
    prompts = [torch.randint(0, cfg.vocab_size, (4 + i % 5,)).tolist()
               for i in range(20)]
    outs = eng.generate(prompts, SamplingParams(max_new_tokens=3), "mc")
    assert len(outs) == 20
    assert all(len(o.output_ids) == 3 for o in outs)
    assert eng.kv.free_pages == eng.kv.num_pages


def test_group_prefill_sharing_matches_separate(setup):
    """Group API (shared prompt prefill + shared KV pages) must produce
    the same GREEDY outputs as n separate requests, use fewer pages while
    running, and return every page at the end."""
    cfg, model, eng = setup
    torch.manual_seed(9)
    prompt = torch.randint(0, cfg.vocab_size, (13,)).tolist()  # 13 % 16 != 0
    sp = SamplingParams(temperature=0.0, max_new_tokens=5)

    # separate requests (no sharing)
    eng.enable_prefix_sharing = False
    outs_sep = eng.generate([prompt] * 3, sp, "sep")
    eng.enable_prefix_sharing = True

    # group API
    eng.add_request_group("grp", prompt, sp, 3)
    outs = {}
    min_free = eng.kv.num_pages
    while eng.has_work():
        for o in eng.step():
            outs[o.rid] = o
        min_free = min(min_free, eng.kv.free_pages)
    assert len(outs) == 3
    for s in range(3):
        assert outs[f"grp-s{s}"].output_ids == outs_sep[s].output_ids, s
    assert eng.kv.free_pages == eng.kv.num_pages   # refcounts all released
    # shared full prompt pages were not duplicated per child
    sep_pages_at_peak = 3 * ((13 + 5 + 15) // 16 + 1)
    assert min_free >= eng.kv.num_pages - sep_pages_at_peak


def test_group_sampled_outputs_differ(setup):
    """Stochastic group samples draw independently (per-row RNG keys)."""
    cfg, model, eng = setup
    torch.manual_seed(10)
    prompt = torch.randint(0, cfg.vocab_size, (8,)).tolist()
    eng.add_request_group("st", prompt, SamplingParams(temperature=1.0,
                                                       max_new_tokens=6), 4)
    outs = {}
    while eng.has_work():
        for o in eng.step():
            outs[o.rid] = o
    seqs = [tuple(outs[f"st-s{s}"].output_ids) for s in range(4)]
    assert len(outs) == 4
    assert len(set(seqs)) > 1, seqs    # not all identical


def test_group_abort_before_fork_emits_all(setup):
    """A group parent aborted before its fork must emit one aborted output
    per child rid (the coordinator expects n results per group)."""
    cfg, model, eng = setup
    prompt = [3, 4, 5]
    eng.add_request_group("ab", prompt, SamplingParams(max_new_tokens=50), 3)
    eng.abort_request(abort_all=True)
    outs = {}
    while eng.has_work():
        for o in eng.step():
            outs[o.rid] = o
    assert set(outs) == {"ab-s0", "ab-s1", "ab-s2"}
    assert all(o.finish_reason == "abort" for o in outs.values())


def test_release_resume_memory(setup):
    cfg, model, eng = setup
    torch.manual_seed(30)
    prompt = torch.randint(0, cfg.vocab_size, (6,)).tolist()
    eng.generate([prompt], SamplingParams(max_new_tokens=3), "pre")
    eng.release_memory()
    assert eng.kv.k_cache == []
    eng.resume_memory()
    outs = eng.generate([prompt], SamplingParams(temperature=0.0,
                                                 max_new_tokens=3), "post")
    assert len(outs[0].output_ids) == 3
