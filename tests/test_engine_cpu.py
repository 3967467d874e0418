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


# ------------------------------------------------------- radix prefix cache


def _fresh_engine(cfg, model, kv_bytes=8 << 20, radix=True, **kw):
    e = Engine(cfg, device="cpu", dtype=torch.float32,
               kv_bytes_budget=kv_bytes, enable_radix_cache=radix, **kw)
    e.model.load_state_dict(model.state_dict())
    return e


def test_radix_cache_hits_and_greedy_equality(setup):
    """Cross-request prefix reuse (SGLang radix-cache capability,
    SURVEY.md §2.2.2): a shared 40-token system prompt is prefilled once;
    later requests seed their page table from the trie and skip it.
    Greedy outputs must equal the no-cache engine's exactly."""
    cfg, model, _ = setup
    sys_prefix = list(range(100, 140))
    p1, p2 = sys_prefix + [7, 8, 9], sys_prefix + [21, 22]
    sp = SamplingParams(temperature=0.0, max_new_tokens=6)
    e0 = _fresh_engine(cfg, model, radix=False)
    ref = {i: e0.generate([p], sp, f"r{i}")[0].output_ids
           for i, p in enumerate((p1, p2))}
    e = _fresh_engine(cfg, model)
    o1 = e.generate([p1], sp, "a")[0]
    assert e.radix.num_nodes > 0          # finished seq donated its pages
    h0 = e.radix.hit_tokens
    o2 = e.generate([p2], sp, "b")[0]
    assert e.radix.hit_tokens - h0 >= 32  # 2 full pages of the shared prefix
    assert o1.output_ids == ref[0] and o2.output_ids == ref[1]
    # generated tokens are cached too (multi-turn / continuation reuse):
    # re-asking p1 hits its full cached pages
    h1 = e.radix.hit_tokens
    assert e.generate([p1], sp, "c")[0].output_ids == ref[0]
    assert e.radix.hit_tokens - h1 >= 32
    e.flush_radix()
    assert e.radix.num_nodes == 0
    assert e.kv.free_pages == e.kv.num_pages   # accounting closes


def test_radix_eviction_under_page_pressure(setup):
    """A tiny KV pool forces LRU leaf eviction: new work must still
    complete, and accounting must close when the tree is flushed."""
    cfg, model, _ = setup
    # ~24 pages total: each 40+6-token request needs 3; cache grows until
    # the allocator runs dry and eviction kicks in
    from polyrl_amd.rollout.kv_cache import PagedKVCache
    bt = PagedKVCache.bytes_per_token(cfg.num_hidden_layers,
                                      cfg.num_key_value_heads, cfg.head_dim)
    e = _fresh_engine(cfg, model, kv_bytes=bt * 16 * 24)
    assert e.kv.num_pages <= 32
    sp = SamplingParams(temperature=0.0, max_new_tokens=6)
    e0 = _fresh_engine(cfg, model, radix=False, kv_bytes=bt * 16 * 24)
    for i in range(12):
        p = [(1000 + 37 * i + j) % cfg.vocab_size for j in range(40)]
        out = e.generate([p], sp, f"q{i}")[0]
        refo = e0.generate([p], sp, f"q{i}")[0]
        assert out.output_ids == refo.output_ids, i
        assert out.finish_reason == "length"
    e.flush_radix()
    assert e.kv.free_pages == e.kv.num_pages


def test_radix_composes_with_group_fork(setup):
    """Group prefix-sharing (n samples) on top of a radix-cached system
    prompt: greedy children equal the separate-request reference."""
    cfg, model, _ = setup
    sys_prefix = list(range(200, 240))
    prompt = sys_prefix + [3, 4, 5]
    sp = SamplingParams(temperature=0.0, max_new_tokens=5)
    e0 = _fresh_engine(cfg, model, radix=False)
    ref = e0.generate([prompt], sp, "r")[0].output_ids
    e = _fresh_engine(cfg, model)
    e.generate([sys_prefix + [99]], sp, "warm")      # populate the trie
    h0 = e.radix.hit_tokens
    e.add_request_group("grp", prompt, sp, 3)
    outs = {}
    while e.has_work():
        for o in e.step():
            outs[o.rid] = o
    assert e.radix.hit_tokens > h0                   # parent hit the prefix
    for s in range(3):
        assert outs[f"grp-s{s}"].output_ids == ref
    assert not e.has_work()


def test_radix_flushed_on_weight_update(setup):
    """Weight swap invalidates cached KV (the reference flushes SGLang's
    cache after /update_weights — patches.py:360-387)."""
    from polyrl_amd.rollout.runner import EngineRunner
    cfg, model, _ = setup
    e = _fresh_engine(cfg, model)
    sp = SamplingParams(temperature=0.0, max_new_tokens=4)
    e.generate([list(range(300, 340))], sp, "w")
    assert e.radix.num_nodes > 0
    runner = EngineRunner(e)
    sd = {k: v + 0.01 for k, v in model.state_dict().items()}
    runner.update_weights(sd, version=1)
    assert e.radix.num_nodes == 0


def test_radix_hit_with_chunked_prefill_tail(setup):
    """Radix-seeded request whose uncached tail spans multiple prefill
    chunks (max_num_batched_tokens < tail length): greedy equality holds
    (the chunked-prefill path attends to cached + freshly-written history)."""
    cfg, model, _ = setup
    sys_prefix = list(range(50, 82))          # 2 full pages
    tail = [(7 * j + 3) % cfg.vocab_size for j in range(40)]
    prompt = sys_prefix + tail
    sp = SamplingParams(temperature=0.0, max_new_tokens=5)
    e0 = _fresh_engine(cfg, model, radix=False)
    ref = e0.generate([prompt], sp, "r")[0].output_ids
    # chunk budget 16 => the 40-token tail prefills over 3 chunks
    e = _fresh_engine(cfg, model, max_num_batched_tokens=16)
    e.generate([sys_prefix + [1, 2]], sp, "warm")
    h0 = e.radix.hit_tokens
    out = e.generate([prompt], sp, "q")[0]
    assert e.radix.hit_tokens - h0 >= 32
    assert out.output_ids == ref


def test_radix_abort_mid_prefill_accounting(setup):
    """Abort a radix-seeded request before its prefill completes: pages
    (cached + fresh) must all return / stay cache-owned — accounting closes
    after a flush."""
    cfg, model, _ = setup
    sys_prefix = list(range(50, 82))
    prompt = sys_prefix + [(11 * j + 1) % cfg.vocab_size for j in range(40)]
    sp = SamplingParams(temperature=0.0, max_new_tokens=5)
    e = _fresh_engine(cfg, model, max_num_batched_tokens=16)
    e.generate([sys_prefix + [9]], sp, "warm")
    e.add_request("victim", prompt, sp)
    e.step()                                   # admit + first chunk only
    e.abort_request("victim")
    while e.has_work():
        e.step()
    e.flush_radix()
    assert e.kv.free_pages == e.kv.num_pages


def test_kv_shrink_seq_rollback_accounting(setup):
    """shrink_seq rolls back abandoned chunk growth exactly: length and
    trailing pages return; shared pages survive via refcount."""
    from polyrl_amd.rollout.kv_cache import PagedKVCache
    kv = PagedKVCache(1, 1, 8, num_pages=8, page_size=4, device="cpu")
    assert kv.allocate(0, 6)           # 2 pages, len 6
    free0 = kv.free_pages
    assert kv.allocate(0, 7)           # grow to 13 -> 4 pages
    kv.shrink_seq(0, 7)                # roll the growth back
    assert kv.seq_len(0) == 6
    assert kv.free_pages == free0
    kv.free_seq(0)
    assert kv.free_pages == kv.num_pages


def test_radix_correct_after_mixed_chunk_rollback(setup):
    """Page pressure during chunked decode triggers the mixed-allocation
    rollback; with the radix cache on, later identical prompts must still
    decode EXACTLY like a fresh engine (an inflated _seq_len would donate
    never-written KV)."""
    cfg, model, _ = setup
    from polyrl_amd.rollout.kv_cache import PagedKVCache
    bt = PagedKVCache.bytes_per_token(cfg.num_hidden_layers,
                                      cfg.num_key_value_heads, cfg.head_dim)
    sp = SamplingParams(temperature=0.0, max_new_tokens=24)
    prompts = [[(59 * i + j) % cfg.vocab_size for j in range(20)]
               for i in range(6)]
    e0 = _fresh_engine(cfg, model, radix=False, kv_bytes=bt * 16 * 1024)
    refs = [e0.generate([p], sp, f"r{i}")[0].output_ids
            for i, p in enumerate(prompts)]
    # tiny pool + chunked decode: concurrent requests exhaust pages mid-chunk
    e = _fresh_engine(cfg, model, kv_bytes=bt * 16 * 22,
                      decode_chunk_size=8, max_num_batched_tokens=64)
    for rep in range(2):               # second round hits the radix
        for i, p in enumerate(prompts):
            e.add_request(f"x{rep}-{i}", p, sp)
        outs = {}
        while e.has_work():
            for o in e.step():
                outs[o.rid] = o
        for i in range(6):
            o = outs[f"x{rep}-{i}"]
            if o.finish_reason == "length":       # survived the pressure
                assert o.output_ids == refs[i], (rep, i)
    e.flush_radix()
    assert e.kv.free_pages == e.kv.num_pages


def test_max_model_len_fence(setup):
    """A request whose prompt+response would exceed max_model_len stops at
    the fence with finish_reason='length' (never writes past the KV
    geometry the hipGraph page tables are sized for)."""
    cfg, model, _ = setup
    e = Engine(cfg, device="cpu", dtype=torch.float32,
               kv_bytes_budget=8 << 20, max_model_len=24)
    e.model.load_state_dict(model.state_dict())
    out = e.generate([list(range(7, 27))],          # 20-token prompt
                     SamplingParams(temperature=0.0, max_new_tokens=50),
                     "fence")[0]
    assert out.finish_reason == "length"
    assert len(out.output_ids) <= 24 - 20 + 1
    assert e.kv.free_pages == e.kv.num_pages


# ----------------------------------------------- round-2 ADVICE regressions


def test_release_memory_drops_captured_graphs(setup):
    """captured decode graphs hold device pointers into
    the KV tensors; release_memory()/resume_memory() must drop them so
    replays never touch freed memory."""
    cfg, model, _ = setup
    eng = _fresh_engine(cfg, model, radix=False)
    eng._graphs[4] = {"fake": "capture state"}
    eng._graph_pool = object()
    eng.release_memory()
    assert eng._graphs == {}
    assert eng._graph_pool is None
    eng.resume_memory()
    outs = eng.generate([[1, 2, 3]], SamplingParams(temperature=0.0,
                                                    max_new_tokens=4), "rr")
    assert len(outs[0].output_ids) == 4


def test_aborted_request_does_not_donate_radix(setup):
    """an aborted request's KV pages hold PRE-update
    weights' KV (it is emitted on the step AFTER flush_radix), so it must
    not insert into the radix trie — a token-exact continuation would
    otherwise radix-match stale KV and skip recomputing under the new
    weights."""
    cfg, model, _ = setup
    eng = _fresh_engine(cfg, model, radix=True, decode_chunk_size=4)
    prompt = list(range(40, 80))           # 40 tokens, page-aligned span
    eng.add_request("a0", prompt, SamplingParams(temperature=0.0,
                                                 max_new_tokens=32))
    for _ in range(3):                     # prefill + 2 decode chunks
        eng.step()
    assert eng.num_running() == 1
    eng.abort_request(rid="a0")
    outs = []
    while eng.has_work():
        outs.extend(eng.step())
    assert outs and outs[0].finish_reason == "abort"
    assert len(outs[0].output_ids) > 0     # partial output captured
    # the aborted prefix must NOT be radix-matchable
    pages, matched = eng.radix.match(prompt + outs[0].output_ids)
    assert matched == 0, matched
    if pages:
        eng.kv.unref_pages(pages)
    # control: a NORMALLY finished request does donate
    eng.add_request("b0", prompt, SamplingParams(temperature=0.0,
                                                 max_new_tokens=8))
    while eng.has_work():
        eng.step()
    pages2, matched2 = eng.radix.match(prompt + [0])
    assert matched2 > 0
    eng.kv.unref_pages(pages2)


def test_decode_bucket_table():
    """Bucketed graphed-decode helper (EXPERIMENTAL round-3 path): buckets
    are monotone, cover 1..256 with <=50% padding waste, and B beyond the
    table maps to itself."""
    from polyrl_amd.rollout.engine import DECODE_BUCKETS, decode_bucket
    assert list(DECODE_BUCKETS) == sorted(set(DECODE_BUCKETS))
    for b in range(1, 257):
        bb = decode_bucket(b)
        assert bb >= b and bb in DECODE_BUCKETS
        assert bb <= b * 3 / 2, (b, bb)   # bounded padding waste
    assert decode_bucket(300) == 300          # beyond table: exact


def test_decode_scratch_slot_reservation(setup):
    """The dummy-row KV sink: reserved once, stable across calls, page
    zeroed, re-reserved after the KV cache is rebuilt (release/resume)."""
    _, _, eng = setup
    s1 = eng._decode_scratch_slot()
    s2 = eng._decode_scratch_slot()
    assert s1 == s2                       # idempotent
    page = s1 // eng.kv.page_size
    assert eng.kv.k_cache[0][page].abs().sum().item() == 0.0
    free_before = eng.kv.free_pages
    eng._decode_scratch_slot()
    assert eng.kv.free_pages == free_before   # no double reservation
    # release so later tests on this module-scoped engine see all pages
    eng.kv.free_seq(eng._SCRATCH_SEQ)
    assert eng.kv.free_pages == free_before + 1


def test_decode_scratch_survives_release_resume(setup):
    """release_memory/resume_memory rebuild the KV cache; the scratch
    reservation must re-reserve lazily on the fresh cache (the seq maps
    reset) instead of handing out a stale slot."""
    _, _, eng = setup
    s1 = eng._decode_scratch_slot()
    assert eng.kv.seq_len(eng._SCRATCH_SEQ) == 1
    eng.release_memory()
    eng.resume_memory()
    assert eng.kv.seq_len(eng._SCRATCH_SEQ) == 0    # fresh cache: no seq
    s2 = eng._decode_scratch_slot()
    assert eng.kv.seq_len(eng._SCRATCH_SEQ) == 1
    page = s2 // eng.kv.page_size
    assert 0 <= page < eng.kv.num_pages
    assert eng.kv.k_cache[0][page].abs().sum().item() == 0.0
    eng.kv.free_seq(eng._SCRATCH_SEQ)               # leave the engine clean
