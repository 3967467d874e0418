"""Round-3 entry point: validate POLYRL_DECODE_BUCKETS on hardware in one
gpurun call — greedy equality vs buckets-off under an EOS-staggered load
(B shrinks as requests finish), plus the capture-count saving.

    python profiles/validate_decode_buckets.py
"""
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def run(buckets: bool):
    os.environ["POLYRL_DECODE_BUCKETS"] = "1" if buckets else "0"
    import torch

    from polyrl_amd.models import create_model, get_model_config
    from polyrl_amd.rollout.engine import Engine, SamplingParams
    cfg = get_model_config("llama-tiny")
    torch.manual_seed(3)
    model = create_model(cfg, kind="actor", dtype="bfloat16", device="cuda")
    eng = Engine(cfg, device="cuda", dtype=torch.bfloat16,
                 kv_bytes_budget=512 << 20, max_running_requests=64)
    eng.model.load_state_dict(model.state_dict())
    torch.manual_seed(4)
    # staggered budgets: finished requests shrink B every few steps —
    # the capture-storm scenario the buckets exist for
    prompts = [torch.randint(0, cfg.vocab_size, (8 + i % 5,)).tolist()
               for i in range(40)]
    for i, p in enumerate(prompts):
        eng.add_request(f"r{i}", p, SamplingParams(
            temperature=0.0, max_new_tokens=4 + (i * 7) % 29))
    outs = {}
    while eng.has_work():
        for o in eng.step():
            outs[o.rid] = o.output_ids
    captures = len(eng._graphs)
    del eng, model
    torch.cuda.empty_cache()
    return outs, captures


def main():
    base, cap_base = run(buckets=False)
    bkt, cap_bkt = run(buckets=True)
    mismatch = [r for r in base if base[r] != bkt[r]]
    print(f"captures: exact-B={cap_base}  bucketed={cap_bkt}")
    print(f"greedy equality: {len(base) - len(mismatch)}/{len(base)} match")
    if mismatch:
        r = mismatch[0]
        print(f"FIRST MISMATCH {r}: base={base[r]} bkt={bkt[r]}")
        raise SystemExit("bucketed decode diverged — do NOT enable")
    assert cap_bkt <= cap_base, "bucketing should not increase captures"
    print("OK: bucketed graphed decode is token-exact; "
          f"capture saving {cap_base} -> {cap_bkt}")


if __name__ == "__main__":
    main()
