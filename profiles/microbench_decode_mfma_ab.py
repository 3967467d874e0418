"""Round-3 entry point: A/B the EXPERIMENTAL MFMA decode kernel against the
production VALU kernel — numerics first, then timing — on the llama/qwen
decode shapes.  Run BOTH halves in one gpurun call:

    POLYRL_DECODE_MFMA=1 python -m pytest tests/test_ops_gpu.py \
        -k decode_mfma -q          # numerics gate (must pass first)
    python profiles/microbench_decode_mfma_ab.py   # timing A/B

The env flag is read ONCE per process (static in attention_decode.hip), so
this script spawns a subprocess per variant.
"""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os, sys, torch
sys.path.insert(0, %(repo)r)
import polyrl_amd.ops as ops

def probe(B, Hq, Hk, D, ctx, page_size=16, iters=50):
    torch.manual_seed(1)
    max_pages = (ctx + page_size - 1) // page_size
    num_pages = B * max_pages + 1
    kc = (torch.randn(num_pages, page_size, Hk, D, device="cuda") / 4).bfloat16()
    vc = torch.randn_like(kc)
    q = (torch.randn(B, Hq, D, device="cuda") / 4).bfloat16()
    pt = torch.arange(B * max_pages, dtype=torch.int32,
                      device="cuda").reshape(B, max_pages)
    cl = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
    scale = D ** -0.5
    for _ in range(5):
        ops.paged_attention_decode(q, kc, vc, pt, cl, scale)
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        ops.paged_attention_decode(q, kc, vc, pt, cl, scale)
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3      # us

rows = []
# (B, Hq, Hk, D, ctx): llama3-8b decode (G=4) + qwen2.5-1.5b (G=6 falls
# back under MFMA — shows the fallback is harmless) + G=8 + D=64 shapes,
# small-B latency-bound and large-B throughput-bound points
for shape in [(16, 32, 8, 128, 2048), (32, 32, 8, 128, 2048),
              (32, 32, 8, 128, 8192), (128, 32, 8, 128, 768),
              (256, 32, 8, 128, 768), (32, 12, 2, 128, 2048),
              (32, 32, 4, 128, 2048), (32, 16, 4, 64, 2048)]:
    rows.append({"shape": shape, "us": probe(*shape)})
print("JSON:" + __import__("json").dumps(rows))
"""


def run(mfma: bool):
    env = dict(os.environ)
    env["POLYRL_DECODE_MFMA"] = "1" if mfma else "0"
    r = subprocess.run([sys.executable, "-c", WORKER % {"repo": REPO}],
                       capture_output=True, text=True, env=env, timeout=900)
    if r.returncode != 0:
        print(r.stdout[-2000:], r.stderr[-2000:])
        raise SystemExit(f"variant mfma={mfma} failed")
    line = [l for l in r.stdout.splitlines() if l.startswith("JSON:")][-1]
    return json.loads(line[5:])


def main():
    base = run(mfma=False)
    mfma = run(mfma=True)
    print(f"{'B,Hq,Hk,D,ctx':>24} {'valu us':>9} {'mfma us':>9} {'speedup':>8}")
    for b, m in zip(base, mfma):
        sp = b["us"] / m["us"]
        print(f"{str(tuple(b['shape'])):>24} {b['us']:9.1f} {m['us']:9.1f} "
              f"{sp:7.2f}x")
    out = {"base": base, "mfma": mfma}
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/decode_mfma_ab.json", "w") as f:
        json.dump(out, f, indent=1)
    print("wrote gpurun_out/decode_mfma_ab.json")


if __name__ == "__main__":
    main()
