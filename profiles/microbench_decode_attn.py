"""Paged decode attention probe: time + roofline across context lengths.

    python profiles/microbench_decode_attn.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

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
        o = ops.paged_attention_decode(q, kc, vc, pt, cl, scale)
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        o = ops.paged_attention_decode(q, kc, vc, pt, cl, scale)
    e.record()
    torch.cuda.synchronize()
    us = s.elapsed_time(e) / iters * 1e3
    kv_bytes = 2 * B * ctx * Hk * D * 2          # K+V read once
    bound_us = kv_bytes / 8e12 * 1e6
    print(f"B={B:4d} Hq={Hq} Hk={Hk} ctx={ctx:5d}: {us:8.1f} us "
          f"(KV {kv_bytes/2**20:7.1f} MiB, bound {bound_us:6.1f} us, "
          f"{us/bound_us:4.1f}x)")


if __name__ == "__main__":
    assert torch.cuda.is_available()
    for ctx in (128, 384, 1024, 4096):
        probe(128, 32, 8, 128, ctx)
    probe(32, 32, 8, 128, 4096)
    probe(32, 32, 8, 128, 8192)
    probe(256, 28, 4, 128, 1024)     # qwen2.5-7b shape
