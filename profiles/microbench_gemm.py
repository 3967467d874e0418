"""Round-2 microbench for the experimental 256^2 MFMA GEMM
(ops/csrc/experimental/gemm_bf16.hip).  NOT a pytest file — run manually on
an MI355X:

    timeout 240 python profiles/microbench_gemm.py

Refchecks at 512^2 x512 vs torch.matmul (bf16 tolerance), then A/B-times the
bench's trainer GEMM shapes vs hipBLASLt (torch.matmul).  The kernel is the
COARSE-sync variant (one barrier + vmcnt(0) per K-tile); the guide's fine
8-phase interleave is the follow-up if this lands within ~25% of the 1.3 PF
template number.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
assert torch.cuda.is_available(), "GPU required"

d = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                 "polyrl_amd", "ops", "csrc", "experimental")
so = os.path.join(d, "polyrl_gemm_exp.so")
if os.path.exists(so):           # prebuilt in-tree (travels to the GPU box)
    import importlib.util
    spec = importlib.util.spec_from_file_location("polyrl_gemm_exp", so)
    ext = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(ext)
else:                            # fallback: JIT build on the box
    from torch.utils.cpp_extension import load
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    ext = load(name="polyrl_gemm_exp",
               sources=[os.path.join(d, "gemm_binding.hip"),
                        os.path.join(d, "gemm_bf16.hip")],
               extra_cuda_cflags=["-O3"], verbose=False)

def refcheck(M, N, K, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    A = (torch.rand((M, K), generator=g, device="cuda") * 2 - 1).to(torch.bfloat16)
    W = (torch.rand((N, K), generator=g, device="cuda") * 2 - 1).to(torch.bfloat16)
    C = ext.gemm_nt(A, W)
    ref = (A.float() @ W.float().T)
    err = (C.float() - ref).abs()
    rel = (err / ref.abs().clamp(min=1.0)).max().item()
    print(f"refcheck {M}x{N}x{K}: max rel err {rel:.3e}")
    assert rel < 2e-2, f"NUMERICS FAIL {rel}"

def bench(M, N, K, iters=20):
    A = torch.randn((M, K), device="cuda", dtype=torch.bfloat16)
    W = torch.randn((N, K), device="cuda", dtype=torch.bfloat16)
    torch.cuda.synchronize()
    # interleaved A/B (guide rule: within-probe comparison)
    def run(fn):
        fn(); torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters): fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters
    t_ours, t_blas = [], []
    for _ in range(3):
        t_ours.append(run(lambda: ext.gemm_nt(A, W)))
        t_blas.append(run(lambda: A @ W.T))
    tf = lambda t: 2.0 * M * N * K / t / 1e12
    print(f"{M}x{N}x{K}: ours {tf(min(t_ours)):7.1f} TF | "
          f"hipBLASLt {tf(min(t_blas)):7.1f} TF | "
          f"speedup {min(t_blas)/min(t_ours):.2f}x")

def refcheck_decode(N, K, ksplit, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    x = (torch.rand((256, K), generator=g, device="cuda") * 2 - 1).to(torch.bfloat16)
    W = (torch.rand((N, K), generator=g, device="cuda") * 2 - 1).to(torch.bfloat16)
    C = ext.gemm_decode(x, W, ksplit)
    ref = x.float() @ W.float().T
    rel = ((C - ref).abs() / ref.abs().clamp(min=1.0)).max().item()
    print(f"decode refcheck 256x{N}x{K} ks={ksplit}: max rel err {rel:.3e}")
    assert rel < 2e-2, f"DECODE NUMERICS FAIL {rel}"


def bench_decode(N, K, ksplit, iters=50):
    x = torch.randn((256, K), device="cuda", dtype=torch.bfloat16)
    W = torch.randn((N, K), device="cuda", dtype=torch.bfloat16)
    def run(fn):
        fn(); torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters): fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters
    t_ours, t_blas = [], []
    for _ in range(3):
        t_ours.append(run(lambda: ext.gemm_decode(x, W, ksplit)))
        t_blas.append(run(lambda: x @ W.T))
    # weight-read bound: N*K*2 bytes at ~6.3 TB/s achievable HBM
    bound_us = N * K * 2 / 6.3e12 * 1e6
    print(f"decode 256x{N}x{K} ks={ksplit}: ours {min(t_ours)*1e6:7.1f} us "
          f"| hipBLASLt {min(t_blas)*1e6:7.1f} us | weight-read bound "
          f"{bound_us:6.1f} us | ours/bound {min(t_ours)*1e6/bound_us:.2f}x")


if __name__ == "__main__":
    import sys as _sys
    if "--decode-only" not in _sys.argv:
        refcheck(512, 512, 512)
        refcheck(512, 512, 128)
        refcheck(2048, 1024, 4096, seed=1)
        for (m, n, k) in [(8192, 4096, 4096), (8192, 14336, 4096),
                          (8192, 4096, 14336), (8192, 6144, 4096),
                          (4096, 4096, 4096), (8192, 8192, 8192)]:
            bench(m, n, k)
    # decode regime (M=256): llama-3-8B projections + vocab head
    refcheck_decode(512, 512, 1)
    refcheck_decode(512, 512, 4)
    refcheck_decode(1024, 4096, 8, seed=1)
    for (n, k, ks) in [(4096, 4096, 8), (6144, 4096, 8), (14336, 4096, 8),
                       (4096, 14336, 16), (128256, 4096, 2)]:
        bench_decode(n, k, ks)
    print("MICROBENCH DONE")
