"""GEMM layout microbench — quantifies the in-situ trainer layouts.

For every hot trainer linear (llama3-8b shapes, 8192-token micros) time the
three layouts the autograd engine actually issues:

  fwd   (nt): Y  = X  @ W^T      X[M,K] row-major, W[N,K] row-major
  dgrad (nn): dX = dY @ W        dY[M,N] row-major, W[N,K] row-major
  wgrad (tn): dW = dY^T @ X      dY^T is a transposed view

hipBLASLt reached 1.5-1.6 PF on ISOLATED nt shapes (profiles/PROFILES.md)
while the in-trainer average is 0.84 PF — this separates "layout penalty"
from "interference/odd-shape penalty" so the fix targets the right slot.

Run on MI355X:  python profiles/microbench_layouts.py
"""
import torch

# (name, N_out, K_in) for llama3-8b; M = tokens per micro-batch
LINEARS = [
    ("qkv_q", 4096, 4096),
    ("qkv_kv", 1024, 4096),
    ("o_proj", 4096, 4096),
    ("gate_up", 28672, 4096),   # fused gate+up
    ("gate", 14336, 4096),
    ("down", 4096, 14336),
    ("lm_head", 128256, 4096),
]
MS = [8192, 4096]


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3   # us


def m_sweep():
    """Does M%256 alignment alone recover the in-situ GEMM losses?
    In-trainer micros measured M in {6472, 7000, 7216, 7428} at 0.88-0.92
    PF on the [M,14336]x[14336,4096] slots; this sweeps raw vs 256-aligned
    M over the two hot layouts."""
    dev = "cuda"
    shapes = [6472, 6656, 7000, 7168, 7216, 7424, 7428, 7680, 8192]
    N, K = 4096, 14336        # down fwd (nt) / gate dgrad (nn) slot
    print(f"{'M':>6} {'M%256':>6} | {'nt_us':>8} {'nt_TF':>7} | "
          f"{'nn_us':>8} {'nn_TF':>7} | {'tn_us':>8} {'tn_TF':>7}")
    for M in shapes:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
        dy = torch.randn(M, N, dtype=torch.bfloat16, device=dev)
        fl = 2.0 * M * N * K
        t_nt = bench(lambda: x @ w.t())
        t_nn = bench(lambda: dy @ w)
        t_tn = bench(lambda: dy.t() @ x)
        print(f"{M:>6} {M % 256:>6} | {t_nt:8.1f} {fl/t_nt/1e6:7.0f} | "
              f"{t_nn:8.1f} {fl/t_nn/1e6:7.0f} | "
              f"{t_tn:8.1f} {fl/t_tn/1e6:7.0f}")
        # reverse slot: gate fwd (nt) at [M,4096] -> 14336
        w2 = torch.randn(K, N, dtype=torch.bfloat16, device=dev)
        dy2 = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        x2 = torch.randn(M, N, dtype=torch.bfloat16, device=dev)
        t2 = bench(lambda: x2 @ w2.t())
        t2n = bench(lambda: dy2 @ w2)
        print(f"{'':>6} {'rev':>6} | {t2:8.1f} {fl/t2/1e6:7.0f} | "
              f"{t2n:8.1f} {fl/t2n/1e6:7.0f} |")


def main():
    assert torch.cuda.is_available()
    dev = "cuda"
    torch.manual_seed(0)
    m_sweep()
    print(f"{'layer':>8} {'M':>6} {'N':>7} {'K':>6} | "
          f"{'nt_us':>8} {'nt_TF':>7} | {'nn_us':>8} {'nn_TF':>7} | "
          f"{'tn_us':>8} {'tn_TF':>7}")
    for M in MS:
        for name, N, K in LINEARS:
            x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
            w = torch.randn(N, K, dtype=torch.bfloat16, device=dev)
            dy = torch.randn(M, N, dtype=torch.bfloat16, device=dev)
            fl = 2.0 * M * N * K
            t_nt = bench(lambda: x @ w.t())
            t_nn = bench(lambda: dy @ w)
            t_tn = bench(lambda: dy.t() @ x)
            print(f"{name:>8} {M:>6} {N:>7} {K:>6} | "
                  f"{t_nt:8.1f} {fl/t_nt/1e6:7.0f} | "
                  f"{t_nn:8.1f} {fl/t_nn/1e6:7.0f} | "
                  f"{t_tn:8.1f} {fl/t_tn/1e6:7.0f}")


if __name__ == "__main__":
    main()
