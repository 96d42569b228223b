"""Measure hipBLASLt bandwidth on the decode-regime (skinny-M) GEMMs.

At decode the per-step GEMMs are M=conc (16) by the weight shapes of the
flagship models; they are weight-streaming-bound, so report effective
TB/s of W bytes (the only traffic that matters at M=16). This sizes the
headroom for a hand-written skinny GEMM (see skinny_gemm.hip).
"""
import time

import torch

SHAPES = [
    # (name, K, N)  y[M,N] = x[M,K] @ W[N,K]^T   (llama-3-70b per-layer)
    ("qkv   ", 8192, 8192 + 2048),
    ("o     ", 8192, 8192),
    ("gateup", 8192, 57344),
    ("down  ", 28672, 8192),
    ("lmhead", 8192, 128256),
    # mixtral-8x7b dense parts
    ("mx_qkv", 4096, 4096 + 2048),
]


def main(M=16, iters=50):
    if M >= 1024:
        return main_big(M)

    torch.cuda.init()
    dev = "cuda:0"
    print(f"M={M} bf16, {iters} iters, W bytes / time")
    for name, K, N in SHAPES:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        for _ in range(5):
            y = x @ w.t()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            y = x @ w.t()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        gb = N * K * 2 / 1e9
        print(f"{name} K={K:6d} N={N:6d}  {dt*1e6:8.1f} us  "
              f"{gb/dt/1000:6.2f} TB/s")
    del y  # noqa: F841




def main_big(M):
    """Prefill-regime GEMMs: report TFLOP/s vs the 2.5 PF bf16 dense peak."""
    import time
    dev = "cuda:0"
    print(f"M={M} bf16 prefill shapes")
    for name, K, N in SHAPES[:5]:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            y = x @ w.t()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            y = x @ w.t()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        fl = 2.0 * M * N * K
        print(f"{name} K={K:6d} N={N:6d}  {dt*1e3:8.2f} ms  "
              f"{fl/dt/1e12:7.1f} TF  ({fl/dt/25e12:4.1f}% of 2.5PF)")
    del y  # noqa: F841


if __name__ == "__main__":
    import sys
    main(M=int(sys.argv[1]) if len(sys.argv) > 1 else 16)
