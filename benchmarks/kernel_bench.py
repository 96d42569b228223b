"""Microbenchmarks for the native kernels (run on an MI355X box).

Prints per-kernel achieved bandwidth / TFLOPs vs the hardware ceilings
(HBM ~6.3 TB/s achievable, bf16 MFMA 2.5 PF dense)."""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from dynamo_amd import ops  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / iters


def bench_rmsnorm(rows=8192, D=8192):
    x = torch.randn(rows, D, dtype=torch.bfloat16, device="cuda")
    w = torch.ones(D, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: ops.rmsnorm(x, w, 1e-5))
    gb = 2 * rows * D * 2 / 1e9
    print(f"rmsnorm      [{rows}x{D}]: {t*1e6:8.1f} us  {gb/t:7.2f} GB/s")


def bench_silu_mul(rows=8192, I=14336):
    x = torch.randn(rows, 2 * I, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: ops.silu_mul(x))
    gb = 3 * rows * I * 2 / 1e9
    print(f"silu_mul     [{rows}x{I}]: {t*1e6:8.1f} us  {gb/t:7.2f} GB/s")


def bench_decode_attn(B=16, ctx=8192, Hq=64, Hkv=8, ps=64):
    hd = 128
    npages = B * ((ctx + ps - 1) // ps)
    kc = torch.randn(npages, Hkv, ps, hd, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    pt = torch.arange(npages, dtype=torch.int32, device="cuda").view(B, -1)
    q = torch.randn(B, Hq, hd, dtype=torch.bfloat16, device="cuda")
    ctxl = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
    scratch = ops.DecodeScratch(B, Hq, hd, ctx, "cuda")
    t = timeit(lambda: ops.paged_attention_decode(
        q, kc, vc, pt, ctxl, hd ** -0.5, scratch))
    vct = vc.permute(0, 1, 3, 2).contiguous()
    tv = timeit(lambda: ops.paged_attention_decode(
        q, kc, vct, pt, ctxl, hd ** -0.5, scratch, v_transposed=True))
    gb = 2 * B * ctx * Hkv * hd * 2 / 1e9  # K+V bytes
    print(f"decode_attn  [B{B} ctx{ctx} Hq{Hq}/{Hkv}]: {t*1e6:8.1f} us  "
          f"{gb/t:7.2f} GB/s | vt {tv*1e6:8.1f} us {gb/tv:7.2f} GB/s")


def bench_prefill_attn(S=8192, Hq=32, Hkv=8, ps=64, v_transposed=False):
    hd = 128
    npages = (S + ps - 1) // ps
    kc = torch.randn(npages, Hkv, ps, hd, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn_like(kc)
    pt = torch.arange(npages, dtype=torch.int32, device="cuda").view(1, -1)
    q = torch.randn(S, Hq, hd, dtype=torch.bfloat16, device="cuda")
    starts = torch.tensor([0], dtype=torch.int32, device="cuda")
    qlen = torch.tensor([S], dtype=torch.int32, device="cuda")
    ctxl = torch.tensor([S], dtype=torch.int32, device="cuda")
    tiles = ops.build_prefill_tiles([S], "cuda",
                                    ops.prefill_tile_rows(Hq, Hkv))
    if v_transposed:
        vc = vc.permute(0, 1, 3, 2).contiguous()
    t = timeit(lambda: ops.attention_prefill_paged(
        q, kc, vc, pt, starts, qlen, ctxl, hd ** -0.5, tiles,
        v_transposed=v_transposed), iters=5)
    # causal flops: 2 gemms * 2*S*S/2*hd per head
    fl = 2 * 2 * Hq * (S * S / 2) * hd
    print(f"prefill_attn [S{S} Hq{Hq}/{Hkv} vt{int(v_transposed)}]: "
          f"{t*1e3:8.2f} ms  {fl/t/1e12:7.1f} TFLOP/s (causal)")


def bench_kv_append(T=8192, Hkv=8, ps=64):
    hd = 128
    npages = (T + ps - 1) // ps
    kc = torch.zeros(npages, Hkv, ps, hd, dtype=torch.bfloat16, device="cuda")
    vc = torch.zeros_like(kc)
    k = torch.randn(T, Hkv, hd, dtype=torch.bfloat16, device="cuda")
    v = torch.randn_like(k)
    slots = torch.arange(T, dtype=torch.int64, device="cuda")
    t = timeit(lambda: ops.kv_cache_append(kc, vc, k, v, slots))
    gb = 4 * T * Hkv * hd * 2 / 1e9
    print(f"kv_append    [T{T}]: {t*1e6:8.1f} us  {gb/t:7.2f} GB/s")


def bench_sampling(B=16, V=128256):
    logits = torch.randn(B, V, device="cuda")
    t = timeit(lambda: ops.greedy_sample(logits))
    print(f"greedy       [B{B} V{V}]: {t*1e6:8.1f} us")


def bench_gemm(M=16, K=8192, N=8192):
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    t = timeit(lambda: torch.nn.functional.linear(a, w))
    fl = 2 * M * N * K
    gb = (M * K + N * K + M * N) * 2 / 1e9
    print(f"gemm(blaslt) [{M}x{K}x{N}]: {t*1e6:8.1f} us  {fl/t/1e12:7.1f} TF  "
          f"{gb/t:7.2f} GB/s")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--which", default="all")
    a = ap.parse_args()
    torch.manual_seed(0)
    w = a.which
    if w in ("all", "rmsnorm"):
        bench_rmsnorm()
    if w in ("all", "silu"):
        bench_silu_mul()
    if w in ("all", "decode"):
        bench_decode_attn()
        bench_decode_attn(B=16, ctx=8192, Hq=32, Hkv=8)
    if w in ("all", "prefill"):
        bench_prefill_attn()
        bench_prefill_attn(v_transposed=True)
        bench_prefill_attn(S=8192, Hq=64)
        bench_prefill_attn(S=8192, Hq=64, v_transposed=True)
        bench_prefill_attn(S=2048)
        bench_prefill_attn(Hq=64)          # GQA 8 (llama-70b TP1)
        bench_prefill_attn(S=2048, Hq=64)
    if w in ("all", "append"):
        bench_kv_append()
    if w in ("all", "sample"):
        bench_sampling()
    if w in ("all", "gemm"):
        bench_gemm()
        bench_gemm(M=8192)
