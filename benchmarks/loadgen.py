"""AIPerf-style load generator for the OpenAI frontend.

Parity with the reference's benchmarking discipline
(ai-dynamo/dynamo docs/.../llama-3-3-70b-topology.mdx:55-63 AIPerf: fixed
ISL/OSL via min/max tokens, concurrency-driven closed loop;
benchmarks/router prefix-ratio sweeps): generates synthetic prompts of
exact ISL (optionally with a shared prefix ratio to exercise KV-aware
routing), drives fixed concurrency or Poisson arrivals, and reports
output tok/s, TTFT and ITL percentiles.

Usage:
  python benchmarks/loadgen.py --url http://127.0.0.1:8000 \\
      --model mock-model --isl 8192 --osl 1024 --concurrency 64 \\
      --requests 160 [--prefix-ratio 0.5] [--rate 5.0]
"""
from __future__ import annotations

import argparse
import asyncio
import json
import random
import statistics
import sys
import time

import httpx


def pct(xs, p):
    if not xs:
        return None
    xs = sorted(xs)
    i = min(len(xs) - 1, int(p / 100 * len(xs)))
    return xs[i]


async def one_request(client, args, prompt_tokens, stats, session=None,
                      collect=None, osl=None):
    t0 = time.monotonic()
    first = None
    ntok = 0
    last = t0
    itls = []
    body = {"model": args.model, "prompt": prompt_tokens,
            "max_tokens": osl or args.osl, "stream": True,
            "ignore_eos": True}
    if session:
        body["user"] = session    # sticky-session pin
    try:
        async with client.stream("POST", f"{args.url}/v1/completions",
                                 json=body) as r:
            if r.status_code != 200:
                stats["errors"] += 1
                return
            async for line in r.aiter_lines():
                if not line.startswith("data: ") or line == "data: [DONE]":
                    continue
                now = time.monotonic()
                if first is None:
                    first = now
                else:
                    itls.append((now - last) * 1000)
                last = now
                ntok += 1
    except httpx.HTTPError:
        stats["errors"] += 1
        return
    stats["ttft"].append(first - t0 if first else None)
    stats["itl"].extend(itls)
    stats["tokens"] += ntok
    stats["latency"].append(time.monotonic() - t0)
    if collect is not None and first is not None:
        collect.append(first - t0)


async def run_multiturn(args):
    """Multiturn conversations (reference parity: lib/bench multiturn_bench
    TTFT benchmark): each conversation appends its history every turn and
    pins to one worker via sticky sessions, so turns > 1 hit the prefix
    cache and TTFT collapses to the new-chunk prefill."""
    rng = random.Random(args.seed)
    stats = {"ttft": [], "itl": [], "latency": [], "tokens": 0, "errors": 0}
    per_turn = [[] for _ in range(args.turns)]
    chunk = max(1, args.isl // args.turns)
    limits = httpx.Limits(max_connections=args.concurrency + 8)
    async with httpx.AsyncClient(timeout=None, limits=limits) as client:
        sem = asyncio.Semaphore(args.concurrency)

        async def conversation(i):
            crng = random.Random(args.seed * 1000 + i)
            history = []
            async with sem:
                for t in range(args.turns):
                    history = history + [crng.randrange(args.vocab)
                                         for _ in range(chunk)]
                    await one_request(client, args, list(history), stats,
                                      session=f"conv-{i}",
                                      collect=per_turn[t])
                    # the assistant's reply becomes part of the context
                    history = history + [crng.randrange(args.vocab)
                                         for _ in range(args.osl)]

        t0 = time.monotonic()
        await asyncio.gather(*[conversation(i)
                               for i in range(args.requests)])
        wall = time.monotonic() - t0

    out = {
        "conversations": args.requests, "turns": args.turns,
        "errors": stats["errors"], "wall_s": round(wall, 3),
        "output_tok_s": round(stats["tokens"] / wall, 2),
        "ttft_p50_by_turn_s": [round(pct(t, 50), 4) if t else None
                               for t in per_turn],
        "itl_p50_ms": round(pct(stats["itl"], 50), 3) if stats["itl"] else None,
        "config": {"isl_chunk": chunk, "osl": args.osl,
                   "concurrency": args.concurrency},
    }
    print(json.dumps(out))
    return out


async def run_trace(args):
    """Replay a converted trace (tools/trace_convert.py output): each
    record fires at ts_s with its own isl/osl; records sharing a
    prefix_group share a prompt prefix of prefix_len tokens (the
    reference's mooncake-trace replay discipline)."""
    import json as js
    rng = random.Random(args.seed)
    vocab = args.vocab
    recs = [js.loads(ln) for ln in open(args.trace) if ln.strip()]
    prefixes = {}

    def make_prompt(rec):
        isl = max(1, rec.get("isl") or args.isl)
        plen = min(int(rec.get("prefix_len") or 0), isl - 1)
        g = rec.get("prefix_group")
        pre = []
        if g is not None and plen > 0:
            key = (g, plen)
            if key not in prefixes:
                grng = random.Random(hash((args.seed, g)) & 0xFFFFFFFF)
                prefixes[key] = [grng.randrange(vocab) for _ in range(plen)]
            pre = prefixes[key]
        return pre + [rng.randrange(vocab) for _ in range(isl - len(pre))]

    stats = {"ttft": [], "itl": [], "latency": [], "tokens": 0, "errors": 0}
    limits = httpx.Limits(max_connections=args.concurrency + 8)
    async with httpx.AsyncClient(timeout=None, limits=limits) as client:
        sem = asyncio.Semaphore(args.concurrency)
        t0 = time.monotonic()

        async def fire(rec):
            delay = rec["ts_s"] - (time.monotonic() - t0)
            if delay > 0:
                await asyncio.sleep(delay)
            async with sem:
                await one_request(client, args, make_prompt(rec), stats,
                                  osl=max(1, rec.get("osl") or args.osl))

        await asyncio.gather(*[fire(r) for r in recs])
        wall = time.monotonic() - t0

    ttfts = [t for t in stats["ttft"] if t is not None]
    out = {
        "requests": len(recs),
        "errors": stats["errors"],
        "wall_s": round(wall, 3),
        "output_tok_s": round(stats["tokens"] / wall, 2),
        "ttft_p50_s": round(pct(ttfts, 50), 4) if ttfts else None,
        "ttft_p95_s": round(pct(ttfts, 95), 4) if ttfts else None,
        "itl_p50_ms": round(pct(stats["itl"], 50), 3) if stats["itl"] else None,
        "config": {"trace": args.trace, "concurrency": args.concurrency},
    }
    print(json.dumps(out))
    return out


async def run(args):
    rng = random.Random(args.seed)
    vocab = args.vocab
    shared_prefix = [rng.randrange(vocab)
                     for _ in range(int(args.isl * args.prefix_ratio))]

    def make_prompt():
        own = [rng.randrange(vocab) for _ in range(args.isl - len(shared_prefix))]
        return shared_prefix + own

    stats = {"ttft": [], "itl": [], "latency": [], "tokens": 0, "errors": 0}
    limits = httpx.Limits(max_connections=args.concurrency + 8)
    async with httpx.AsyncClient(timeout=None, limits=limits) as client:
        sem = asyncio.Semaphore(args.concurrency)

        async def bounded(i):
            if args.rate > 0:  # open loop: Poisson arrivals
                await asyncio.sleep(rng.expovariate(args.rate) * i / max(i, 1))
            async with sem:
                await one_request(client, args, make_prompt(), stats)

        t0 = time.monotonic()
        if args.rate > 0:
            tasks = []
            t = 0.0
            for i in range(args.requests):
                t += rng.expovariate(args.rate)

                async def delayed(d=t, i=i):
                    await asyncio.sleep(d)
                    async with sem:
                        await one_request(client, args, make_prompt(), stats)
                tasks.append(asyncio.create_task(delayed()))
            await asyncio.gather(*tasks)
        else:
            await asyncio.gather(*[bounded(i) for i in range(args.requests)])
        wall = time.monotonic() - t0

    ttfts = [t for t in stats["ttft"] if t is not None]
    out = {
        "requests": args.requests,
        "errors": stats["errors"],
        "wall_s": round(wall, 3),
        "output_tok_s": round(stats["tokens"] / wall, 2),
        "ttft_p50_s": round(pct(ttfts, 50), 4) if ttfts else None,
        "ttft_p95_s": round(pct(ttfts, 95), 4) if ttfts else None,
        "itl_p50_ms": round(pct(stats["itl"], 50), 3) if stats["itl"] else None,
        "itl_p95_ms": round(pct(stats["itl"], 95), 3) if stats["itl"] else None,
        "latency_p50_s": round(pct(stats["latency"], 50), 3)
        if stats["latency"] else None,
        "config": {"isl": args.isl, "osl": args.osl,
                   "concurrency": args.concurrency,
                   "prefix_ratio": args.prefix_ratio, "rate": args.rate},
    }
    print(json.dumps(out))
    return out


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--url", default="http://127.0.0.1:8000")
    p.add_argument("--model", default="")
    p.add_argument("--isl", type=int, default=8192)
    p.add_argument("--osl", type=int, default=1024)
    p.add_argument("--concurrency", type=int, default=16)
    p.add_argument("--requests", type=int, default=32)
    p.add_argument("--prefix-ratio", type=float, default=0.0)
    p.add_argument("--turns", type=int, default=0,
                   help="multiturn mode: turns per conversation (0 = off)")
    p.add_argument("--rate", type=float, default=0.0,
                   help=">0: open-loop Poisson req/s; 0: closed loop")
    p.add_argument("--vocab", type=int, default=512)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--trace", default=None,
                   help="replay a converted trace file "
                        "(tools/trace_convert.py output)")
    args = p.parse_args()
    if args.trace:
        asyncio.run(run_trace(args))
    elif args.turns > 0:
        asyncio.run(run_multiturn(args))
    else:
        asyncio.run(run(args))


if __name__ == "__main__":
    main()
