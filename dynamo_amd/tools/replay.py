"""Replay a recorded request stream against a running frontend.

Reference parity: the reference's request recorder (lib/llm recorder.rs)
and trace-replay component (components/src/dynamo/replay). Records come
from `python -m dynamo_amd.frontend --record FILE` (JSONL, one request per
line with its original arrival timestamp and response chunks).

  python -m dynamo_amd.tools.replay FILE --url http://127.0.0.1:8000 \
      [--speed 2.0] [--max-requests N]

Requests are re-issued over /v1/completions with the recorded token ids
and sampling, preserving recorded inter-arrival gaps scaled by --speed.
Prints per-request latency and an aggregate summary.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import sys
import time


async def _one(client, url: str, rec: dict, results: list):
    body = {
        "model": rec.get("model", ""),
        "prompt": rec["token_ids"],
        "max_tokens": rec.get("stop", {}).get("max_tokens", 128),
        "temperature": rec.get("sampling", {}).get("temperature", 0.0),
        "top_p": rec.get("sampling", {}).get("top_p", 1.0),
        "top_k": rec.get("sampling", {}).get("top_k", 0),
        "seed": rec.get("sampling", {}).get("seed", 0),
        "ignore_eos": rec.get("stop", {}).get("ignore_eos", False),
    }
    t0 = time.monotonic()
    r = await client.post(url + "/v1/completions", json=body)
    dt = time.monotonic() - t0
    ok = r.status_code == 200
    ntok = (r.json()["usage"]["completion_tokens"] if ok else 0)
    results.append((ok, dt, ntok, rec.get("latency_s")))


async def replay(path: str, url: str, speed: float, max_requests: int):
    import httpx
    recs = []
    with open(path) as fh:
        for line in fh:
            line = line.strip()
            if line:
                recs.append(json.loads(line))
    if max_requests:
        recs = recs[:max_requests]
    if not recs:
        print("no records", file=sys.stderr)
        return 1
    base_ts = recs[0].get("ts", 0.0)
    results: list = []
    tasks = []
    start = time.monotonic()
    async with httpx.AsyncClient(timeout=120) as client:
        for rec in recs:
            delay = max(0.0, (rec.get("ts", base_ts) - base_ts) / speed)
            now = time.monotonic() - start
            if delay > now:
                await asyncio.sleep(delay - now)
            tasks.append(asyncio.create_task(_one(client, url, rec, results)))
        await asyncio.gather(*tasks)
    okc = sum(1 for ok, *_ in results if ok)
    lats = sorted(dt for ok, dt, *_ in results if ok)
    toks = sum(n for ok, _, n, _ in results if ok)
    wall = time.monotonic() - start
    p50 = lats[len(lats) // 2] if lats else 0.0
    print(json.dumps({
        "requests": len(results), "ok": okc,
        "wall_s": round(wall, 3), "output_tokens": toks,
        "tok_per_s": round(toks / wall, 2) if wall > 0 else 0.0,
        "latency_p50_s": round(p50, 4),
        "latency_max_s": round(lats[-1], 4) if lats else 0.0,
    }))
    return 0 if okc == len(results) else 1


def main():
    ap = argparse.ArgumentParser("dynamo_amd.tools.replay")
    ap.add_argument("file")
    ap.add_argument("--url", default="http://127.0.0.1:8000")
    ap.add_argument("--speed", type=float, default=1.0)
    ap.add_argument("--max-requests", type=int, default=0)
    a = ap.parse_args()
    sys.exit(asyncio.run(replay(a.file, a.url, a.speed, a.max_requests)))


if __name__ == "__main__":
    main()
