"""Trace converters for the load generator (reference parity:
lib/bench mooncake/trace converters, lib/data-gen mooncake.rs).

Converts public trace formats into the replay format
`benchmarks/loadgen.py --trace` consumes (JSONL of
{ts_s, isl, osl, prefix_group}):

  mooncake: FAST25 conversation-trace JSONL with
    {timestamp (ms), input_length, output_length, hash_ids}
  aiperf / generic: {ts, isl, osl} passthrough with field aliasing

  python -m dynamo_amd.tools.trace_convert --format mooncake \
      --in trace.jsonl --out replay.jsonl [--speed 1.0] [--limit N]
"""
from __future__ import annotations

import argparse
import json
import sys


def convert_mooncake(line: dict, t0: float) -> dict:
    ts = float(line.get("timestamp", 0.0)) / 1000.0
    # prefix_group: requests sharing leading hash_ids share KV prefix —
    # loadgen maps the group id onto a shared prompt prefix
    hids = line.get("hash_ids") or []
    return {
        "ts_s": ts - t0,
        "isl": int(line.get("input_length", 0)),
        "osl": int(line.get("output_length", 0)),
        "prefix_group": (int(hids[0]) if hids else None),
        "prefix_len": len(hids) * int(line.get("block_size", 512)),
    }


def convert_generic(line: dict, t0: float) -> dict:
    ts = float(line.get("ts", line.get("ts_s", line.get("timestamp", 0.0))))
    if ts > 1e7:          # ms epoch heuristics
        ts /= 1000.0
    return {
        "ts_s": ts - t0,
        "isl": int(line.get("isl", line.get("input_length",
                                            line.get("input_tokens", 0)))),
        "osl": int(line.get("osl", line.get("output_length",
                                            line.get("output_tokens", 0)))),
        "prefix_group": line.get("prefix_group"),
    }


def convert(fmt: str, lines, speed: float = 1.0, limit: int = 0):
    conv = {"mooncake": convert_mooncake, "generic": convert_generic}[fmt]
    out = []
    t0 = None
    for raw in lines:
        raw = raw.strip()
        if not raw:
            continue
        rec = json.loads(raw)
        if t0 is None:
            t0 = (float(rec.get("timestamp", 0)) / 1000.0
                  if fmt == "mooncake" else
                  convert_generic(rec, 0.0)["ts_s"])
        r = conv(rec, t0)
        r["ts_s"] = max(0.0, r["ts_s"] / speed)
        out.append(r)
        if limit and len(out) >= limit:
            break
    out.sort(key=lambda r: r["ts_s"])
    return out


def main():
    p = argparse.ArgumentParser("dynamo_amd.tools.trace_convert")
    p.add_argument("--format", choices=["mooncake", "generic"],
                   default="mooncake")
    p.add_argument("--in", dest="inp", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--speed", type=float, default=1.0,
                   help="time compression factor (2.0 = replay 2x faster)")
    p.add_argument("--limit", type=int, default=0)
    a = p.parse_args()
    with open(a.inp) as fh:
        recs = convert(a.format, fh, a.speed, a.limit)
    with open(a.out, "w") as fh:
        for r in recs:
            fh.write(json.dumps(r) + "\n")
    print(f"wrote {len(recs)} records -> {a.out}", file=sys.stderr)


if __name__ == "__main__":
    main()
