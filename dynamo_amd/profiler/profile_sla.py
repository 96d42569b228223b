"""Pre-deployment SLA profiler.

Reference parity: components/src/dynamo/profiler (profile_sla.py
run_profile): sweeps serving concurrency against a running deployment,
measures TTFT/ITL/throughput per level, and derives the planner's
PerfModel bootstrap (prefill_tokens_per_s, decode_tokens_per_s_at_itl,
max_conc_at_itl) for the given SLA targets.

  python -m dynamo_amd.profiler --url http://127.0.0.1:8000 \
      --model llama-3-8b --isl 2048 --osl 64 \
      --concurrencies 1,2,4,8 --itl-slo-ms 15 --out profile.json
"""
from __future__ import annotations

import json
import subprocess
import sys
from pathlib import Path
from typing import List, Optional


def _loadgen_cmd(url, model, isl, osl, conc, requests, vocab):
    root = Path(__file__).resolve().parents[2]
    return [sys.executable, str(root / "benchmarks" / "loadgen.py"),
            "--url", url, "--model", model, "--isl", str(isl),
            "--osl", str(osl), "--concurrency", str(conc),
            "--requests", str(requests), "--vocab", str(vocab)]


def run_profile(url: str, model: str, isl: int, osl: int,
                concurrencies: List[int], requests_per_level: int,
                itl_slo_ms: float, ttft_slo_s: float,
                vocab: int = 512, out: Optional[str] = None) -> dict:
    sweep = []
    for conc in concurrencies:
        r = subprocess.run(
            _loadgen_cmd(url, model, isl, osl, conc,
                         max(requests_per_level, conc), vocab),
            capture_output=True, text=True, timeout=1800)
        if r.returncode != 0:
            raise RuntimeError(f"loadgen failed at conc {conc}: {r.stderr}")
        rec = json.loads(r.stdout.strip().splitlines()[-1])
        rec["concurrency"] = conc
        sweep.append(rec)

    # derive the planner PerfModel from the sweep
    within = [r for r in sweep
              if r.get("itl_p50_ms") is not None
              and r["itl_p50_ms"] <= itl_slo_ms
              and (r.get("ttft_p50_s") is None
                   or r["ttft_p50_s"] <= ttft_slo_s)]
    best = max(within, key=lambda r: r["concurrency"]) if within else None
    c1 = sweep[0]
    prefill_tps = (isl / c1["ttft_p50_s"]
                   if c1.get("ttft_p50_s") else None)
    result = {
        "model": model, "isl": isl, "osl": osl,
        "slo": {"itl_ms": itl_slo_ms, "ttft_s": ttft_slo_s},
        "sweep": sweep,
        "perf_model": {
            "prefill_tokens_per_s": prefill_tps,
            "max_conc_at_itl": best["concurrency"] if best else None,
            "decode_tokens_per_s_at_itl": best["output_tok_s"] if best else None,
        },
        "meets_slo": best is not None,
    }
    if out:
        Path(out).write_text(json.dumps(result, indent=1))
    return result
