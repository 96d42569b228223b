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


def choose_parallelization(profiles: dict, itl_slo_ms: float,
                           ttft_slo_s: float, total_gpus: int,
                           req_per_s: float, isl: int, osl: int) -> dict:
    """TP-config search over per-config profiles (reference parity:
    profiler/profile_sla.py:343's parallelism sweep selection).

    `profiles` maps a config label -> {"parallel": ParallelizationConfig,
    "profile": run_profile() result}. For each config that meets the SLOs,
    compute the replicas needed for `req_per_s` and its GPU cost; pick the
    config serving the load with the fewest GPUs (ties -> higher headroom
    per GPU). Configs whose replica demand exceeds `total_gpus` are
    reported as infeasible."""
    import math

    from dynamo_amd.planner.planner import ParallelizationConfig

    scored = []
    for label, ent in profiles.items():
        par = ent.get("parallel") or ParallelizationConfig()
        pm = ent["profile"]["perf_model"]
        if not ent["profile"].get("meets_slo") or not pm["max_conc_at_itl"]:
            scored.append({"config": label, "feasible": False,
                           "reason": "SLO not met at any concurrency"})
            continue
        tps = pm["decode_tokens_per_s_at_itl"]
        prefill_tps = pm["prefill_tokens_per_s"] or float("inf")
        n_decode = max(1, math.ceil(req_per_s * osl / max(1.0, tps)))
        n_prefill = max(1, math.ceil(req_per_s * isl / prefill_tps))
        replicas = max(n_decode, n_prefill)
        gpus = replicas * par.gpus_per_replica
        scored.append({
            "config": label, "feasible": gpus <= total_gpus,
            "replicas": replicas, "gpus": gpus,
            "tokens_per_gpu": tps / par.gpus_per_replica,
        })
    feasible = [s for s in scored if s.get("feasible")]
    best = (min(feasible, key=lambda s: (s["gpus"], -s["tokens_per_gpu"]))
            if feasible else None)
    return {"candidates": scored,
            "best": best["config"] if best else None}
