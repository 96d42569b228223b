"""Load generator E2E against a mock-worker stack over real HTTP."""
import json
import subprocess
import sys
import time

import httpx
import pytest

from tests.proc_utils import ManagedProcess, worker_cmd


@pytest.mark.timeout(180)
def test_loadgen_against_stack(tmp_path):
    disc = f"file:{tmp_path}/disc"
    w = ManagedProcess(worker_cmd(mock=True, model="tiny-llama",
                                  discovery=disc, page_size=16),
                       ready_marker="WORKER_READY").start()
    front = ManagedProcess(
        [sys.executable, "-m", "dynamo_amd.frontend", "--discovery", disc,
         "--port", "18233"], ready_marker="FRONTEND_READY").start()
    try:
        base = "http://127.0.0.1:18233"
        deadline = time.time() + 60
        with httpx.Client(timeout=10) as c:
            while time.time() < deadline:
                try:
                    if c.get(base + "/health").json()["models"]:
                        break
                except httpx.TransportError:
                    pass
                time.sleep(0.3)
        out = subprocess.run(
            [sys.executable, "benchmarks/loadgen.py", "--url", base,
             "--model", "tiny-llama", "--isl", "64", "--osl", "8",
             "--concurrency", "4", "--requests", "8"],
            capture_output=True, text=True, timeout=90,
            cwd=front.env.get("PWD") or None)
        assert out.returncode == 0, out.stderr[-1000:]
        r = json.loads(out.stdout.strip().splitlines()[-1])
        assert r["errors"] == 0
        assert r["output_tok_s"] > 0
        assert r["ttft_p50_s"] is not None
    finally:
        front.stop()
        w.stop()


@pytest.mark.timeout(180)
def test_loadgen_multiturn(tmp_path):
    """Multiturn mode: per-turn TTFT report against the mock stack, with
    sticky sessions (2 workers) so every turn lands on its pinned worker."""
    import socket
    disc = f"file:{tmp_path}/disc"
    ws = [ManagedProcess(worker_cmd(mock=True, model="tiny-llama",
                                    discovery=disc, page_size=16),
                         ready_marker="WORKER_READY").start()
          for _ in range(2)]
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    front = ManagedProcess(
        [sys.executable, "-m", "dynamo_amd.frontend", "--discovery", disc,
         "--port", str(port)], ready_marker="FRONTEND_READY").start()
    try:
        base = f"http://127.0.0.1:{port}"
        deadline = time.time() + 60
        with httpx.Client(timeout=10) as c:
            while time.time() < deadline:
                try:
                    if c.get(base + "/health").json()["models"]:
                        break
                except httpx.TransportError:
                    pass
                time.sleep(0.3)
        out = subprocess.run(
            [sys.executable, "benchmarks/loadgen.py", "--url", base,
             "--model", "tiny-llama", "--isl", "96", "--osl", "8",
             "--concurrency", "4", "--requests", "4", "--turns", "3"],
            capture_output=True, text=True, timeout=90)
        assert out.returncode == 0, out.stderr
        summary = json.loads(out.stdout.strip().splitlines()[-1])
        assert summary["errors"] == 0
        assert summary["conversations"] == 4 and summary["turns"] == 3
        assert len(summary["ttft_p50_by_turn_s"]) == 3
        assert all(v is not None for v in summary["ttft_p50_by_turn_s"])
        # 4 convs x 3 turns x 8 output tokens
        assert summary["output_tok_s"] > 0
    finally:
        front.stop()
        for w in ws:
            w.stop()


@pytest.mark.timeout(300)
def test_profiler_sweep(tmp_path):
    """SLA profiler: concurrency sweep -> planner PerfModel bootstrap."""
    import socket
    disc = f"file:{tmp_path}/disc"
    w = ManagedProcess(worker_cmd(mock=True, model="tiny-llama",
                                  discovery=disc, page_size=16),
                       ready_marker="WORKER_READY").start()
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    front = ManagedProcess(
        [sys.executable, "-m", "dynamo_amd.frontend", "--discovery", disc,
         "--port", str(port)], ready_marker="FRONTEND_READY").start()
    try:
        base = f"http://127.0.0.1:{port}"
        deadline = time.time() + 60
        with httpx.Client(timeout=10) as c:
            while time.time() < deadline:
                try:
                    if c.get(base + "/health").json()["models"]:
                        break
                except httpx.TransportError:
                    pass
                time.sleep(0.3)
        from dynamo_amd.profiler import run_profile
        res = run_profile(base, "tiny-llama", isl=64, osl=8,
                          concurrencies=[1, 2], requests_per_level=4,
                          itl_slo_ms=10_000.0, ttft_slo_s=60.0,
                          out=str(tmp_path / "prof.json"))
        assert len(res["sweep"]) == 2
        assert res["meets_slo"] is True
        pm = res["perf_model"]
        assert pm["max_conc_at_itl"] == 2
        assert pm["decode_tokens_per_s_at_itl"] > 0
        assert (tmp_path / "prof.json").exists()
        # the derived numbers plug straight into the planner's PerfModel
        from dynamo_amd.planner.planner import PerfModel
        PerfModel(prefill_tokens_per_s=pm["prefill_tokens_per_s"] or 1.0,
                  decode_tokens_per_s_at_itl=pm["decode_tokens_per_s_at_itl"],
                  max_conc_at_itl=pm["max_conc_at_itl"])
    finally:
        front.stop()
        w.stop()


@pytest.mark.timeout(180)
def test_trace_convert_and_replay(tmp_path):
    """Mooncake-format trace -> converter -> loadgen --trace replay
    against a live mock stack (reference parity: lib/bench trace
    converters + offline_replay_bench)."""
    import os
    ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    raw = tmp_path / "mooncake.jsonl"
    recs = [
        {"timestamp": 1000, "input_length": 40, "output_length": 3,
         "hash_ids": [7, 8], "block_size": 8},
        {"timestamp": 1050, "input_length": 40, "output_length": 3,
         "hash_ids": [7, 8], "block_size": 8},     # same prefix group
        {"timestamp": 1200, "input_length": 24, "output_length": 2,
         "hash_ids": [], "block_size": 8},
    ]
    raw.write_text("\n".join(json.dumps(r) for r in recs))
    conv = tmp_path / "replay.jsonl"
    r = subprocess.run(
        [sys.executable, "-m", "dynamo_amd.tools.trace_convert",
         "--format", "mooncake", "--in", str(raw), "--out", str(conv),
         "--speed", "10.0"],
        capture_output=True, text=True, cwd=ROOT)
    assert r.returncode == 0, r.stderr
    lines = [json.loads(ln) for ln in conv.read_text().splitlines()]
    assert len(lines) == 3
    assert lines[0]["ts_s"] == 0.0
    assert lines[0]["prefix_group"] == 7 and lines[0]["prefix_len"] == 16
    assert lines[2]["prefix_group"] is None

    disc = f"file:{tmp_path}/disc"
    w = ManagedProcess(worker_cmd(mock=True, model="tiny-llama",
                                  discovery=disc, page_size=16),
                       ready_marker="WORKER_READY").start()
    front = ManagedProcess(
        [sys.executable, "-m", "dynamo_amd.frontend", "--discovery", disc,
         "--port", "18237"], ready_marker="FRONTEND_READY").start()
    try:
        base = "http://127.0.0.1:18237"
        deadline = time.time() + 60
        with httpx.Client(timeout=10) as c:
            while time.time() < deadline:
                try:
                    if c.get(base + "/health").json()["models"]:
                        break
                except httpx.TransportError:
                    pass
                time.sleep(0.3)
        out = subprocess.run(
            [sys.executable, "benchmarks/loadgen.py", "--url", base,
             "--model", "tiny-llama", "--trace", str(conv),
             "--concurrency", "4", "--vocab", "512"],
            capture_output=True, text=True, cwd=ROOT, timeout=120)
        assert out.returncode == 0, out.stdout + out.stderr
        rep = json.loads(out.stdout.strip().splitlines()[-1])
        assert rep["requests"] == 3 and rep["errors"] == 0
        assert rep["output_tok_s"] > 0
    finally:
        front.stop()
        w.stop()
