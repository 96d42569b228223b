"""bench.py contract tests on CPU (gloo, tiny model): the driver's
invocation pattern must work and print exactly one valid JSON line."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(nproc, extra, timeout=280):
    if nproc == 1:
        cmd = [sys.executable, "bench.py"]
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={nproc}", "--master-addr", "127.0.0.1",
               "--master-port", "29741", "bench.py"]
    cmd += ["--device", "cpu", "--model", "tiny-llama", "--isl", "96",
            "--osl", "12", "--conc-per-gpu", "2", "--steps", "3",
            "--warmup", "1", "--page-size", "16", "--kv-pool-pages", "256"]
    cmd += extra
    out = subprocess.run(cmd, cwd=ROOT, capture_output=True, text=True,
                         timeout=timeout)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    return json.loads(lines[0])


@pytest.mark.timeout(300)
def test_bench_single():
    r = run_bench(1, [])
    assert r["n_gpus"] == 1
    assert r["value"] > 0
    assert r["scaling"] == "weak"
    assert r["config"]["parallelism"] == "agg1"
    assert r["config"]["ttft_p50_s"] is not None


@pytest.mark.timeout(300)
def test_bench_disagg_world2():
    r = run_bench(2, [])
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "disagg_p1tp1_d1tp1"
    assert r["value"] > 0
    assert r["ms_per_step"] > 0


@pytest.mark.timeout(300)
def test_bench_disagg_world4_tp2():
    """The config-#4 shape at CPU scale: TP2 prefill group + TP2 decode
    group with cross-group page transfer (the driver's SCALE run shape)."""
    r = run_bench(4, [])
    assert r["n_gpus"] == 4
    assert r["config"]["parallelism"] == "disagg_p2tp2_d2tp2"
    assert r["value"] > 0


@pytest.mark.timeout(300)
def test_bench_dp_world2():
    r = run_bench(2, ["--mode", "dp"])
    assert r["config"]["parallelism"] == "dp2"
    assert r["value"] > 0


@pytest.mark.timeout(300)
def test_bench_moe_ep_world2():
    """config #5 path: Mixtral disagg with expert parallelism."""
    r = run_bench(2, ["--model", "tiny-mixtral"])
    assert r["n_gpus"] == 2
    assert r["value"] > 0
