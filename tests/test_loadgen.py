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
