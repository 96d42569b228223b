"""Multi-process E2E on CPU: subprocess mock workers + frontend subprocess
over file discovery, exercised through real HTTP."""
import sys
import time

import httpx
import pytest

from tests.proc_utils import ManagedProcess, worker_cmd


@pytest.mark.timeout(180)
def test_multiprocess_workers_and_frontend(tmp_path):
    disc = f"file:{tmp_path}/disc"
    workers = []
    front = None
    try:
        for _ in range(2):
            workers.append(ManagedProcess(
                worker_cmd(mock=True, model="tiny-llama", discovery=disc,
                           page_size=16),
                ready_marker="WORKER_READY").start())
        import socket
        with socket.socket() as s:          # pick a free port (avoids
            s.bind(("127.0.0.1", 0))        # collisions across runs)
            port = s.getsockname()[1]
        front = ManagedProcess(
            [sys.executable, "-m", "dynamo_amd.frontend", "--discovery", disc,
             "--port", str(port)],
            ready_marker="FRONTEND_READY").start()
        base = f"http://127.0.0.1:{port}"
        # wait for model registration to reach the frontend
        deadline = time.time() + 60
        with httpx.Client(timeout=30) as client:
            while time.time() < deadline:
                try:
                    r = client.get(base + "/health")
                    if r.status_code == 200 and r.json()["models"]:
                        break
                except httpx.TransportError:
                    pass
                time.sleep(0.3)
            else:
                raise TimeoutError("frontend never saw the model")
            r = client.post(base + "/v1/completions", json={
                "model": "tiny-llama", "prompt": "multi process hello",
                "max_tokens": 6})
            assert r.status_code == 200, r.text
            assert r.json()["usage"]["completion_tokens"] == 6
            # streaming
            with client.stream("POST", base + "/v1/completions", json={
                    "model": "tiny-llama", "prompt": "stream", "max_tokens": 4,
                    "stream": True}) as rs:
                lines = [ln for ln in rs.iter_lines()
                         if ln.startswith("data: ")]
            assert lines[-1] == "data: [DONE]"
    finally:
        if front:
            front.stop()
        for w in workers:
            w.stop()


@pytest.mark.timeout(180)
def test_record_and_replay(tmp_path):
    """Record real HTTP traffic, then replay it with the replay tool."""
    import json
    import socket
    import subprocess
    disc = f"file:{tmp_path}/disc"
    rec = tmp_path / "rec.jsonl"
    workers, front = [], None
    try:
        workers.append(ManagedProcess(
            worker_cmd(mock=True, model="tiny-llama", discovery=disc,
                       page_size=16),
            ready_marker="WORKER_READY").start())
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        front = ManagedProcess(
            [sys.executable, "-m", "dynamo_amd.frontend", "--discovery", disc,
             "--port", str(port), "--record", str(rec)],
            ready_marker="FRONTEND_READY").start()
        base = f"http://127.0.0.1:{port}"
        with httpx.Client(timeout=30) as client:
            deadline = time.time() + 60
            while time.time() < deadline:
                try:
                    if client.get(base + "/health").json()["models"]:
                        break
                except httpx.TransportError:
                    pass
                time.sleep(0.3)
            for i in range(3):
                r = client.post(base + "/v1/completions", json={
                    "model": "tiny-llama", "prompt": [10 + i, 20, 30],
                    "max_tokens": 4})
                assert r.status_code == 200
        assert len(rec.read_text().splitlines()) == 3
        out = subprocess.run(
            [sys.executable, "-m", "dynamo_amd.tools.replay", str(rec),
             "--url", base, "--speed", "100"],
            capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, out.stderr
        summary = json.loads(out.stdout.strip().splitlines()[-1])
        assert summary["requests"] == 3 and summary["ok"] == 3
        assert summary["output_tokens"] == 12
    finally:
        if front:
            front.stop()
        for w in workers:
            w.stop()


@pytest.mark.timeout(120)
def test_worker_status_server(tmp_path):
    """--status-port serves liveness + engine state JSON out-of-band."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    disc = f"file:{tmp_path}/disc"
    w = ManagedProcess(
        worker_cmd(mock=True, model="tiny-llama", discovery=disc,
                   page_size=16) + ["--status-port", str(port)],
        ready_marker="STATUS_READY").start()
    try:
        with httpx.Client(timeout=10) as c:
            deadline = time.time() + 30
            body = None
            while time.time() < deadline:
                try:
                    body = c.get(f"http://127.0.0.1:{port}/status").json()
                    break
                except httpx.TransportError:
                    time.sleep(0.2)
            assert body is not None
            assert body["status"] == "ok"
            assert body["model"] == "tiny-llama"
            assert body["worker_type"] == "aggregated"
            assert body["paused"] is False
            assert body["total_kv_pages"] > 0
    finally:
        w.stop()
