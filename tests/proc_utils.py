"""ManagedProcess-style harness for multi-process E2E tests
(reference parity: tests/utils/managed_process.py ManagedProcess)."""
from __future__ import annotations

import os
import signal
import subprocess
import sys
import time
from typing import List, Optional


class ManagedProcess:
    def __init__(self, args: List[str], ready_marker: Optional[str] = None,
                 timeout: float = 120.0, env: Optional[dict] = None):
        self.args = args
        self.ready_marker = ready_marker
        self.timeout = timeout
        self.proc: Optional[subprocess.Popen] = None
        self.ready_line = ""
        self.env = dict(os.environ, **(env or {}))
        self.env.setdefault("PYTHONUNBUFFERED", "1")

    def start(self) -> "ManagedProcess":
        self.proc = subprocess.Popen(
            self.args, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            text=True, env=self.env,
            cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        if self.ready_marker:
            deadline = time.time() + self.timeout
            while time.time() < deadline:
                line = self.proc.stdout.readline()
                if not line:
                    if self.proc.poll() is not None:
                        raise RuntimeError(
                            f"process died rc={self.proc.returncode}: "
                            f"{' '.join(self.args)}")
                    continue
                if self.ready_marker in line:
                    self.ready_line = line.strip()
                    return self
            self.stop()
            raise TimeoutError(f"no ready marker for {' '.join(self.args)}")
        return self

    def stop(self, grace: float = 5.0):
        if self.proc is None or self.proc.poll() is not None:
            return
        self.proc.send_signal(signal.SIGTERM)
        try:
            self.proc.wait(grace)
        except subprocess.TimeoutExpired:
            self.proc.kill()
            self.proc.wait(5)

    def kill(self):
        if self.proc and self.proc.poll() is None:
            self.proc.kill()
            self.proc.wait(5)

    def tail(self, n=40) -> str:
        try:
            out = self.proc.stdout.read() or ""
            return "\n".join(out.splitlines()[-n:])
        except Exception:
            return ""


def worker_cmd(**kw) -> List[str]:
    args = [sys.executable, "-m", "dynamo_amd.workers"]
    for k, v in kw.items():
        flag = "--" + k.replace("_", "-")
        if v is True:
            args.append(flag)
        elif v is not None and v is not False:
            args += [flag, str(v)]
    return args
