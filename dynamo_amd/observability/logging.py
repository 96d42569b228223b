"""Structured JSONL logging (DYN_LOGGING_JSONL parity,
reference runtime/src/logging.rs)."""
from __future__ import annotations

import json
import logging
import os
import sys
import time


class JsonlFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out = {
            "ts": round(time.time(), 6),
            "level": record.levelname,
            "target": record.name,
            "message": record.getMessage(),
        }
        if record.exc_info:
            out["exception"] = self.formatException(record.exc_info)
        for k in ("request_id", "worker_id", "span"):
            v = getattr(record, k, None)
            if v is not None:
                out[k] = v
        return json.dumps(out, separators=(",", ":"))


def setup_logging(level: str | int = None, jsonl: bool | None = None):
    """Env-configurable: DYN_LOG (level), DYN_LOGGING_JSONL (format)."""
    if level is None:
        level = os.environ.get("DYN_LOG", "INFO").upper()
    if jsonl is None:
        jsonl = os.environ.get("DYN_LOGGING_JSONL", "0") in ("1", "true")
    root = logging.getLogger()
    root.setLevel(level)
    h = logging.StreamHandler(sys.stderr)
    if jsonl:
        h.setFormatter(JsonlFormatter())
    else:
        h.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)s %(name)s %(message)s"))
    root.handlers = [h]
    return root
