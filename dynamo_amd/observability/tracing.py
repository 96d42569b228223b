"""Request tracing: per-request span trees fanned out to JSONL
(reference parity: lib/llm/src/request_trace/ JSONL sink + the
http-request -> handle_payload span chain, runtime/src/logging.rs
make_handle_payload_span). GPU ranges use torch.cuda.nvtx, which maps to
roctx markers on ROCm (the reference's runtime/src/nvtx.rs analog) so
rocprofv3 --marker-trace can correlate host spans with kernels."""
from __future__ import annotations

import contextlib
import contextvars
import json
import os
import threading
import time
import uuid
from typing import Optional

_current_span = contextvars.ContextVar("dynamo_span", default=None)
_tracer: "Optional[RequestTracer]" = None


class RequestTracer:
    """JSONL span sink; thread-safe append."""

    def __init__(self, path: Optional[str] = None):
        self.path = path or os.environ.get("DYN_REQUEST_TRACE_FILE")
        self._lock = threading.Lock()
        self._fh = open(self.path, "a") if self.path else None

    def emit(self, rec: dict):
        if self._fh is None:
            return
        with self._lock:
            self._fh.write(json.dumps(rec, separators=(",", ":")) + "\n")
            self._fh.flush()

    def close(self):
        if self._fh:
            self._fh.close()


def set_tracer(t: Optional[RequestTracer]):
    global _tracer
    _tracer = t


def get_tracer() -> Optional[RequestTracer]:
    return _tracer


def trace_event(name: str, **fields):
    t = _tracer
    if t is None:
        return
    parent = _current_span.get()
    rec = {"ts": round(time.time(), 6), "event": name, **fields}
    if parent:
        rec.setdefault("trace_id", parent[0])
        rec.setdefault("span_id", parent[1])
    t.emit(rec)


@contextlib.contextmanager
def span(name: str, request_id: Optional[str] = None, **fields):
    """Span context: JSONL begin/end + roctx range on GPU."""
    parent = _current_span.get()
    trace_id = parent[0] if parent else (request_id or uuid.uuid4().hex[:16])
    span_id = uuid.uuid4().hex[:8]
    token = _current_span.set((trace_id, span_id))
    t0 = time.time()
    nvtx = None
    try:
        import torch
        if torch.cuda.is_available():
            torch.cuda.nvtx.range_push(name)  # roctx on ROCm
            nvtx = True
    except Exception:
        pass
    try:
        yield
    finally:
        if nvtx:
            import torch
            torch.cuda.nvtx.range_pop()
        _current_span.reset(token)
        t = _tracer
        if t is not None:
            t.emit({"ts": round(t0, 6), "dur_ms": round((time.time() - t0) * 1e3, 3),
                    "span": name, "trace_id": trace_id, "span_id": span_id,
                    "parent_id": parent[1] if parent else None,
                    **({"request_id": request_id} if request_id else {}),
                    **fields})
