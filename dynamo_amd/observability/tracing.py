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


class OtlpExporter:
    """OTLP/HTTP trace exporter (reference parity: OTLP traces pushed to a
    collector gated by OTEL_EXPORT_ENABLED, observability-architecture.md;
    request_trace/ OTLP sink). Spans batch in memory and POST as OTLP JSON
    (`/v1/traces` ExportTraceServiceRequest) on a background thread."""

    def __init__(self, endpoint: Optional[str] = None,
                 service_name: str = "dynamo_amd",
                 batch_size: int = 64, flush_interval: float = 2.0):
        self.endpoint = (endpoint or
                         os.environ.get("OTEL_EXPORTER_OTLP_ENDPOINT",
                                        "http://127.0.0.1:4318")
                         ).rstrip("/") + "/v1/traces"
        self.service_name = service_name
        self.batch_size = batch_size
        self.flush_interval = flush_interval
        self._buf: list = []
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        self.exported = 0
        self.errors = 0

    @staticmethod
    def _span_to_otlp(rec: dict) -> dict:
        t0 = rec.get("ts", time.time())
        dur = rec.get("dur_ms", 0.0) / 1e3
        attrs = [{"key": k, "value": {"stringValue": str(v)}}
                 for k, v in rec.items()
                 if k not in ("ts", "dur_ms", "span", "trace_id", "span_id",
                              "parent_id")]
        out = {
            "traceId": (rec.get("trace_id") or "0").ljust(32, "0")[:32],
            "spanId": (rec.get("span_id") or "0").ljust(16, "0")[:16],
            "name": rec.get("span") or rec.get("event", "event"),
            "kind": 1,
            "startTimeUnixNano": str(int(t0 * 1e9)),
            "endTimeUnixNano": str(int((t0 + dur) * 1e9)),
            "attributes": attrs,
        }
        if rec.get("parent_id"):
            out["parentSpanId"] = rec["parent_id"].ljust(16, "0")[:16]
        return out

    def emit(self, rec: dict):
        with self._lock:
            self._buf.append(self._span_to_otlp(rec))
            if len(self._buf) >= self.batch_size:
                batch, self._buf = self._buf, []
            else:
                return
        self._post(batch)

    def _loop(self):
        while not self._stop.wait(self.flush_interval):
            self.flush()

    def flush(self):
        with self._lock:
            batch, self._buf = self._buf, []
        if batch:
            self._post(batch)

    def _post(self, spans: list):
        body = json.dumps({"resourceSpans": [{
            "resource": {"attributes": [
                {"key": "service.name",
                 "value": {"stringValue": self.service_name}}]},
            "scopeSpans": [{"scope": {"name": "dynamo_amd"},
                            "spans": spans}],
        }]}).encode()
        try:
            import urllib.request
            req = urllib.request.Request(
                self.endpoint, data=body,
                headers={"Content-Type": "application/json"})
            urllib.request.urlopen(req, timeout=5).read()
            self.exported += len(spans)
        except Exception:
            self.errors += 1

    def close(self):
        self._stop.set()
        self.flush()


class RequestTracer:
    """Span sink fan-out: JSONL file + optional OTLP exporter
    (reference request_trace/ fans to JSONL/NATS/OTLP)."""

    def __init__(self, path: Optional[str] = None,
                 otlp: "Optional[OtlpExporter]" = None):
        self.path = path or os.environ.get("DYN_REQUEST_TRACE_FILE")
        self._lock = threading.Lock()
        self._fh = open(self.path, "a") if self.path else None
        if otlp is None and os.environ.get("OTEL_EXPORT_ENABLED") in (
                "1", "true", "True"):
            otlp = OtlpExporter()
        self.otlp = otlp

    def emit(self, rec: dict):
        if self._fh is not None:
            with self._lock:
                self._fh.write(json.dumps(rec, separators=(",", ":")) + "\n")
                self._fh.flush()
        if self.otlp is not None:
            self.otlp.emit(rec)

    def close(self):
        if self._fh:
            self._fh.close()
        if self.otlp:
            self.otlp.close()


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
