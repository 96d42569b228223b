from .logging import setup_logging
from .tracing import RequestTracer, get_tracer, set_tracer, span, trace_event

__all__ = ["setup_logging", "RequestTracer", "get_tracer", "set_tracer",
           "span", "trace_event"]
