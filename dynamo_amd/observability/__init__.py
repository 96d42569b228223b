from .logging import setup_logging
from .roctx import range_pop, range_push, roctx_range
from .tracing import (OtlpExporter, RequestTracer, get_tracer,
                      set_tracer, span, trace_event)

__all__ = ["OtlpExporter", "roctx_range", "range_push", "range_pop",
           "setup_logging", "RequestTracer", "get_tracer", "set_tracer",
           "span", "trace_event"]
