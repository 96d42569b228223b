"""roctx ranges for GPU profiler timelines (rocprofv3 --marker-trace).

Reference parity: the reference annotates with NVTX
(ai-dynamo/dynamo lib/runtime/src/nvtx.rs); the MI355X equivalent is
roctx from roctracer. No-ops when libroctx64 is unavailable or
DYNAMO_ROCTX=0. ctypes binding — no build dependency.
"""
from __future__ import annotations

import ctypes
import os
from contextlib import contextmanager

_lib = None
_enabled = os.environ.get("DYNAMO_ROCTX", "0") == "1"
if _enabled:
    for cand in ("libroctx64.so", "libroctx64.so.4",
                 "/opt/rocm/lib/libroctx64.so"):
        try:
            _lib = ctypes.CDLL(cand)
            _lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
            _lib.roctxRangePop.argtypes = []
            break
        except OSError:
            _lib = None


def range_push(name: str):
    if _lib is not None:
        _lib.roctxRangePushA(name.encode())


def range_pop():
    if _lib is not None:
        _lib.roctxRangePop()


@contextmanager
def roctx_range(name: str):
    range_push(name)
    try:
        yield
    finally:
        range_pop()
