"""Centralized environment-variable configuration.

Parity with the reference's centrally-named env config
(ai-dynamo/dynamo runtime/src/config/environment_names.rs, ~180 DYN_* vars;
we implement the subset that is meaningful in this build). Every CLI reads
its defaults from here.
"""
from __future__ import annotations

import os

ENV_VARS = {
    "DYN_DISCOVERY_BACKEND": "discovery backend spec: memory | file:/path",
    "DYN_NAMESPACE": "default namespace for components",
    "DYN_LOG": "log level (INFO, DEBUG, ...)",
    "DYN_LOGGING_JSONL": "1 = structured JSONL logs",
    "DYN_REQUEST_TRACE_FILE": "JSONL request-trace sink path",
    "DYN_HTTP_PORT": "frontend HTTP port",
    "DYN_ROUTER_MODE": "kv | round_robin | random | least_loaded | p2c",
    "DYN_ROUTER_TEMPERATURE": "softmax temperature for kv routing",
    "DYN_KV_BLOCK_SIZE": "KV page size in tokens",
    "DYN_KV_GPU_MEM_FRACTION": "fraction of free HBM for the KV pool",
    "DYN_HOST_CACHE_PAGES": "KVBM G2 pinned-host tier size (pages)",
    "DYN_TUNABLEOP": "0 disables the pre-tuned hipBLASLt table",
    "DYN_WORKER_TYPE": "aggregated | prefill | decode",
    "DYN_BYPASS_TOKEN_THRESHOLD":
        "conditional disagg: net-new prefill tokens below this bypass the "
        "prefill pool (reference conditional_disagg.rs:15; default 2048)",
}


def get(name: str, default=None, cast=str):
    v = os.environ.get(name)
    if v is None:
        return default
    try:
        if cast is bool:
            return v not in ("0", "false", "False", "")
        return cast(v)
    except (TypeError, ValueError):
        return default


def discovery(default="memory"):
    return get("DYN_DISCOVERY_BACKEND", default)


def namespace(default="dynamo"):
    return get("DYN_NAMESPACE", default)
