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
    # kernel bring-up / A-B toggles (read by the HIP dispatch layer)
    "DYNAMO_DECODE_CHUNK": "decode attention context-chunk tokens override",
    "DYNAMO_DECODE_MFMA": "0 = force the VALU decode path (token-major only)",
    "DYNAMO_DECODE_SWAPPED": "0 = A-operand MFMA decode (token-major only)",
    "DYNAMO_VT3_VARIANT":
        "d-major decode variant select: 0=VT4 (default), 1=VT3, 2=VT2, "
        "5=VT5 (recorded-negative V-LDS variant)",
    "DYNAMO_FUSED_MERGE":
        "1 = fused decode chunk merge (recorded-negative; threadfence cost)",
    "DYNAMO_MOE_BMM": "0 = grouped-kernel MoE decode instead of padded bmm",
    "DYNAMO_MOE_MFMA": "0 = VALU grouped MoE GEMM",
    "DYNAMO_MOE_GRAPHS": "0 = exclude MoE layers from hipGraph capture",
    "DYNAMO_ROCTX": "1 = roctx range annotations for rocprofv3 traces",
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
