"""dynamo_amd — MI355X-native disaggregated LLM inference framework.

A from-scratch CDNA4 (gfx950) build of the capabilities of ai-dynamo/dynamo:
OpenAI-compatible frontend, KV-aware router, disaggregated prefill/decode
worker pools with xGMI KV transfer, paged-KV native engine with hand-written
HIP kernels (paged attention, RMSNorm, RoPE, sampling, MoE), SLA planner,
and a tiered KV-block manager. See SURVEY.md for the blueprint.
"""

import torch as _torch  # noqa: F401  (extensions link against torch libs)

__version__ = "0.1.0"
