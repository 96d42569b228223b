from .kv_router import KvRouter, RouterConfig
from .prefill_router import PrefillRouter

__all__ = ["KvRouter", "RouterConfig", "PrefillRouter"]
