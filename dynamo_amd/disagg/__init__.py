from .transfer import KvPuller, pool_transfer_metadata, register_local_pool

__all__ = ["KvPuller", "pool_transfer_metadata", "register_local_pool"]
