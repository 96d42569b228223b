from .pool import WeightPool, weight_allocator, current_allocator

__all__ = ["WeightPool", "weight_allocator", "current_allocator"]
