from .tp import TPEngineGroup, follower_loop, init_tp

__all__ = ["TPEngineGroup", "follower_loop", "init_tp"]
