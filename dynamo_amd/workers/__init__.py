from .service import WorkerService, make_sampling

__all__ = ["WorkerService", "make_sampling"]
