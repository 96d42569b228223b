from .tunableop import enable_tunableop

__all__ = ["enable_tunableop"]
