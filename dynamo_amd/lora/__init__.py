from .manager import LoRAAdapter, LoRAManager

__all__ = ["LoRAAdapter", "LoRAManager"]
