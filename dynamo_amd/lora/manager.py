"""LoRA adapters for the native engine.

Capability parity with the reference's LoRA stack (ai-dynamo/dynamo
lib/llm/src/lora: downloader/cache/lora-aware routing, and the worker
endpoints load_lora/unload_lora/list_loras,
components/src/dynamo/vllm/worker_factory.py:1378-1413).

Design: low-rank deltas on the attention qkv/o and MLP gate_up/down
projections (y += (x A^T) B^T * alpha/r). One adapter is ACTIVE per engine
at a time (engine-level activation, not per-request batching): multi-
adapter serving is achieved across workers — the router filters workers by
the adapters they have loaded — rather than by mixing adapters inside one
batch. Activating/deactivating invalidates captured decode graphs.
Adapters load from a torch state-dict file or are random-initialized
(seeded) for synthetic serving.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch

log = logging.getLogger("dynamo_amd.lora")

TARGETS = ("qkv", "o", "gate_up", "down")


class LoRAAdapter:
    def __init__(self, name: str, rank: int, alpha: float,
                 weights: Dict[str, tuple]):
        """weights: {f"{layer}.{target}": (A [r, in], B [out, r])}"""
        self.name = name
        self.rank = rank
        self.alpha = alpha
        self.scale = alpha / rank
        self.weights = weights

    @staticmethod
    def random(name: str, model, rank: int = 8, alpha: float = 16.0,
               seed: int = 0, std: float = 0.1) -> "LoRAAdapter":
        g = torch.Generator(device="cpu").manual_seed(seed)
        weights = {}
        for li, layer in enumerate(model.layers):
            mods = _layer_targets(layer)
            for tgt, (w,) in mods.items():
                out_f, in_f = w.shape[-2], w.shape[-1]
                A = torch.empty(rank, in_f).normal_(0, std, generator=g)
                B = torch.empty(out_f, rank).normal_(0, std, generator=g)
                weights[f"{li}.{tgt}"] = (
                    A.to(w.device, w.dtype), B.to(w.device, w.dtype))
        return LoRAAdapter(name, rank, alpha, weights)

    @staticmethod
    def load(name: str, path: str, model, rank: int, alpha: float
             ) -> "LoRAAdapter":
        sd = torch.load(path, map_location="cpu")
        weights = {}
        for li, layer in enumerate(model.layers):
            for tgt, (w,) in _layer_targets(layer).items():
                key = f"{li}.{tgt}"
                A = sd[f"{key}.A"].to(w.device, w.dtype)
                B = sd[f"{key}.B"].to(w.device, w.dtype)
                weights[key] = (A, B)
        return LoRAAdapter(name, rank, alpha, weights)


def _layer_targets(layer) -> Dict[str, tuple]:
    out = {}
    if hasattr(layer, "attn"):
        out["qkv"] = (layer.attn.wqkv,)
        out["o"] = (layer.attn.wo,)
    if hasattr(layer, "mlp"):
        out["gate_up"] = (layer.mlp.w_gate_up,)
        out["down"] = (layer.mlp.w_down,)
    return out


class LoRAManager:
    """Holds loaded adapters and applies/clears the active one."""

    def __init__(self, model):
        self.model = model
        self.adapters: Dict[str, LoRAAdapter] = {}
        self.active: Optional[str] = None

    def load(self, name: str, path: Optional[str] = None, rank: int = 8,
             alpha: float = 16.0, seed: int = 0):
        if path:
            ad = LoRAAdapter.load(name, path, self.model, rank, alpha)
        else:
            ad = LoRAAdapter.random(name, self.model, rank, alpha, seed)
        self.adapters[name] = ad
        log.info("loaded LoRA %s (rank %d)", name, rank)

    def unload(self, name: str):
        if self.active == name:
            self.deactivate()
        self.adapters.pop(name, None)

    def list(self) -> List[str]:
        return sorted(self.adapters)

    def activate(self, name: str):
        ad = self.adapters[name]
        for li, layer in enumerate(self.model.layers):
            if hasattr(layer, "attn"):
                layer.attn.lora = {
                    "qkv": ad.weights[f"{li}.qkv"] + (ad.scale,),
                    "o": ad.weights[f"{li}.o"] + (ad.scale,),
                }
            if hasattr(layer, "mlp"):
                layer.mlp.lora = {
                    "gate_up": ad.weights[f"{li}.gate_up"] + (ad.scale,),
                    "down": ad.weights[f"{li}.down"] + (ad.scale,),
                }
        self.active = name

    def deactivate(self):
        for layer in self.model.layers:
            if hasattr(layer, "attn"):
                layer.attn.lora = None
            if hasattr(layer, "mlp"):
                layer.mlp.lora = None
        self.active = None
