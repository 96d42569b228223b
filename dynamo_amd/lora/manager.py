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
        import os
        if path and os.path.isdir(path) and os.path.exists(
                os.path.join(path, "adapter_config.json")):
            ad = load_peft_adapter(name, path, self.model)
        elif path:
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

        def pick(li, keys):
            # PEFT adapters may target a subset of projections
            return {k: ad.weights[f"{li}.{k}"] + (ad.scale,)
                    for k in keys if f"{li}.{k}" in ad.weights} or None

        for li, layer in enumerate(self.model.layers):
            if hasattr(layer, "attn"):
                layer.attn.lora = pick(li, ("qkv", "o"))
            if hasattr(layer, "mlp"):
                layer.mlp.lora = pick(li, ("gate_up", "down"))
        self.active = name

    def deactivate(self):
        for layer in self.model.layers:
            if hasattr(layer, "attn"):
                layer.attn.lora = None
            if hasattr(layer, "mlp"):
                layer.mlp.lora = None
        self.active = None


# ---------------------------------------------------------------------------
def load_peft_adapter(name: str, path: str, model) -> LoRAAdapter:
    """Load an HF PEFT-format adapter dir (adapter_config.json +
    adapter_model.safetensors) onto this build's FUSED projections.

    PEFT targets q/k/v (and gate/up) separately; the fused qkv/gate_up
    modules take a block form: A = stacked per-target A rows, B =
    block-diagonal per-target B columns — an exact rank-(sum r) adapter
    with y += scale * B(Ax) unchanged. TP: B rows follow each target's
    row shard (q by rank, k/v by the rank's kv-head window, gate/up by
    the I shard); input-side As of o_proj/down_proj are column-sliced to
    this rank's input shard (the row-parallel all-reduce sums the
    partial LoRA terms exactly like the base GEMM).
    """
    import json
    import os

    from safetensors import safe_open

    with open(os.path.join(path, "adapter_config.json")) as f:
        acfg = json.load(f)
    r = int(acfg.get("r", 8))
    alpha = float(acfg.get("lora_alpha", 16.0))
    sd = {}
    with safe_open(os.path.join(path, "adapter_model.safetensors"),
                   framework="pt") as f:
        for k in f.keys():
            sd[k.replace("base_model.model.", "")] = f.get_tensor(k)

    tp = model.tp
    cfg = model.cfg
    hd = cfg.head_dim
    kv_idx = (tp.rank if tp.size <= cfg.num_kv_heads
              else (tp.rank * cfg.num_kv_heads) // tp.size)

    def get(li: int, proj: str, mat: str):
        for stem in (f"model.layers.{li}.self_attn.{proj}",
                     f"model.layers.{li}.mlp.{proj}"):
            t = sd.get(f"{stem}.lora_{mat}.weight")
            if t is not None:
                return t.float()
        return None

    def fused(li: int, parts, row_slices, dev_w):
        """Block-form (A, B) over the present `parts`; row_slices gives
        each part's (row0, nrows) in the fused LOCAL weight."""
        present = [(p, sl) for p, sl in zip(parts, row_slices)
                   if get(li, p, "A") is not None]
        if not present:
            return None
        rt = r * len(present)
        in_f = dev_w.shape[-1]
        A = torch.zeros(rt, in_f)
        B = torch.zeros(dev_w.shape[0], rt)
        for j, (p, (row0, nrows, src0)) in enumerate(present):
            A[j * r:(j + 1) * r] = get(li, p, "A")
            Bp = get(li, p, "B")               # [full_out, r]
            B[row0:row0 + nrows, j * r:(j + 1) * r] = \
                Bp[src0:src0 + nrows]
        return (A.to(dev_w.device, dev_w.dtype),
                B.to(dev_w.device, dev_w.dtype))

    weights = {}
    for li, layer in enumerate(model.layers):
        if hasattr(layer, "attn"):
            at = layer.attn
            qr, kr = at.hq * hd, at.hkv * hd
            qkv = fused(li, ("q_proj", "k_proj", "v_proj"),
                        ((0, qr, tp.rank * qr),
                         (qr, kr, kv_idx * kr),
                         (qr + kr, kr, kv_idx * kr)), at.wqkv)
            if qkv is not None:
                weights[f"{li}.qkv"] = qkv
            Ao = get(li, "o_proj", "A")
            Bo = get(li, "o_proj", "B")
            if Ao is not None and Bo is not None:
                if tp.size > 1:
                    n = Ao.shape[1] // tp.size
                    Ao = Ao[:, tp.rank * n:(tp.rank + 1) * n]
                weights[f"{li}.o"] = (Ao.to(at.wo.device, at.wo.dtype),
                                      Bo.to(at.wo.device, at.wo.dtype))
        if hasattr(layer, "mlp"):
            mlp = layer.mlp
            gu = fused(li, ("gate_proj", "up_proj"),
                       ((0, mlp.I, tp.rank * mlp.I),
                        (mlp.I, mlp.I, tp.rank * mlp.I)), mlp.w_gate_up)
            if gu is not None:
                weights[f"{li}.gate_up"] = gu
            Ad = get(li, "down_proj", "A")
            Bd = get(li, "down_proj", "B")
            if Ad is not None and Bd is not None:
                if tp.size > 1:
                    n = Ad.shape[1] // tp.size
                    Ad = Ad[:, tp.rank * n:(tp.rank + 1) * n]
                weights[f"{li}.down"] = (
                    Ad.to(mlp.w_down.device, mlp.w_down.dtype),
                    Bd.to(mlp.w_down.device, mlp.w_down.dtype))
    # rank in the block form varies per module; scale uses the PEFT r
    ad = LoRAAdapter(name, r, alpha, weights)
    return ad
