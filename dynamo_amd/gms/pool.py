"""GMS: GPU memory service — shareable weight pools.

The MI355X-native analog of the reference's GPU Memory Service
(ai-dynamo/dynamo lib/gpu_memory_service/README.md: out-of-process GPU
memory owner, VMM handles over sockets, zero-copy weight sharing & crash
survival). Here: all model weights are carved from ONE hipMalloc arena
(exported once via hipIpc/dmabuf); a restarting worker opens the handle
and reconstructs every tensor as a zero-copy view — no re-init, no copy,
~instant warm start, and the weights survive worker crashes because the
owner process holds the allocation.

Tensor identity across processes relies on deterministic construction
order (models build their weights in a fixed order), so the manifest is
just the ordered (shape, dtype) list — verified at import time.
"""
from __future__ import annotations

import binascii
import contextlib
from typing import List, Optional, Tuple

import torch

_current: "Optional[WeightPool]" = None


def current_allocator() -> "Optional[WeightPool]":
    return _current


@contextlib.contextmanager
def weight_allocator(pool: "WeightPool"):
    global _current
    prev = _current
    _current = pool
    try:
        yield pool
    finally:
        _current = prev


class WeightPool:
    """Bump allocator over one device arena.

    mode="build":  owner process — allocates the arena, carves + inits.
    mode="import": worker — views an imported arena, carving must replay
                   the identical sequence (verified against the manifest).
    """

    ALIGN = 256

    def __init__(self, nbytes: int = 0, device: str = "cuda:0",
                 buffer: Optional[torch.Tensor] = None,
                 manifest: Optional[List[Tuple[Tuple[int, ...], str]]] = None):
        self.device = torch.device(device)
        if buffer is not None:
            self.buffer = buffer
            self.mode = "import"
        else:
            if self.device.type == "cuda":
                from dynamo_amd import ops
                self.buffer = ops.hip().ipc_alloc(nbytes, self.device.index or 0)
            else:
                self.buffer = torch.empty(nbytes, dtype=torch.uint8,
                                          device=device)
            self.mode = "build"
        self.offset = 0
        self.manifest: List[Tuple[Tuple[int, ...], str]] = manifest or []
        self._carve_idx = 0

    @property
    def needs_init(self) -> bool:
        return self.mode == "build"

    def allocate(self, shape, dtype) -> torch.Tensor:
        numel = 1
        for s in shape:
            numel *= s
        nbytes = numel * torch.empty(0, dtype=dtype).element_size()
        off = (self.offset + self.ALIGN - 1) // self.ALIGN * self.ALIGN
        if off + nbytes > self.buffer.numel():
            raise MemoryError(
                f"weight pool exhausted: need {off + nbytes}, have "
                f"{self.buffer.numel()}")
        self.offset = off + nbytes
        t = self.buffer[off:off + nbytes].view(dtype).view(shape)
        entry = (tuple(shape), str(dtype))
        if self.mode == "build":
            self.manifest.append(entry)
        else:
            expect = tuple(self.manifest[self._carve_idx])
            got = (list(entry[0]), entry[1])
            assert (list(expect[0]), expect[1]) == got, (
                f"weight manifest mismatch at #{self._carve_idx}: "
                f"{expect} vs {entry}")
            self._carve_idx += 1
        return t

    # -- export / import ------------------------------------------------
    def export_meta(self) -> dict:
        meta = {
            "nbytes": int(self.buffer.numel()),
            "used": int(self.offset),
            "manifest": [(list(s), d) for s, d in self.manifest],
        }
        if self.device.type == "cuda":
            from dynamo_amd import ops
            meta["ipc_handle"] = binascii.hexlify(
                ops.hip().ipc_export(self.buffer)).decode()
        return meta

    @staticmethod
    def open(meta: dict, device: str = "cuda:0") -> "WeightPool":
        dev = torch.device(device)
        if dev.type == "cuda":
            from dynamo_amd import ops
            handle = binascii.unhexlify(meta["ipc_handle"])
            buf = ops.hip().ipc_open(handle, meta["nbytes"], dev.index or 0)
        else:
            raise ValueError("import requires a CUDA device (hipIpc)")
        return WeightPool(buffer=buf, device=device,
                          manifest=[(tuple(s), d) for s, d in meta["manifest"]])


def estimate_pool_bytes(model_cfg, tp_size: int = 1) -> int:
    """Upper bound on total weight bytes for a model config (bf16)."""
    m = model_cfg
    D, I, L, V = m.hidden_size, m.intermediate_size, m.num_layers, m.vocab_size
    hq, hkv, hd = m.num_q_heads, m.num_kv_heads, m.head_dim
    attn = (hq + 2 * hkv) * hd * D + D * hq * hd
    if m.num_experts:
        mlp = m.num_experts * (3 * I * D) + m.num_experts * D
    else:
        mlp = 3 * I * D
    per_layer = (attn + mlp) // tp_size + 4 * D
    total = L * per_layer + 2 * V * D + D
    return int(total * 2 * 1.05) + (1 << 20)
