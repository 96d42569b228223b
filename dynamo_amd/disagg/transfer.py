"""Prefill -> decode KV-block transfer over xGMI.

The MI355X-native replacement for the reference's NIXL transfer path
(vLLM NixlConnector + disaggregated_params round-trip,
components/src/dynamo/vllm/handlers.py:3299,3706): the prefill worker's KV
pool is ONE hipMalloc allocation exported once via hipIpc (dmabuf); each
decode worker opens the handle (pages then move with the page-pair copy
kernel — reads over the direct xGMI link, no staging, no host hop).

Same-process "loopback" mode backs CPU tests and co-located workers.
"""
from __future__ import annotations

import binascii
from typing import Dict, List, Optional, Tuple

import torch

# process-local registry for loopback (CPU tests / co-located workers)
_LOCAL_POOLS: Dict[str, "object"] = {}


def register_local_pool(instance_id: str, kv_pool):
    _LOCAL_POOLS[instance_id] = kv_pool


def pool_transfer_metadata(instance_id: str, kv_pool) -> dict:
    """Metadata a prefill worker publishes in discovery."""
    register_local_pool(instance_id, kv_pool)
    meta = {
        "instance_id": instance_id,
        "pool_shape": list(kv_pool.shape),
        "pool_nbytes": int(kv_pool.nbytes),
        "dtype": str(kv_pool.dtype).replace("torch.", ""),
    }
    if kv_pool.device.type == "cuda":
        meta["ipc_handle"] = binascii.hexlify(kv_pool.ipc_export()).decode()
    return meta


class KvPuller:
    """Decode-side: maps peer pools and pulls pages into the local pool."""

    def __init__(self, local_pool):
        self.local = local_pool
        self._mapped: Dict[str, torch.Tensor] = {}
        self._stream = (torch.cuda.Stream()
                        if local_pool.device.type == "cuda" else None)

    def _map(self, meta: dict) -> torch.Tensor:
        iid = meta["instance_id"]
        buf = self._mapped.get(iid)
        if buf is not None:
            return buf
        pool = _LOCAL_POOLS.get(iid)
        if pool is not None:  # same process
            buf = pool.buffer
        else:
            from dynamo_amd import ops
            handle = binascii.unhexlify(meta["ipc_handle"])
            raw = ops.hip().ipc_open(handle, meta["pool_nbytes"],
                                     self.local.device.index or 0)
            dtype = getattr(torch, meta["dtype"])
            buf = raw.view(dtype).view(meta["pool_shape"])
        self._mapped[iid] = buf
        return buf

    def pull(self, meta: dict, src_pages: List[int], dst_pages: List[int]):
        """Copy whole pages (all layers, K+V) from the peer pool into the
        local pool. Synchronous (waits for completion)."""
        assert len(src_pages) == len(dst_pages)
        if not src_pages:
            return
        src = self._map(meta)
        dst = self.local.buffer
        assert list(src.shape[:2]) == list(dst.shape[:2]), "layer/kv mismatch"
        L, two, P_src = src.shape[0], src.shape[1], src.shape[2]
        P_dst = dst.shape[2]
        page_elems = src[0, 0, 0].numel()
        n = len(src_pages)
        base = torch.tensor([src_pages, dst_pages], dtype=torch.int32).T  # [n,2]
        if self.local.device.type == "cuda":
            from dynamo_amd import ops
            # expand pairs across (L, 2) planes on the flattened page dim
            dev = self.local.device
            base_d = base.to(dev)
            offs = torch.arange(L * two, dtype=torch.int32, device=dev)
            pairs = base_d.unsqueeze(0).repeat(L * two, 1, 1)
            pairs[:, :, 0] += offs.view(-1, 1) * P_src
            pairs[:, :, 1] += offs.view(-1, 1) * P_dst
            pairs = pairs.reshape(-1, 2).contiguous()
            src_flat = src.reshape(L * two * P_src, page_elems)
            dst_flat = dst.reshape(L * two * P_dst, page_elems)
            with torch.cuda.stream(self._stream):
                ops.hip().copy_pages(dst_flat, src_flat, pairs)
            self._stream.synchronize()
        else:
            sp = torch.tensor(src_pages, dtype=torch.long)
            dp = torch.tensor(dst_pages, dtype=torch.long)
            dst[:, :, dp] = src[:, :, sp]
