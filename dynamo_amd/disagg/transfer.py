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
    elif getattr(kv_pool, "shm_path", None):
        meta["shm_path"] = kv_pool.shm_path
    return meta


def tp_transfer_metadata(instance_id: str, kv_pool, tp) -> dict:
    """Gather per-rank pool metadata across a TP group (rank0 publishes).

    The deterministic lockstep engines use identical page ids on every
    rank, so a decode TP group pulls pages rank-to-rank: decode rank r
    opens prefill rank r's pool (equal TP degrees; the reference reshardes
    via permute_scatter when degrees differ — out of scope for the target
    configs)."""
    import torch.distributed as dist
    local = pool_transfer_metadata(f"{instance_id}#r{tp.rank}", kv_pool)
    box = [None] * tp.size
    dist.all_gather_object(box, local, group=tp.control_group)
    return {"instance_id": instance_id, "tp_size": tp.size, "ranks": box}


def rank_meta(meta: dict, tp_rank: int) -> dict:
    """Select this rank's slice of (possibly TP-grouped) pool metadata."""
    if "ranks" in meta:
        ranks = meta["ranks"]
        if tp_rank >= len(ranks):
            raise RuntimeError(
                f"KV transfer between unequal TP degrees unsupported: "
                f"need rank {tp_rank}, peer has {len(ranks)}")
        return ranks[tp_rank]
    if tp_rank != 0:
        raise RuntimeError(
            "KV transfer between unequal TP degrees unsupported: "
            f"peer is TP1, local rank {tp_rank}")
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
        elif "ipc_handle" in meta:
            from dynamo_amd import ops
            handle = binascii.unhexlify(meta["ipc_handle"])
            raw = ops.hip().ipc_open(handle, meta["pool_nbytes"],
                                     self.local.device.index or 0)
            dtype = getattr(torch, meta["dtype"])
            buf = raw.view(dtype).view(meta["pool_shape"])
        elif "shm_path" in meta:  # CPU cross-process (shared file mapping)
            dtype = getattr(torch, meta["dtype"])
            shape = meta["pool_shape"]
            numel = 1
            for s in shape:
                numel *= s
            buf = torch.from_file(meta["shm_path"], shared=True,
                                  size=numel, dtype=dtype).view(shape)
        else:
            raise RuntimeError(
                f"no transfer path to pool {iid}: not local, no ipc_handle, "
                "no shm_path")
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
