"""KVBM G2 tier: pinned-host KV page pool with offload/onboard.

The MI355X-native core of the reference's tiered KV block manager
(ai-dynamo/dynamo lib/llm/src/block_manager: G1 HBM / G2 pinned host,
offload.rs offload manager, TransferStrategy CudaAsyncH2D/D2H
block/transfer.rs:97): pages evicted from the device prefix cache are
offloaded to a pinned host pool (async on a dedicated HIP stream, DMA over
PCIe) and onboarded back on a prefix-cache hit, instead of being
recomputed.

Host layout is page-contiguous [Ph, L, 2, hkv, ps, hd] so one D2H/H2D
memcpy moves a whole page; the device side gathers/scatters the (L, 2)
planes of a page through the page-copy kernels via a staging buffer.
"""
from __future__ import annotations

import logging
from collections import OrderedDict
from typing import Dict, Optional

import torch

log = logging.getLogger("dynamo_amd.kvbm")


class TinyLFU:
    """W-TinyLFU frequency sketch + admission filter (reference parity:
    lib/kvbm-logical/src/tinylfu.rs). A 4-row count-min sketch of 8-bit
    counters estimates page-hash access frequency; counters halve after a
    sample window (aging). On a full cache the incoming page is ADMITTED
    only if its estimated frequency beats the LRU victim's, which keeps
    one-shot scans from flushing hot prefixes out of the host tier."""

    ROWS = 4
    CAP = 255

    def __init__(self, size_hint: int):
        self.width = max(64, 1 << (max(1, size_hint).bit_length() + 2))
        self.mask = self.width - 1
        self.counts = [bytearray(self.width) for _ in range(self.ROWS)]
        self.sample = max(256, 8 * size_hint)
        self.ops = 0

    def _idx(self, h: int, i: int) -> int:
        x = (h ^ (0x9e3779b97f4a7c15 * (i + 1))) & (1 << 64) - 1
        x = (x ^ (x >> 33)) * 0xff51afd7ed558ccd & (1 << 64) - 1
        return (x >> 32) & self.mask

    def touch(self, h: int):
        for i in range(self.ROWS):
            row = self.counts[i]
            j = self._idx(h, i)
            if row[j] < self.CAP:
                row[j] += 1
        self.ops += 1
        if self.ops >= self.sample:
            self.ops = 0
            for row in self.counts:
                for j in range(self.width):
                    row[j] >>= 1

    def estimate(self, h: int) -> int:
        return min(self.counts[i][self._idx(h, i)] for i in range(self.ROWS))

    def admit(self, new_h: int, victim_h: int) -> bool:
        return self.estimate(new_h) >= self.estimate(victim_h)


class HostKVTier:
    def __init__(self, kv_pool, num_host_pages: int,
                 disk_path: str = "", num_disk_pages: int = 0,
                 object_dir: str = "", policy: str = "lru"):
        self.pool = kv_pool
        self.device = kv_pool.device
        L, two, P, hkv, ps, hd = kv_pool.shape
        self.planes = L * two
        self.plane_elems = hkv * ps * hd
        self.page_elems = self.planes * self.plane_elems
        self.num_host_pages = num_host_pages
        pin = self.device.type == "cuda"
        self.host = torch.empty(num_host_pages, self.page_elems,
                                dtype=kv_pool.dtype, pin_memory=pin)
        self.flat = kv_pool.buffer.reshape(self.planes * P, self.plane_elems)
        self.P = P
        # hash -> host page (insertion-ordered for LRU)
        self.map: "OrderedDict[int, int]" = OrderedDict()
        self.free = list(range(num_host_pages))
        # eviction policy: plain LRU, or LRU victim + TinyLFU admission
        self.lfu = TinyLFU(num_host_pages) if policy == "tinylfu" else None
        if self.device.type == "cuda":
            self.stream = torch.cuda.Stream(device=self.device)
            self.staging = torch.empty(self.page_elems, dtype=kv_pool.dtype,
                                       device=self.device)
            self._fence_event = torch.cuda.Event()
        self.stats = {"offloaded": 0, "onboarded": 0, "evicted_host": 0,
                      "hits": 0, "spilled_disk": 0, "onboarded_disk": 0}
        self.events = []  # (kind, hash) host-tier events
        # G3: disk tier under G2 — host LRU evictions spill there
        self.disk = None
        elem_bytes = torch.empty(0, dtype=kv_pool.dtype).element_size()
        if num_disk_pages > 0 and disk_path:
            from .disk_tier import DiskKVTier
            self.disk = DiskKVTier(disk_path, num_disk_pages,
                                   self.page_elems * elem_bytes)
        # G4: shared content-addressed object store (cross-WORKER reuse);
        # publishes happen off-thread so evictions never stall the engine
        self.objects = None
        self._obj_pool = None
        if object_dir:
            from concurrent.futures import ThreadPoolExecutor
            from .object_tier import ObjectKVTier
            self.objects = ObjectKVTier(object_dir,
                                        self.page_elems * elem_bytes)
            self._obj_pool = ThreadPoolExecutor(max_workers=1)
        self.stats["published_object"] = 0
        self.stats["onboarded_object"] = 0

    def _plane_ids(self, pid: int) -> torch.Tensor:
        ids = [k * self.P + pid for k in range(self.planes)]
        return torch.tensor(ids, dtype=torch.int32, device=self.device)

    def contains(self, h: int) -> bool:
        return (h in self.map
                or (self.disk is not None and self.disk.contains(h))
                or (self.objects is not None and self.objects.contains(h)))

    def _page_bytes(self, hp: int) -> bytes:
        if self.device.type == "cuda":
            # the page may have an in-flight D2H copy on our stream
            self.stream.synchronize()
        return self.host[hp].view(torch.uint8).numpy().tobytes()

    def _alloc_host(self, new_h: Optional[int] = None) -> Optional[int]:
        if self.free:
            return self.free.pop()
        if self.map:
            victim_h = next(iter(self.map))       # LRU victim
            if (self.lfu is not None and new_h is not None
                    and not self.lfu.admit(new_h, victim_h)):
                # TinyLFU admission: the incoming page is colder than the
                # coldest resident - refuse the insert instead of evicting
                self.stats["admission_rejects"] = (
                    self.stats.get("admission_rejects", 0) + 1)
                return None
            old_h, hp = self.map.popitem(last=False)
            self.stats["evicted_host"] += 1
            if self.disk is not None and self.disk.put(old_h,
                                                       self._page_bytes(hp)):
                self.stats["spilled_disk"] += 1
            else:
                self.events.append(("removed_host", old_h))
            return hp
        return None

    def _onboard_from_disk(self, h: int) -> Optional[int]:
        """Promote a G3/G4 page back into a host slot; returns host page."""
        data = self.disk.get(h) if self.disk is not None else None
        if data is None and self.objects is not None:
            data = self.objects.get(h)
            if data is not None:
                self.stats["onboarded_object"] += 1
        if data is None:
            return None
        hp = self._alloc_host(h)
        if hp is None:
            return None
        self.host[hp].view(torch.uint8).copy_(
            torch.frombuffer(bytearray(data), dtype=torch.uint8))
        self.map[h] = hp
        self.map.move_to_end(h)
        self.stats["onboarded_disk"] += 1
        return hp

    # -- device -> host (called from PageAllocator eviction hook) --------
    def offload(self, pid: int, h: int):
        if self.lfu is not None:
            self.lfu.touch(h)
        if h in self.map:
            return
        hp = self._alloc_host(h)
        if hp is None:
            return
        if self.device.type == "cuda":
            from dynamo_amd import ops
            with torch.cuda.stream(self.stream):
                ops.hip().gather_pages(self.staging, self.flat,
                                       self._plane_ids(pid))
                self.host[hp].copy_(self.staging, non_blocking=True)
                self._fence_event.record(self.stream)
        else:
            ids = [k * self.P + pid for k in range(self.planes)]
            self.host[hp].copy_(self.flat[ids].reshape(-1))
        self.map[h] = hp
        self.map.move_to_end(h)
        self.stats["offloaded"] += 1
        self.events.append(("stored_host", h))
        if self.objects is not None and not self.objects.contains(h):
            data = self._page_bytes(hp)
            self._obj_pool.submit(self.objects.put, h, data)
            self.stats["published_object"] += 1

    # -- host -> device (prefix-cache onboard) ---------------------------
    def onboard(self, h: int, pid: int) -> bool:
        if self.lfu is not None:
            self.lfu.touch(h)
        hp = self.map.get(h)
        if hp is None:
            hp = self._onboard_from_disk(h)   # G3 -> G2 promote
        if hp is None:
            return False
        if self.device.type == "cuda":
            from dynamo_amd import ops
            with torch.cuda.stream(self.stream):
                self.staging.copy_(self.host[hp], non_blocking=True)
                ops.hip().scatter_pages(self.staging, self.flat,
                                        self._plane_ids(pid))
                self._fence_event.record(self.stream)
        else:
            ids = [k * self.P + pid for k in range(self.planes)]
            self.flat[ids] = self.host[hp].reshape(self.planes,
                                                   self.plane_elems)
        self.map.move_to_end(h)
        self.stats["onboarded"] += 1
        self.stats["hits"] += 1
        return True

    def fence(self):
        """Make the current compute stream wait for in-flight transfers.
        Call once before each forward pass."""
        if self.device.type == "cuda":
            torch.cuda.current_stream(self.device).wait_event(self._fence_event)

    def drain_events(self):
        ev, self.events = self.events, []
        return ev

    def clear(self):
        """Drop every cached page in G2/G3/G4: called from
        PageAllocator.clear() so a KV reset (clear_kv_blocks /
        update_weights) cannot onboard pages computed under old weights."""
        if self.device.type == "cuda":
            self.stream.synchronize()  # let in-flight offloads land first
        self.map.clear()
        self.free = list(range(self.num_host_pages))
        self.events = []
        if self.disk is not None:
            for h in list(getattr(self.disk, "map", {})):
                self.disk.remove(h)
        if self.objects is not None:
            # wait for queued publishes, then drop our published objects
            self._obj_pool.shutdown(wait=True)
            from concurrent.futures import ThreadPoolExecutor
            self._obj_pool = ThreadPoolExecutor(max_workers=1)
            self.objects.clear()
