"""KVBM G3 tier: disk-backed KV page pool under the pinned-host G2 tier.

Reference parity: the G3 (NVMe/disk) tier of the reference's tiered block
manager (ai-dynamo/dynamo lib/kvbm-engine src/lib.rs:9-24 G1-G4 tier model;
lib/llm/src/block_manager offload.rs; TransferStrategy Nixl(Read|Write) for
G2<->G3, block/transfer.rs:74-86). MI355X-native stance: plain pread/pwrite
on a preallocated file (O_DIRECT/io_uring is a later optimization) — pages
only reach G3 when evicted from G2, so this path is latency-tolerant.

Layout: one file, fixed page_bytes records, slot = offset/page_bytes.
Thread-safe for the engine's single-threaded use + async executor spills.
"""
from __future__ import annotations

import os
from collections import OrderedDict
from typing import Optional


class DiskKVTier:
    def __init__(self, path: str, num_pages: int, page_bytes: int):
        self.path = path
        self.num_pages = num_pages
        self.page_bytes = page_bytes
        self.fd = os.open(path, os.O_RDWR | os.O_CREAT, 0o600)
        os.ftruncate(self.fd, num_pages * page_bytes)  # sparse preallocate
        self.map: "OrderedDict[int, int]" = OrderedDict()  # hash -> slot
        self.free = list(range(num_pages))
        self.stats = {"stored": 0, "loaded": 0, "evicted": 0}

    def contains(self, h: int) -> bool:
        return h in self.map

    def _alloc(self) -> Optional[int]:
        if self.free:
            return self.free.pop()
        if self.map:
            _old, slot = self.map.popitem(last=False)  # LRU
            self.stats["evicted"] += 1
            return slot
        return None

    def put(self, h: int, data: bytes) -> bool:
        if h in self.map:
            self.map.move_to_end(h)
            return True
        assert len(data) == self.page_bytes
        slot = self._alloc()
        if slot is None:
            return False
        os.pwrite(self.fd, data, slot * self.page_bytes)
        self.map[h] = slot
        self.stats["stored"] += 1
        return True

    def get(self, h: int) -> Optional[bytes]:
        slot = self.map.get(h)
        if slot is None:
            return None
        self.map.move_to_end(h)
        self.stats["loaded"] += 1
        return os.pread(self.fd, self.page_bytes, slot * self.page_bytes)

    def remove(self, h: int):
        slot = self.map.pop(h, None)
        if slot is not None:
            self.free.append(slot)

    def close(self):
        if self.fd >= 0:
            os.close(self.fd)
            self.fd = -1

    def __del__(self):  # pragma: no cover
        try:
            self.close()
        except Exception:
            pass
