"""KVBM G4 tier: content-addressed shared object store for KV pages.

Reference parity: the G4 (object/S3) tier of the reference's tiered block
manager (ai-dynamo/dynamo lib/kvbm-engine src/lib.rs:9-24 G1-G4 model,
`object/` S3 store). MI355X-native stance: a shared filesystem directory
of content-addressed page files (one file per block hash) standing in for
the object store — unlike G2/G3, this tier is SHARED across workers, so a
page prefilled by one engine can be onboarded by another without
recompute. Writes are atomic (tmp + rename); reads are lock-free.
"""
from __future__ import annotations

import os
import tempfile
from typing import Optional


class ObjectKVTier:
    def __init__(self, root: str, page_bytes: int,
                 max_objects: int = 1_000_000):
        self.root = root
        self.page_bytes = page_bytes
        self.max_objects = max_objects
        os.makedirs(root, exist_ok=True)
        self.stats = {"put": 0, "hit": 0, "miss": 0}

    def _path(self, h: int) -> str:
        hx = format(h & 0xFFFFFFFFFFFFFFFF, "016x")
        return os.path.join(self.root, hx[:2], hx)

    def contains(self, h: int) -> bool:
        return os.path.exists(self._path(h))

    def put(self, h: int, data: bytes) -> bool:
        assert len(data) == self.page_bytes
        path = self._path(h)
        if os.path.exists(path):
            return True
        os.makedirs(os.path.dirname(path), exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=os.path.dirname(path))
        try:
            os.write(fd, data)
            os.close(fd)
            os.replace(tmp, path)       # atomic publish
        except OSError:
            try:
                os.close(fd)
            except OSError:
                pass
            if os.path.exists(tmp):
                os.unlink(tmp)
            return False
        self.stats["put"] += 1
        return True

    def clear(self):
        """Remove every published object (KV reset: content computed under
        old weights is invalid for every worker sharing this store)."""
        import shutil
        for sub in os.listdir(self.root):
            p = os.path.join(self.root, sub)
            if os.path.isdir(p) and len(sub) == 2:
                shutil.rmtree(p, ignore_errors=True)

    def get(self, h: int) -> Optional[bytes]:
        try:
            with open(self._path(h), "rb") as fh:
                data = fh.read()
            if len(data) != self.page_bytes:
                return None
            self.stats["hit"] += 1
            return data
        except OSError:
            self.stats["miss"] += 1
            return None
