"""Paged KV cache: page allocator with prefix caching + the device pool.

MI355X-native design decisions:
  - pages are 64 tokens by default (one page = one LDS tile of the prefill
    kernel and a natural xGMI transfer granule);
  - the whole pool is ONE hipMalloc allocation ([L, 2, P, Hkv, ps, hd]) so a
    single hipIpc handle exports every layer to a peer decode worker;
  - 288 GB HBM3E means the pool is sized generously (gpu_mem_fraction of
    whatever is left after weights).

Roles mirrored from the reference: block pool + prefix reuse (vLLM-side
behavior the reference orchestrates), KV event emission for the KV-aware
router (ai-dynamo/dynamo lib/kv-router/src/protocols.rs:1000-1355
KvCacheEvent Stored/Removed/Cleared), hash chain parity via dynamo_amd._core
(lib/kv-hashing/src/compute.rs:15-35).
"""
from __future__ import annotations

from collections import OrderedDict, deque
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from dynamo_amd import _core


@dataclass
class KvEvent:
    kind: str                  # "stored" | "removed" | "cleared"
    hashes: List[int] = field(default_factory=list)
    parent: Optional[int] = None


class PageAllocator:
    """Refcounted page allocator with hash-addressed prefix cache.

    Pages with refcount 0 that carry a block hash are kept in an LRU of
    evictable cached pages; allocation evicts from it when the free list is
    empty (emitting a `removed` KV event)."""

    def __init__(self, num_pages: int, page_size: int, enable_prefix: bool = True):
        self.num_pages = num_pages
        self.page_size = page_size
        self.enable_prefix = enable_prefix
        self.free: deque[int] = deque(range(num_pages))
        self.ref = [0] * num_pages
        self.page_hash: List[Optional[int]] = [None] * num_pages
        self.hash_to_page: Dict[int, int] = {}
        self.evictable: "OrderedDict[int, None]" = OrderedDict()  # page -> None
        self.events: List[KvEvent] = []
        self.host_tier = None  # KVBM G2 (kvbm.host_tier.HostKVTier)

    # -- stats ------------------------------------------------------------
    @property
    def num_free(self) -> int:
        return len(self.free) + len(self.evictable)

    @property
    def usage(self) -> float:
        return 1.0 - self.num_free / max(1, self.num_pages)

    # -- allocation -------------------------------------------------------
    def alloc(self) -> int:
        if self.free:
            pid = self.free.popleft()
        elif self.evictable:
            pid, _ = self.evictable.popitem(last=False)  # LRU evict
            h = self.page_hash[pid]
            if h is not None:
                if self.host_tier is not None:
                    self.host_tier.offload(pid, h)  # demote to G2
                del self.hash_to_page[h]
                self.page_hash[pid] = None
                self.events.append(KvEvent("removed", [h]))
        else:
            raise MemoryError("KV pool exhausted")
        self.ref[pid] = 1
        return pid

    def incref(self, pid: int):
        if self.ref[pid] == 0 and pid in self.evictable:
            del self.evictable[pid]
        self.ref[pid] += 1

    def decref(self, pid: int):
        self.ref[pid] -= 1
        assert self.ref[pid] >= 0
        if self.ref[pid] == 0:
            if self.page_hash[pid] is not None and self.enable_prefix:
                self.evictable[pid] = None  # retain for prefix reuse
            else:
                h = self.page_hash[pid]
                if h is not None:
                    del self.hash_to_page[h]
                    self.page_hash[pid] = None
                    self.events.append(KvEvent("removed", [h]))
                self.free.append(pid)

    # -- prefix cache -----------------------------------------------------
    def lookup(self, h: int) -> Optional[int]:
        """Find a cached page by hash and take a reference. Falls back to
        the host tier (onboard = H2D copy into a fresh device page)."""
        if not self.enable_prefix:
            return None
        pid = self.hash_to_page.get(h)
        if pid is not None:
            self.incref(pid)
            return pid
        if self.host_tier is not None and self.host_tier.contains(h):
            try:
                pid = self.alloc()
            except MemoryError:
                return None
            if self.host_tier.onboard(h, pid):
                # register without re-emitting stored (content unchanged on
                # this worker from the router's perspective)
                if h not in self.hash_to_page:
                    self.page_hash[pid] = h
                    self.hash_to_page[h] = pid
                    self.events.append(KvEvent("stored", [h], None))
                return pid
            self.decref(pid)
        return None

    def register_hash(self, pid: int, h: int, parent: Optional[int]):
        """Mark a now-full page as carrying block hash `h` (emits `stored`)."""
        if self.page_hash[pid] is not None:
            return
        if h in self.hash_to_page:
            # another page already holds this content; keep this one unhashed
            return
        self.page_hash[pid] = h
        self.hash_to_page[h] = pid
        self.events.append(KvEvent("stored", [h], parent))

    def drain_events(self) -> List[KvEvent]:
        ev, self.events = self.events, []
        return ev

    def clear(self):
        """Reset all page state (clear_kv_blocks / update_weights).

        Preserves the host_tier LINK (it is wired once at engine init) but
        clears the tier maps too: post-clear lookups must not onboard pages
        whose KV was computed before the reset (stale after update_weights).
        """
        self.free = deque(range(self.num_pages))
        self.ref = [0] * self.num_pages
        self.page_hash = [None] * self.num_pages
        self.hash_to_page = {}
        self.evictable = OrderedDict()
        self.events = []
        if self.host_tier is not None:
            self.host_tier.clear()
        self.events.append(KvEvent("cleared"))


class SequenceKV:
    """Per-sequence page table + hash chain bookkeeping."""

    def __init__(self, alloc: PageAllocator, salt: int = 0):
        self.alloc = alloc
        self.salt = salt
        self.pages: List[int] = []
        self.num_cached_tokens = 0  # tokens whose KV was reused from cache
        # incremental hash-chain state: hashes registered so far
        self._num_hashed = 0
        self._parent = _core.chain_root(salt)
        self._parent_hash = None

    def match_prefix(self, tokens: List[int]) -> int:
        """Reuse cached full pages covering a prefix of `tokens`.

        Returns the number of reused tokens. Never reuses the *entire*
        prompt (at least one token must be recomputed to produce logits)."""
        ps = self.alloc.page_size
        hashes = _core.chain_hashes(tokens, ps, self.salt)
        reused = 0
        for i, h in enumerate(hashes):
            # keep at least one token to compute
            if (i + 1) * ps >= len(tokens):
                break
            pid = self.alloc.lookup(h)
            if pid is None:
                break
            self.pages.append(pid)
            reused = (i + 1) * ps
            # advance incremental hash state past reused pages
            self._parent = h
            self._parent_hash = h
            self._num_hashed = i + 1
        self.num_cached_tokens = reused
        return reused

    def ensure_capacity(self, num_tokens: int) -> int:
        """Allocate pages so `num_tokens` tokens fit. Returns pages added."""
        ps = self.alloc.page_size
        need = (num_tokens + ps - 1) // ps
        added = 0
        while len(self.pages) < need:
            self.pages.append(self.alloc.alloc())
            added += 1
        return added

    def commit_full_pages(self, tokens: List[int], num_computed: int):
        """Register hashes for pages that just became full (KV written for
        the first `num_computed` tokens of `tokens`). Incremental: only new
        full pages are hashed."""
        ps = self.alloc.page_size
        full = num_computed // ps
        for i in range(self._num_hashed, full):
            h = _core.hash_block(self._parent, tokens[i * ps:(i + 1) * ps])
            prev = None if i == 0 else self._parent_hash
            self.alloc.register_hash(self.pages[i], h, prev)
            self._parent_hash = h
            self._parent = h
            self._num_hashed = i + 1

    def release(self):
        for pid in self.pages:
            self.alloc.decref(pid)
        self.pages = []


class KVCachePool:
    """Device memory for the paged KV cache of all layers.

    GPU: one hipMalloc allocation (ipc-exportable); CPU: a torch tensor.
    Layout [L, 2, P, Hkv_local, ps, hd] — [l, 0] is K, [l, 1] is V.
    """

    def __init__(self, num_layers: int, num_pages: int, num_kv_heads: int,
                 page_size: int, head_dim: int, device: str,
                 dtype=torch.bfloat16, shm_export: bool = False,
                 v_transposed: bool = False):
        # v_transposed: V pages are stored d-major; vcache() views the same
        # storage as [P, Hkv, hd, ps] (page bytes and transfer granularity
        # are unchanged - copies and exports stay layout-agnostic)
        self.v_transposed = v_transposed
        self.shape = (num_layers, 2, num_pages, num_kv_heads, page_size, head_dim)
        self.num_pages = num_pages
        self.page_size = page_size
        self.device = torch.device(device)
        self.dtype = dtype
        self.shm_path = None
        numel = 1
        for s in self.shape:
            numel *= s
        nbytes = numel * torch.tensor([], dtype=dtype).element_size()
        if self.device.type == "cuda":
            from dynamo_amd import ops
            self._raw = ops.hip().ipc_alloc(nbytes, self.device.index or 0)
            self.buffer = self._raw.view(dtype).view(self.shape)
        elif shm_export:
            # CPU stand-in for hipIpc: a file-backed shared mapping a peer
            # process can open by path (cross-process disagg pulls in
            # CPU/gloo tests mirror the GPU hipIpc path)
            import os
            import tempfile
            d = "/dev/shm" if os.path.isdir("/dev/shm") else None
            fd, self.shm_path = tempfile.mkstemp(prefix="dynamo_kvpool_",
                                                 dir=d)
            os.close(fd)
            self.buffer = torch.from_file(self.shm_path, shared=True,
                                          size=numel, dtype=dtype).view(
                                              self.shape)
            self.buffer.zero_()
        else:
            self.buffer = torch.zeros(self.shape, dtype=dtype, device=device)
        self.nbytes = nbytes

    def __del__(self):  # pragma: no cover
        if getattr(self, "shm_path", None):
            import os
            try:
                os.unlink(self.shm_path)
            except OSError:
                pass

    def kcache(self, layer: int) -> torch.Tensor:
        return self.buffer[layer, 0]

    def vcache(self, layer: int) -> torch.Tensor:
        v = self.buffer[layer, 1]
        if self.v_transposed:
            _, _, P, hkv, ps, hd = self.shape
            return v.reshape(P, hkv, hd, ps)
        return v

    def ipc_export(self) -> bytes:
        from dynamo_amd import ops
        return ops.hip().ipc_export(self._raw)

    # page_bytes of ONE layer's K (or V) page — transfer granularity
    @property
    def layer_page_numel(self) -> int:
        _, _, _, hkv, ps, hd = self.shape
        return hkv * ps * hd
