"""KV-aware router: prefix-overlap + load cost routing.

Native re-creation of the reference's kv-router
(ai-dynamo/dynamo lib/kv-router): the C++ indexer (dynamo_amd._core,
RadixTree::find_matches parity) is fed by per-worker KV-event streams; the
worker selector implements the DefaultWorkerSelector cost
(kv-router/src/scheduling/selector/default.rs:93,217):

  logit = prefill_load_scale * max(0, prefill_blocks - overlap_credit)
        + decode_cost_blocks
        + decode_active_request_weight * active_requests

with softmax temperature sampling (default.rs:23-69) or argmin at T=0.
Load state comes from get_perf_metrics polling + local in-flight tracking.
"""
from __future__ import annotations

import asyncio
import logging
import math
import random
import time
from collections import OrderedDict
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from dynamo_amd import _core
from dynamo_amd.runtime import DistributedRuntime, PushClient

log = logging.getLogger("dynamo_amd.router")


@dataclass
class RouterConfig:
    """KvRouterConfig parity (kv-router/src/scheduling/config.rs:660)."""
    mode: str = "kv"               # kv|round_robin|random|least_loaded|p2c
    block_size: int = 64
    block_salt: int = 0
    overlap_score_weight: float = 1.0
    decode_active_request_weight: float = 1.0
    prefill_load_scale: float = 1.0
    router_temperature: float = 0.0
    metrics_poll_interval: float = 1.0
    # overload rejection (reference parity: http/service/busy_threshold.rs,
    # discovery/worker_monitor.rs load thresholds): when EVERY worker's
    # load exceeds the threshold, select() raises AllWorkersBusy and the
    # frontend returns 503 instead of queueing.
    busy_threshold: float = 0.0     # 0 = disabled
    # sticky sessions (reference parity: lib/llm session_affinity/): cap on
    # remembered session -> worker pins (LRU evicted beyond this).
    max_sessions: int = 4096
    # lower-tier (G2/G3) overlap credit (reference parity: kv-router
    # indexer/lower_tier.rs): a prefix resident in a worker's HOST tier
    # onboards over PCIe instead of being recomputed, so it counts toward
    # the overlap credit at a discount. 0 disables the host indexer.
    host_overlap_weight: float = 0.8


class AllWorkersBusy(Exception):
    """Raised by select() when every worker is above busy_threshold."""


@dataclass
class WorkerState:
    instance_id: str
    active_requests: int = 0        # local in-flight tracking
    active_blocks: int = 0
    kv_usage: float = 0.0
    num_waiting: int = 0
    total_kv_pages: int = 1
    last_metrics_ts: float = 0.0
    event_task: Optional[asyncio.Task] = None
    metrics_task: Optional[asyncio.Task] = None
    # event-driven global load (reference selector/default.rs:217 uses
    # event-derived state, not per-frontend counters): the worker reports
    # running/waiting/blocks after every step; `sent_since_report` counts
    # requests THIS frontend dispatched since the last report (covers the
    # send->schedule window and multi-frontend blindness)
    reported_running: int = -1      # -1 = no report yet (fall back local)
    reported_waiting: int = 0
    reported_blocks: int = 0
    # deltas THIS frontend caused since the last report (sends not yet
    # visible in the report; completions the report predates)
    sent_since_report: int = 0
    sent_blocks_since_report: int = 0
    done_since_report: int = 0
    done_blocks_since_report: int = 0

    @property
    def eff_active_requests(self) -> float:
        if self.reported_running < 0:
            return self.active_requests
        return max(0, self.reported_running + self.reported_waiting
                   + self.sent_since_report - self.done_since_report)

    @property
    def eff_active_blocks(self) -> float:
        if self.reported_running < 0:
            return self.active_blocks
        return max(0, self.reported_blocks + self.sent_blocks_since_report
                   - self.done_blocks_since_report)


class KvRouter:
    """Routes token sequences to worker instances of one component."""

    def __init__(self, runtime: DistributedRuntime, namespace: str,
                 component: str, cfg: RouterConfig | None = None,
                 endpoint: str = "generate"):
        self.runtime = runtime
        self.namespace = namespace
        self.component = component
        self.cfg = cfg or RouterConfig()
        self.client = PushClient(runtime, namespace, component, endpoint)
        self.indexer = _core.KvIndexer()
        # G2/G3 prefix digests per worker (stored_host/removed_host events)
        self.host_indexer = _core.KvIndexer()
        self.workers: Dict[str, WorkerState] = {}
        self._sessions: "OrderedDict[str, str]" = OrderedDict()
        # local worker inhibition (reference: distributed-runtime.md "Local
        # Worker Inhibition", DYN_RUNTIME_INHIBITED_DURATION_SECS): after a
        # failed send, exclude the instance locally until the inhibition
        # expires — discovery-lease expiry alone leaves a window where
        # migration retries re-pick the same dead worker.
        self._inhibited: Dict[str, float] = {}   # iid -> expiry monotonic
        self.inhibit_duration = 30.0
        self._rr = 0
        self._tasks: List[asyncio.Task] = []
        self._started = False

    async def start(self):
        if not self._started:
            self._started = True
            self._tasks.append(asyncio.create_task(self._watch_loop()))
        return self

    async def stop(self):
        for t in self._tasks:
            t.cancel()
        for w in self.workers.values():
            if w.event_task:
                w.event_task.cancel()
            if w.metrics_task:
                w.metrics_task.cancel()

    # -- discovery + event/metrics ingestion ---------------------------
    async def _watch_loop(self):
        while True:
            try:
                self._sync_workers()
            except Exception:
                log.exception("router watch failed")
            await asyncio.sleep(self.cfg.metrics_poll_interval)

    def _sync_workers(self):
        live = {i.instance_id: i for i in self.client.instances()}
        for iid, inst in live.items():
            ws = self.workers.setdefault(iid, WorkerState(iid))
            if ws.event_task is None:  # not yet subscribed (may have been
                # pre-created by begin_request accounting)
                ws.event_task = asyncio.create_task(
                    self._consume_events(inst.address, iid))
                ws.metrics_task = asyncio.create_task(
                    self._consume_metrics(inst.address, iid))
        for iid in list(self.workers):
            if iid not in live:
                ws = self.workers.pop(iid)
                if ws.event_task:
                    ws.event_task.cancel()
                if ws.metrics_task:
                    ws.metrics_task.cancel()
                self.indexer.remove_worker(self._wid(iid))
                self.host_indexer.remove_worker(self._wid(iid))

    @staticmethod
    def _wid(instance_id: str) -> int:
        # hash the FULL instance id: a prefix slice would alias two
        # instances sharing the first hex chars into one indexer worker.
        # Signed 64-bit (the C++ indexer takes int64).
        import hashlib
        v = int.from_bytes(
            hashlib.blake2b(instance_id.encode(), digest_size=8).digest(),
            "big")
        return v - (1 << 64) if v >= (1 << 63) else v

    async def _consume_events(self, address: str, iid: str):
        wid = self._wid(iid)
        try:
            async for batch in self.runtime.client.call_stream(
                    address, f"{self.component}.kv_events", {}):
                for ev in batch:
                    if ev["kind"] == "stored":
                        self.indexer.apply_stored(wid, ev["hashes"])
                    elif ev["kind"] == "removed":
                        self.indexer.apply_removed(wid, ev["hashes"])
                    elif ev["kind"] == "stored_host":
                        self.host_indexer.apply_stored(wid, ev["hashes"])
                    elif ev["kind"] == "removed_host":
                        self.host_indexer.apply_removed(wid, ev["hashes"])
                    elif ev["kind"] == "cleared":
                        self.indexer.clear_worker(wid)
                        self.host_indexer.clear_worker(wid)
        except (ConnectionError, OSError, asyncio.CancelledError):
            pass  # worker died; watch loop will clean up
        except Exception:
            log.exception("kv_events consumer for %s failed", iid)

    def _ingest_metrics(self, iid: str, m: dict):
        ws = self.workers.get(iid)
        if not ws or not m:
            return
        ws.kv_usage = m.get("kv_usage", 0.0)
        ws.num_waiting = m.get("num_waiting", 0)
        ws.total_kv_pages = max(1, m.get("total_kv_pages", 1))
        ws.reported_running = m.get("num_running", -1)
        ws.reported_waiting = m.get("num_waiting", 0)
        ws.reported_blocks = m.get("active_blocks", 0)
        ws.sent_since_report = 0
        ws.sent_blocks_since_report = 0
        ws.done_since_report = 0
        ws.done_blocks_since_report = 0
        ws.last_metrics_ts = time.time()

    async def _consume_metrics(self, address: str, iid: str):
        """Event-driven load state: one snapshot per engine step pushed by
        the worker (reference parity: FPM over the event plane instead of
        router polling). Falls back to 1 Hz get_perf_metrics polling for
        workers without the streaming endpoint."""
        try:
            async for m in self.runtime.client.call_stream(
                    address, f"{self.component}.metrics_events", {}):
                self._ingest_metrics(iid, m)
        except asyncio.CancelledError:
            raise
        except Exception:
            while iid in self.workers:   # polling fallback
                try:
                    m = await self.runtime.client.call(
                        address, f"{self.component}.get_perf_metrics", {})
                    self._ingest_metrics(iid, m)
                except asyncio.CancelledError:
                    raise
                except Exception:
                    pass
                await asyncio.sleep(self.cfg.metrics_poll_interval)

    # -- selection ------------------------------------------------------
    def select(self, token_ids: List[int],
               session_id: Optional[str] = None,
               override: Optional[dict] = None) -> Optional[str]:
        """Pick a worker instance_id for this token sequence.

        session_id pins a session to its previous worker while that worker
        is alive (sticky sessions); busy_threshold > 0 rejects with
        AllWorkersBusy when every worker is overloaded; `override` applies
        per-request router-config overrides (router_config_override parity:
        kv-router/src/scheduling/config.rs) for the cost-formula weights
        and sampling temperature."""
        insts = self.client.instances()
        if not insts:
            return None
        # drop locally-inhibited instances (dead-worker watch-lag window);
        # if that would leave nothing, fall back to the full list
        if self._inhibited:
            now = time.monotonic()
            self._inhibited = {i: t for i, t in self._inhibited.items()
                               if t > now}
            ok = [i for i in insts if i.instance_id not in self._inhibited]
            if ok:
                insts = ok
        if self.cfg.busy_threshold > 0:
            if all(self._load(i.instance_id) > self.cfg.busy_threshold
                   for i in insts):
                raise AllWorkersBusy(
                    f"all {len(insts)} workers above busy threshold "
                    f"{self.cfg.busy_threshold}")
        if session_id is not None:
            alive = {i.instance_id for i in insts}
            pinned = self._sessions.get(session_id)
            if pinned in alive:
                self._sessions.move_to_end(session_id)
                return pinned
            iid = self._select_inner(insts, token_ids, override)
            self._sessions[session_id] = iid
            self._sessions.move_to_end(session_id)
            while len(self._sessions) > self.cfg.max_sessions:
                self._sessions.popitem(last=False)
            return iid
        return self._select_inner(insts, token_ids, override)

    def _select_inner(self, insts, token_ids: List[int],
                      override: Optional[dict] = None) -> Optional[str]:
        cfg = self.cfg
        if override:
            import dataclasses
            allowed = {"mode", "overlap_score_weight",
                       "decode_active_request_weight", "prefill_load_scale",
                       "router_temperature", "host_overlap_weight"}
            cfg = dataclasses.replace(
                cfg, **{k: v for k, v in override.items() if k in allowed})
        mode = cfg.mode
        if mode == "round_robin":
            self._rr += 1
            return insts[self._rr % len(insts)].instance_id
        if mode == "random":
            return random.choice(insts).instance_id
        if mode in ("least_loaded", "p2c"):
            cand = insts if mode == "least_loaded" else random.sample(
                insts, min(2, len(insts)))
            return min(cand, key=lambda i: self._load(i.instance_id)).instance_id

        # kv mode
        bs = cfg.block_size
        hashes = _core.chain_hashes(token_ids, bs, cfg.block_salt)
        matches = self.indexer.find_matches(hashes)
        host_matches = (self.host_indexer.find_matches(hashes)
                        if cfg.host_overlap_weight > 0 else {})
        prefill_blocks = (len(token_ids) + bs - 1) // bs
        logits = []
        for inst in insts:
            ws = self.workers.get(inst.instance_id) or WorkerState(
                inst.instance_id)
            wid = self._wid(inst.instance_id)
            overlap = matches.get(wid, 0)
            # a deeper HOST-resident prefix beats a shallower device one at
            # the onboard discount (lower_tier.rs semantics)
            overlap = max(overlap,
                          cfg.host_overlap_weight * host_matches.get(wid, 0))
            cost = (cfg.prefill_load_scale
                    * max(0.0, prefill_blocks
                          - cfg.overlap_score_weight * overlap)
                    + ws.eff_active_blocks
                    + cfg.decode_active_request_weight
                    * ws.eff_active_requests)
            logits.append(cost)
        T = cfg.router_temperature
        if T <= 0:
            best = min(range(len(insts)), key=lambda i: logits[i])
            return insts[best].instance_id
        # softmax sampling over -cost/T
        ws_ = [math.exp(-(c - min(logits)) / T) for c in logits]
        total = sum(ws_)
        r = random.random() * total
        acc = 0.0
        for i, w in enumerate(ws_):
            acc += w
            if r <= acc:
                return insts[i].instance_id
        return insts[-1].instance_id

    def _load(self, iid: str) -> float:
        ws = self.workers.get(iid)
        if not ws:
            return 0.0
        return ws.eff_active_requests + ws.kv_usage

    def inhibit(self, iid: str, duration: Optional[float] = None):
        """Locally exclude `iid` from selection after a failed send."""
        self._inhibited[iid] = time.monotonic() + (
            duration if duration is not None else self.inhibit_duration)

    # -- request accounting ---------------------------------------------
    def begin_request(self, iid: str, token_ids: List[int]):
        ws = self.workers.setdefault(iid, WorkerState(iid))
        blocks = (len(token_ids) + self.cfg.block_size - 1) // self.cfg.block_size
        ws.active_requests += 1
        ws.active_blocks += blocks
        ws.sent_since_report += 1
        ws.sent_blocks_since_report += blocks

    def end_request(self, iid: str, token_ids: List[int]):
        ws = self.workers.get(iid)
        if ws:
            blocks = (len(token_ids) + self.cfg.block_size - 1) // self.cfg.block_size
            ws.active_requests = max(0, ws.active_requests - 1)
            ws.active_blocks = max(0, ws.active_blocks - blocks)
            ws.done_since_report += 1
            ws.done_blocks_since_report += blocks
