"""PrefillRouter: disaggregated prefill/decode orchestration.

Re-creates the reference's PrefillRouter
(ai-dynamo/dynamo lib/llm/src/kv_router/prefill_router/mod.rs:182):
  1. conditional-disagg check — short net-new prefills (after prefix
     overlap) bypass the prefill pool and run aggregated on the decode
     worker (conditional_bypass.rs:69: bypass when net-new < threshold or
     overlap ratio high);
  2. otherwise select a prefill worker (overlap credit zeroed for the
     decode pool's selector — admission.rs:41), run prefill to first token,
     and inject disaggregated_params into the decode request
     (admission.rs:64 consume_prefill_stream).
"""
from __future__ import annotations

import logging
from typing import AsyncIterator, List, Optional

from dynamo_amd.runtime import DistributedRuntime
from .kv_router import KvRouter, RouterConfig

log = logging.getLogger("dynamo_amd.prefill_router")


class PrefillRouter:
    def __init__(self, runtime: DistributedRuntime, namespace: str,
                 prefill_component: str = "prefill",
                 decode_component: str = "backend",
                 cfg: RouterConfig | None = None,
                 bypass_token_threshold: int | None = None,
                 bypass_overlap_ratio: float = 0.7,
                 bypass_decode_busy_waiting: int = 4):
        self.runtime = runtime
        self.namespace = namespace
        self.cfg = cfg or RouterConfig()
        self.prefill_router = KvRouter(runtime, namespace, prefill_component,
                                       self.cfg)
        self.decode_router = KvRouter(runtime, namespace, decode_component,
                                      self.cfg)
        if bypass_token_threshold is None:
            from dynamo_amd import envs
            bypass_token_threshold = envs.get("DYN_BYPASS_TOKEN_THRESHOLD",
                                              2048, int)
        self.bypass_token_threshold = bypass_token_threshold
        self.bypass_overlap_ratio = bypass_overlap_ratio
        # busy gating (conditional_disagg.rs:111 parity): never bypass onto
        # a decode pool whose queues are already deep
        self.bypass_decode_busy_waiting = bypass_decode_busy_waiting

    async def start(self):
        await self.prefill_router.start()
        await self.decode_router.start()
        return self

    async def stop(self):
        await self.prefill_router.stop()
        await self.decode_router.stop()

    def has_prefill_pool(self) -> bool:
        return bool(self.prefill_router.client.instances())

    def _decode_pool_busy(self) -> bool:
        ws = [w for w in self.decode_router.workers.values()]
        if not ws:
            return False
        return all(w.num_waiting >= self.bypass_decode_busy_waiting
                   for w in ws)

    def _should_bypass(self, token_ids: List[int]) -> bool:
        if not self.has_prefill_pool():
            return True
        if self._decode_pool_busy():
            return False   # busy gating: keep long prefills off decode
        from dynamo_amd import _core
        bs = self.cfg.block_size
        hashes = _core.chain_hashes(token_ids, bs, self.cfg.block_salt)
        matches = self.decode_router.indexer.find_matches(hashes)
        best_overlap_blocks = max(matches.values(), default=0)
        net_new = len(token_ids) - best_overlap_blocks * bs
        if net_new < self.bypass_token_threshold:
            return True
        if best_overlap_blocks * bs / max(1, len(token_ids)) > self.bypass_overlap_ratio:
            return True
        return False

    async def generate(self, payload: dict) -> AsyncIterator[dict]:
        """Full disagg pipeline for one request; yields LLMEngineOutput
        chunks from the decode worker (plus the prefill-produced first
        token)."""
        token_ids = payload["token_ids"]
        if self._should_bypass(token_ids):
            iid = self.decode_router.select(token_ids)
            self.decode_router.begin_request(iid, token_ids)
            try:
                async for chunk in self.decode_router.client.generate(
                        payload, instance_id=iid):
                    yield chunk
            finally:
                self.decode_router.end_request(iid, token_ids)
            return

        # 1) prefill
        p_iid = self.prefill_router.select(token_ids)
        self.prefill_router.begin_request(p_iid, token_ids)
        disagg = None
        try:
            async for chunk in self.prefill_router.client.generate(
                    payload, instance_id=p_iid):
                disagg = chunk.get("disaggregated_params") or disagg
        finally:
            self.prefill_router.end_request(p_iid, token_ids)
        if disagg is None:
            raise RuntimeError("prefill worker returned no "
                               "disaggregated_params")
        # first token produced by prefill
        if disagg.get("first_token") is not None:
            yield {"token_ids": [disagg["first_token"]]}

        # 2) decode with injected prefill_result
        d_iid = self.decode_router.select(token_ids)
        d_payload = dict(payload)
        d_payload["prefill_result"] = disagg
        self.decode_router.begin_request(d_iid, token_ids)
        try:
            async for chunk in self.decode_router.client.generate(
                    d_payload, instance_id=d_iid):
                yield chunk
        finally:
            self.decode_router.end_request(d_iid, token_ids)
