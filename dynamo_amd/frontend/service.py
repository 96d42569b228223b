"""Frontend model management + request pipeline.

The Python analog of the reference's frontend assembly
(ai-dynamo/dynamo lib/llm/src/entrypoint/input/common.rs:523-535 pipeline
Frontend -> Preprocessor -> Migration -> Backend -> PrefillRouter ->
ServiceBackend, and discovery/watcher.rs:184 ModelWatcher): watches
discovery for model cards, builds a per-model router (KV-aware, with
disagg orchestration when a prefill pool exists), tokenizes/detokenizes,
and retries/migrates streams on worker death (migration.rs:130 parity —
delivered tokens are replayed to the new worker).
"""
from __future__ import annotations

import asyncio
import logging
import time
import uuid
from dataclasses import dataclass, field
from typing import AsyncIterator, Dict, List, Optional

from dynamo_amd.observability import span, trace_event
from dynamo_amd.runtime import DistributedRuntime, EndpointError, NoInstancesError
from dynamo_amd.router import KvRouter, PrefillRouter, RouterConfig
from .tokenizer import ChatTemplater, make_tokenizer

log = logging.getLogger("dynamo_amd.frontend")


@dataclass
class ModelEntry:
    name: str
    card: dict
    component: str
    router: KvRouter = None
    prefill_router: Optional[PrefillRouter] = None
    tokenizer: object = None
    templater: ChatTemplater = None
    migration_limit: int = 3


class ModelManager:
    """Watches discovery; maintains per-model routers + tokenizers."""

    def __init__(self, runtime: DistributedRuntime, namespace: str = "dynamo",
                 router_cfg: RouterConfig | None = None,
                 record_path: Optional[str] = None):
        self.runtime = runtime
        self.namespace = namespace
        self.router_cfg = router_cfg or RouterConfig()
        self.models: Dict[str, ModelEntry] = {}
        self._watch_task: Optional[asyncio.Task] = None
        self.request_count = 0
        # request recording (reference parity: lib/llm recorder.rs +
        # components replay): JSONL of {request, chunks, latency}
        self.record_path = record_path
        self._record_fh = None

    def _record(self, rec: dict):
        import json
        if self._record_fh is None:
            self._record_fh = open(self.record_path, "a")
        self._record_fh.write(json.dumps(rec) + "\n")
        self._record_fh.flush()

    async def start(self, watch_interval: float = 1.0):
        await self.runtime.start()
        await self.refresh()
        self._watch_task = asyncio.create_task(self._watch(watch_interval))
        return self

    async def stop(self):
        if self._watch_task:
            self._watch_task.cancel()
        for e in self.models.values():
            if e.router:
                await e.router.stop()
            if e.prefill_router:
                await e.prefill_router.stop()
        if self._record_fh is not None:
            self._record_fh.close()
            self._record_fh = None

    async def _watch(self, interval: float):
        while True:
            try:
                await self.refresh()
            except Exception:
                log.exception("model watch failed")
            await asyncio.sleep(interval)

    async def refresh(self):
        insts = self.runtime.discovery.list(self.namespace)
        # group instances by model name
        seen: Dict[str, dict] = {}
        for inst in insts:
            card = inst.model_card
            if not card:
                continue
            name = card["name"]
            wt = (card.get("runtime_config") or {}).get("worker_type",
                                                        "aggregated")
            info = seen.setdefault(name, {"card": card, "components": {}})
            info["components"][inst.component] = wt

        for name, info in seen.items():
            if name in self.models:
                entry = self.models[name]
            else:
                card = info["card"]
                decode_comp = None
                prefill_comp = None
                for comp, wt in info["components"].items():
                    if wt == "prefill":
                        prefill_comp = comp
                    else:
                        decode_comp = comp
                if decode_comp is None:
                    continue  # only prefill workers so far
                rc = RouterConfig(
                    mode=self.router_cfg.mode,
                    block_size=card.get("kv_cache_block_size", 64),
                    block_salt=card.get("block_salt", 0),
                    router_temperature=self.router_cfg.router_temperature)
                entry = ModelEntry(
                    name=name, card=card, component=decode_comp,
                    router=KvRouter(self.runtime, self.namespace, decode_comp,
                                    rc),
                    tokenizer=make_tokenizer(
                        card.get("tokenizer")
                        or {"type": "byte",
                            "vocab_size": card["model_config"]["vocab_size"]}),
                    templater=ChatTemplater(card.get("chat_template")),
                )
                await entry.router.start()
                self.models[name] = entry
            # (re)wire disagg when a prefill pool appears
            pf = [c for c, wt in info["components"].items() if wt == "prefill"]
            if pf and self.models[name].prefill_router is None:
                e = self.models[name]
                e.prefill_router = PrefillRouter(
                    self.runtime, self.namespace, prefill_component=pf[0],
                    decode_component=e.component, cfg=e.router.cfg)
                await e.prefill_router.start()
        # drop models with no live instances
        for name in list(self.models):
            if name not in seen:
                e = self.models.pop(name)
                await e.router.stop()

    # -- multimodal: encode workers (E/PD disaggregation) ---------------
    def encoder_instances(self, model: str):
        """Live encode workers feeding `model` (reference parity:
        EncoderRouter, lib/llm/src/kv_router/encoder_router.rs)."""
        out = []
        for inst in self.runtime.discovery.list(self.namespace, "encoder"):
            m = inst.metadata or {}
            if m.get("worker_type") == "encoder" and (
                    not m.get("model") or m["model"] == model):
                out.append(inst)
        return out

    async def encode_images(self, model: str, images: list) -> list:
        """Round-robin an encode worker; returns embedding specs
        [{"b64","shape","dtype"}]. Raises if no encoder is registered."""
        insts = self.encoder_instances(model)
        if not insts:
            raise RuntimeError(f"no encode workers for model {model!r}")
        self._enc_rr = getattr(self, "_enc_rr", 0) + 1
        inst = insts[self._enc_rr % len(insts)]
        r = await self.runtime.client.call(
            inst.address, "encoder.encode", {"images": images})
        return r["embeddings"]

    def encoder_tokens_per_image(self, model: str) -> int:
        insts = self.encoder_instances(model)
        return int(insts[0].metadata.get("tokens_per_image", 0)) if insts else 0

    # ------------------------------------------------------------------
    def get(self, model: str) -> ModelEntry:
        if model in self.models:
            return self.models[model]
        if len(self.models) == 1:
            return next(iter(self.models.values()))
        raise KeyError(f"model {model!r} not found; have {list(self.models)}")

    async def generate_tokens(self, entry: ModelEntry, token_ids: List[int],
                              sampling: dict, stop: dict,
                              request_id: Optional[str] = None,
                              session_id: Optional[str] = None,
                              extra: Optional[dict] = None
                              ) -> AsyncIterator[dict]:
        """Route + stream with migration retry (replays delivered tokens)."""
        self.request_count += 1
        rid = request_id or f"{uuid.uuid4().hex[:16]}"
        delivered: List[int] = []
        recorded: List[dict] = []
        t_start = time.time()
        attempts = 0
        trace_event("request_start", request_id=rid, model=entry.name,
                    prompt_tokens=len(token_ids))
        while True:
            payload = {
                "request_id": rid if not attempts else f"{rid}-m{attempts}",
                "token_ids": list(token_ids) + delivered,
                "sampling_options": sampling,
                "stop_conditions": dict(
                    stop, max_tokens=max(1, stop.get("max_tokens", 128)
                                         - len(delivered))),
                "annotations": {"trace_id": rid},
            }
            if extra:
                payload.update(extra)
            try:
                # direct routing hint (reference parity: RoutingHints
                # backend_instance_id / RouterMode::Direct): bypass the
                # KV-aware selection and pin to the named instance
                hint = ((extra or {}).get("routing") or {}).get(
                    "backend_instance_id")
                # embedding requests are prefill-only: no P/D split
                if hint is not None:
                    alive = {i.instance_id
                             for i in entry.router.client.instances()}
                    if hint not in alive:
                        raise NoInstancesError(
                            f"pinned instance {hint} is not alive")
                    entry.router.begin_request(hint, payload["token_ids"])
                    gen = self._direct_gen(entry, payload, hint)
                elif (not sampling.get("embed")
                        and entry.prefill_router is not None
                        and entry.prefill_router.has_prefill_pool()):
                    gen = entry.prefill_router.generate(payload)
                else:
                    iid = entry.router.select(
                        payload["token_ids"], session_id=session_id,
                        override=(extra or {}).get("router_config_override"))
                    if iid is None:
                        raise NoInstancesError(f"no workers for {entry.name}")
                    entry.router.begin_request(iid, payload["token_ids"])
                    gen = self._direct_gen(entry, payload, iid)
                async for chunk in gen:
                    for t in chunk.get("token_ids", []):
                        delivered.append(t)
                    if self.record_path:
                        recorded.append(chunk)
                    yield chunk
                trace_event("request_end", request_id=rid,
                            output_tokens=len(delivered), attempts=attempts)
                if self.record_path:
                    self._record({"ts": t_start, "request_id": rid,
                                  "model": entry.name,
                                  "token_ids": list(token_ids),
                                  "sampling": sampling, "stop": stop,
                                  "latency_s": time.time() - t_start,
                                  "chunks": recorded})
                return
            except (EndpointError, ConnectionError, OSError) as e:
                attempts += 1
                if attempts > entry.migration_limit:
                    raise
                log.warning("stream migration for %s (attempt %d): %s",
                            rid, attempts, e)
                await asyncio.sleep(0.05)

    async def _direct_gen(self, entry, payload, iid):
        try:
            async for chunk in entry.router.client.generate(
                    payload, instance_id=iid):
                yield chunk
        except (EndpointError, ConnectionError, OSError):
            # locally inhibit the failed instance so the migration retry
            # cannot re-pick it during the discovery watch-lag window
            entry.router.inhibit(iid)
            raise
        finally:
            entry.router.end_request(iid, payload["token_ids"])
