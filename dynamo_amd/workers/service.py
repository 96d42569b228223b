"""WorkerService: serves an engine (native or mock) on the request plane.

Fulfills the worker contract the reference defines for its engine adapters
(components/src/dynamo/vllm/main.py:149 worker()):
  - registers a model card + runtime config in discovery
  - serves endpoints: generate, clear_kv_blocks, get_perf_metrics,
    kv_events (stream), release_kv (disagg), lora stubs
  - prefill workers finish at 1 token, hold KV and return
    disaggregated_params; decode workers pull KV over xGMI before decoding

Request payload (PreprocessedRequest parity,
lib/llm/src/protocols/common/preprocessor.rs:243):
  {request_id, token_ids, sampling_options{temperature, top_p, top_k, seed},
   stop_conditions{max_tokens, stop_token_ids, ignore_eos},
   routing{worker_instance_id?}, prefill_result?, annotations?}
Response chunks (LLMEngineOutput parity, llm_backend.rs:163):
  {token_ids: [t], finish_reason?, disaggregated_params?}
"""
from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, AsyncIterator, Dict, Optional

from dynamo_amd.engine.engine import LLMEngine
from dynamo_amd.engine.scheduler import Request, SamplingParams
from dynamo_amd.engine.kv_cache import SequenceKV
from dynamo_amd.observability import trace_event
from dynamo_amd.runtime import DistributedRuntime, RequestContext
from dynamo_amd.disagg.transfer import KvPuller, pool_transfer_metadata

log = logging.getLogger("dynamo_amd.worker")


def make_sampling(payload: dict) -> SamplingParams:
    so = payload.get("sampling_options") or {}
    sc = payload.get("stop_conditions") or {}
    return SamplingParams(
        max_tokens=int(sc.get("max_tokens", 128)),
        temperature=float(so.get("temperature", 0.0)),
        top_p=float(so.get("top_p", 1.0)),
        top_k=int(so.get("top_k", 0)),
        stop_token_ids=list(sc.get("stop_token_ids", [])),
        ignore_eos=bool(sc.get("ignore_eos", False)),
        seed=int(so.get("seed", 0)),
        embed=bool(so.get("embed", False)),
        logprobs=int(so.get("logprobs", 0)),
    )


class WorkerService:
    def __init__(self, engine: LLMEngine, runtime: DistributedRuntime,
                 namespace: str = "dynamo", component: str = "backend",
                 model_name: Optional[str] = None,
                 kv_transfer_meta: Optional[dict] = None):
        # kv_transfer_meta: pre-gathered TP-group pool metadata (built by
        # the worker CLI with a collective BEFORE the event loop starts)
        self._kv_transfer_meta = kv_transfer_meta
        self.engine = engine
        self.runtime = runtime
        self.namespace = namespace
        self.component_name = component
        self.model_name = model_name or engine.cfg.model.name
        self.comp = runtime.namespace(namespace).component(component)
        self.queues: Dict[str, asyncio.Queue] = {}
        self.kv_event_subs: list = []
        self.metrics_subs: list = []
        self._work = asyncio.Event()
        self._loop_task: Optional[asyncio.Task] = None
        self._engine_lock = asyncio.Lock()
        self._req_counter = 0
        self._puller: Optional[KvPuller] = None
        self.worker_type = engine.cfg.worker_type
        self._paused = asyncio.Event()
        self._paused.set()  # set = running

    # ------------------------------------------------------------------
    def _mdc_sum(self, card: dict) -> str:
        import hashlib
        import json
        return hashlib.blake2b(
            json.dumps(card, sort_keys=True).encode(),
            digest_size=8).hexdigest()

    def model_card(self) -> dict:
        cfg = self.engine.cfg
        card = self._card_body()
        card["mdc_sum"] = self._mdc_sum(card)
        return card

    def _card_body(self) -> dict:
        cfg = self.engine.cfg
        body = self._card_extras()
        body.update({
            "name": self.model_name,
            "model_config": cfg.model.to_dict(),
            "context_length": cfg.max_model_len,
            "kv_cache_block_size": cfg.page_size,
            "block_salt": cfg.block_salt,
            "runtime_config": {
                "total_kv_blocks": self.engine.alloc.num_pages,
                "max_num_seqs": cfg.max_num_seqs,
                "max_num_batched_tokens": cfg.max_batched_tokens,
                "worker_type": self.worker_type,
            },
        })
        return body

    def _card_extras(self) -> dict:
        """Tokenizer + chat template advertised from a local HF checkpoint
        dir (single-node: the frontend shares the filesystem). Falls back
        to the byte tokenizer + default template when absent."""
        import json as _json
        import os as _os
        wp = getattr(self.engine.cfg.model, "weights_path", "")
        out: dict = {}
        if not wp:
            return out
        tok = _os.path.join(wp, "tokenizer.json")
        if _os.path.exists(tok):
            out["tokenizer"] = {"type": "hf", "path": tok}
        gcfg = _os.path.join(wp, "generation_config.json")
        if _os.path.exists(gcfg):
            try:
                with open(gcfg) as f:
                    eos = _json.load(f).get("eos_token_id")
                if eos is not None:
                    out["eos_token_ids"] = (eos if isinstance(eos, list)
                                            else [eos])
            except (OSError, ValueError):
                pass
        tcfg = _os.path.join(wp, "tokenizer_config.json")
        if _os.path.exists(tcfg):
            try:
                with open(tcfg) as f:
                    tmpl = _json.load(f).get("chat_template")
                if tmpl:
                    out["chat_template"] = tmpl
            except (OSError, ValueError):
                pass
        return out

    async def start(self):
        self.comp.serve_endpoint("generate", self.generate)
        self.comp.serve_endpoint("clear_kv_blocks", self.clear_kv_blocks)
        self.comp.serve_endpoint("get_perf_metrics", self.get_perf_metrics)
        self.comp.serve_endpoint("metrics_events", self.metrics_events)
        self.comp.serve_endpoint("kv_events", self.kv_events)
        self.comp.serve_endpoint("release_kv", self.release_kv)
        self.comp.serve_endpoint("pause", self.pause)
        self.comp.serve_endpoint("resume", self.resume)
        self.comp.serve_endpoint("load_lora", self.load_lora)
        self.comp.serve_endpoint("unload_lora", self.unload_lora)
        self.comp.serve_endpoint("list_loras", self.list_loras)
        self.comp.serve_endpoint("update_weights", self.update_weights)
        metadata = {"worker_type": self.worker_type}
        if self._kv_transfer_meta is not None:
            # TP group: every rank's pool handle, gathered by the CLI
            # (decode rank r pulls from prefill rank r; page ids agree via
            # lockstep determinism)
            metadata["kv_transfer"] = self._kv_transfer_meta
            metadata["tp_size"] = self._kv_transfer_meta.get("tp_size", 1)
        elif self.engine.runner.kv_pool is not None:
            metadata["kv_transfer"] = pool_transfer_metadata(
                self.comp.instance_id, self.engine.runner.kv_pool)
        await self.comp.register(model_card=self.model_card(),
                                 metadata=metadata)
        self._loop_task = asyncio.create_task(self._engine_loop())
        return self

    async def stop(self, drain: bool = True, drain_timeout: float = 10.0):
        """Stop serving. Default is GRACEFUL (reference parity:
        push_endpoint.rs:46-56 inflight counter +
        graceful-shutdown-architecture.md): deregister from discovery
        first (no new routing), then let in-flight requests finish before
        killing the engine loop. drain=False aborts immediately (tests /
        fault injection)."""
        self.comp.deregister()
        if drain and self.engine.has_work():
            deadline = asyncio.get_event_loop().time() + drain_timeout
            while (self.engine.has_work()
                   and asyncio.get_event_loop().time() < deadline):
                await asyncio.sleep(0.02)
        if self._loop_task:
            self._loop_task.cancel()

    @property
    def instance_id(self) -> str:
        return self.comp.instance_id

    # ------------------------------------------------------------------
    async def _engine_loop(self):
        from dynamo_amd.engine.engine import StepOutput
        while True:
            await self._paused.wait()  # snapshot lifecycle pause
            if not self.engine.has_work():
                self._work.clear()
                await self._work.wait()
                await self._paused.wait()
            try:
                async with self._engine_lock:
                    outputs = await asyncio.to_thread(self.engine.step)
            except Exception:
                # a step exception must not kill the loop: abort the
                # requests that were in flight (clients get a clean
                # error-finish) and keep serving
                log.exception("engine step failed; aborting in-flight work")
                async with self._engine_lock:
                    self._abort_inflight("error")
                continue
            for so in outputs:
                q = self.queues.get(so.req_id)
                if q is not None:
                    q.put_nowait(so)
            self._fan_kv_events()
            self._fan_metrics()
            await asyncio.sleep(0)

    def _abort_inflight(self, reason: str):
        """Abort every scheduled request, pushing a finished StepOutput so
        waiting generate() handlers terminate their streams."""
        from dynamo_amd.engine.engine import StepOutput
        for rid in list(self.engine.requests):
            try:
                self.engine.abort(rid)
            except Exception:
                log.exception("abort %s failed", rid)
            q = self.queues.get(rid)
            if q is not None:
                q.put_nowait(StepOutput(rid, None, True, reason))

    async def _drain_idle(self, timeout: float) -> bool:
        """Wait until the engine has no scheduled work. True if idle."""
        deadline = asyncio.get_event_loop().time() + timeout
        while (self.engine.has_work()
               and asyncio.get_event_loop().time() < deadline):
            await asyncio.sleep(0.01)
        return not self.engine.has_work()

    def _fan_kv_events(self):
        """Drain engine KV events to subscribers — called from the engine
        loop AND from endpoints that mutate KV while the engine is idle
        (clear_kv_blocks/release_kv), else their events would stall until
        the next step."""
        events = self.engine.drain_kv_events()
        if events and self.kv_event_subs:
            batch = [{"kind": e.kind, "hashes": list(e.hashes),
                      "parent": e.parent} for e in events]
            for q in self.kv_event_subs:
                q.put_nowait(batch)

    def _metrics_snapshot(self) -> dict:
        m = self.engine.last_metrics
        return {
            "worker_id": self.instance_id,
            "worker_type": self.worker_type,
            "step": m.step,
            "num_running": self.engine.scheduler.num_running(),
            "num_waiting": self.engine.scheduler.num_waiting(),
            "kv_usage": self.engine.alloc.usage,
            "total_kv_pages": self.engine.alloc.num_pages,
            "active_blocks": (self.engine.alloc.num_pages
                              - len(self.engine.alloc.free)
                              - len(self.engine.alloc.evictable)),
            "num_tokens_step": m.num_tokens_step,
            "prefill_tokens_step": m.prefill_tokens_step,
            "decode_tokens_step": m.decode_tokens_step,
            "step_time_ms": m.step_time_ms,
            "ts": time.time(),
        }

    def _fan_metrics(self):
        """Push a post-step metrics snapshot to subscribers (event-driven
        load state for routers — the reference fans ForwardPassMetrics
        through its event plane instead of letting routers poll,
        components/src/dynamo/common/forward_pass_metrics.py:14-28).
        Replace-don't-queue: only the latest snapshot matters."""
        if not self.metrics_subs:
            return
        snap = self._metrics_snapshot()
        for q in self.metrics_subs:
            while not q.empty():   # drop stale unconsumed snapshots
                try:
                    q.get_nowait()
                except asyncio.QueueEmpty:
                    break
            q.put_nowait(snap)

    async def metrics_events(self, payload, ctx):
        """Streaming FPM subscription (one snapshot per engine step,
        coalesced to the latest while the subscriber is slow)."""
        q: asyncio.Queue = asyncio.Queue()
        self.metrics_subs.append(q)
        ctx.on_cancel(lambda: q.put_nowait(None))
        try:
            yield self._metrics_snapshot()   # immediate state on subscribe
            while not ctx.cancelled:
                snap = await q.get()
                if snap is None:
                    break
                yield snap
        finally:
            self.metrics_subs.remove(q)

    # ------------------------------------------------------------------
    async def generate(self, payload: dict, ctx: RequestContext
                       ) -> AsyncIterator[dict]:
        want = payload.get("mdc_sum")
        if want is not None:
            # model-deployment-card checksum (PreprocessedRequest mdc_sum
            # parity): reject requests preprocessed against a STALE card
            have = self.model_card()["mdc_sum"]
            if want != have:
                raise RuntimeError(
                    f"mdc_sum mismatch: request {want} != worker {have} "
                    "(frontend holds a stale model card)")
        if payload.get("_HEALTH_CHECK"):
            # canary parity (PreprocessedRequest _HEALTH_CHECK flag): verify
            # the engine loop is alive without generating anything
            yield {"token_ids": [], "health": "ok",
                   "worker_type": self.worker_type,
                   "num_running": self.engine.scheduler.num_running()}
            return
        self._req_counter += 1
        req_id = payload.get("request_id") or f"req-{self._req_counter}"
        tokens = list(payload["token_ids"])
        sp = make_sampling(payload)
        is_prefill_role = self.worker_type == "prefill"
        if is_prefill_role:
            sp = SamplingParams(max_tokens=1, temperature=sp.temperature,
                                top_p=sp.top_p, top_k=sp.top_k,
                                seed=sp.seed, ignore_eos=True)

        trace_event("handle_payload", request_id=req_id,
                    worker_id=self.instance_id, worker_type=self.worker_type,
                    prompt_tokens=len(tokens))
        q: asyncio.Queue = asyncio.Queue()
        self.queues[req_id] = q
        completed = False
        def _decode_tensor(spec):
            import base64
            import numpy as np
            import torch
            arr = np.frombuffer(base64.b64decode(spec["b64"]),
                                dtype=np.dtype(spec.get("dtype", "float16")))
            return torch.from_numpy(arr.reshape(spec["shape"]).copy()).float()

        pe = payload.get("prompt_embeds")
        prompt_embeds = None
        if pe is not None:
            # PreprocessedRequest prompt_embeds (b64) parity
            prompt_embeds = _decode_tensor(pe)
        # multimodal embedding spans: [{"offset", "b64", "shape", "dtype"}]
        mm = payload.get("mm_embeds") or []
        embed_spans = [(int(s["offset"]), _decode_tensor(s)) for s in mm]
        try:
            pr = payload.get("prefill_result")
            async with self._engine_lock:
                if pr is not None:
                    # disagg decode side: attach + pull KV (TP engines
                    # broadcast the same spec to every rank)
                    req = await self._attach_remote_kv(req_id, tokens, sp,
                                                       pr)
                else:
                    req = self.engine.add_request(
                        req_id, tokens, sp, prompt_embeds=prompt_embeds)
                    if embed_spans:
                        req.embed_spans = embed_spans
                    if is_prefill_role:
                        req.hold_kv = True
            self._work.set()
            # event-driven: a cancel frame wakes the queue wait via a
            # sentinel — no polling timeouts (reference behavior:
            # push_endpoint responds to cancellation immediately)
            ctx.on_cancel(lambda: q.put_nowait(None))

            while True:
                so = await q.get()
                if so is None or ctx.cancelled:
                    async with self._engine_lock:
                        self.engine.abort(req_id)
                    return
                chunk: dict = {"token_ids": ([so.new_token]
                                             if so.new_token is not None
                                             else [])}
                if so.embedding is not None:
                    chunk["embedding"] = so.embedding
                if so.logprobs is not None:
                    chunk["logprobs"] = [so.logprobs]
                # coalesce: if the consumer fell behind the engine, fold
                # every already-queued plain token into ONE frame instead
                # of sending one frame per token
                while (not so.finished and so.logprobs is None
                       and so.embedding is None and not q.empty()):
                    nxt = q.get_nowait()
                    if nxt is None:
                        q.put_nowait(None)   # re-deliver cancel sentinel
                        break
                    if nxt.embedding is not None or nxt.logprobs is not None:
                        chunk["embedding"] = nxt.embedding
                        if nxt.logprobs is not None:
                            chunk.setdefault("logprobs", []).append(
                                nxt.logprobs)
                    if nxt.new_token is not None:
                        chunk["token_ids"].append(nxt.new_token)
                    so = nxt
                if so.finished:
                    chunk["finish_reason"] = so.finish_reason
                    if is_prefill_role:
                        chunk["disaggregated_params"] = \
                            self._disagg_params(req)
                    completed = True
                    yield chunk
                    return
                yield chunk
        finally:
            self.queues.pop(req_id, None)
            if not completed:
                # stream ended early (cancel frame, client disconnect,
                # handler close): stop generating for this request
                async with self._engine_lock:
                    self.engine.abort(req_id)

    def _disagg_params(self, req: Request) -> dict:
        return {
            "prefill_instance_id": self.instance_id,
            "prefill_component": self.component_name,
            "page_ids": list(req.kv.pages),
            "num_tokens": req.num_computed,
            "first_token": req.output_tokens[0] if req.output_tokens else None,
        }

    async def _attach_remote_kv(self, req_id: str, tokens, sp, pr: dict
                                ) -> Request:
        """Decode side of the disagg handoff: add the request and pull its
        KV pages over xGMI (every TP rank pulls its own shard from the
        matching prefill rank — parallel/tp.attach_remote)."""
        insts = self.runtime.discovery.list(self.namespace)
        src_meta = None
        src_addr = None
        for inst in insts:
            if inst.instance_id == pr["prefill_instance_id"]:
                src_meta = inst.metadata.get("kv_transfer")
                src_addr = inst.address
                break
        if src_meta is None:
            raise RuntimeError(
                f"prefill instance {pr['prefill_instance_id']} not found")
        spec = {
            "request_id": req_id,
            "token_ids": list(tokens),
            "sampling": sp.__dict__.copy(),
            "num_tokens": int(pr["num_tokens"]),
            "page_ids": [int(p) for p in pr["page_ids"]],
            "first_token": pr.get("first_token"),
            "src_meta": src_meta,
            "arrival": time.monotonic(),
        }
        req = await asyncio.to_thread(self.engine.attach_request, spec)
        # release the prefill side's hold
        try:
            await self.runtime.client.call(
                src_addr, f"{pr.get('prefill_component', 'prefill')}."
                "release_kv", {"request_id": req_id})
        except Exception:
            log.warning("release_kv to %s failed",
                        pr["prefill_instance_id"])
        return req

    # ------------------------------------------------------------------
    async def clear_kv_blocks(self, payload, ctx):
        # the allocator reset is only safe with no in-flight sequences
        # (their SequenceKV.release would decref fresh refcounts); drain
        # first and REFUSE if live traffic keeps the engine busy
        if not await self._drain_idle(float(payload.get("drain_timeout", 5.0))
                                      if isinstance(payload, dict) else 5.0):
            if not (isinstance(payload, dict) and payload.get("force")):
                yield {"status": "busy",
                       "num_running": self.engine.scheduler.num_running()}
                return
            async with self._engine_lock:
                self._abort_inflight("abort")
        async with self._engine_lock:
            self.engine.clear_kv()
        self._fan_kv_events()
        yield {"status": "ok"}

    async def get_perf_metrics(self, payload, ctx):
        yield self._metrics_snapshot()

    async def kv_events(self, payload, ctx):
        """Streaming subscription: batches of KV events (router feed).

        On subscribe, the current cached-block state is replayed as one
        `stored` batch so late subscribers converge (the reference's router
        rebuilds state from the event stream + dedup refcounts)."""
        q: asyncio.Queue = asyncio.Queue()
        async with self._engine_lock:
            snapshot = list(self.engine.alloc.hash_to_page.keys())
            self.kv_event_subs.append(q)
        if snapshot:
            yield [{"kind": "stored", "hashes": snapshot, "parent": None}]
        ctx.on_cancel(lambda: q.put_nowait(None))
        try:
            while not ctx.cancelled:
                batch = await q.get()
                if batch is None:
                    break
                # coalesce any backlog into one frame
                while not q.empty():
                    more = q.get_nowait()
                    if more is None:
                        q.put_nowait(None)
                        break
                    batch.extend(more)
                yield batch
        finally:
            self.kv_event_subs.remove(q)

    async def release_kv(self, payload, ctx):
        async with self._engine_lock:
            self.engine.release_held(payload["request_id"])
        self._fan_kv_events()
        yield {"status": "ok"}

    # -- snapshot lifecycle: pause -> (snapshot externally) -> resume ----
    # (reference: common/snapshot/lifecycle.py:32-108 + engine pause
    # controller handlers.py:339; with GMS the weights already live in an
    # external process, so pause/resume is the whole engine-side story)
    async def pause(self, payload, ctx):
        self._paused.clear()
        async with self._engine_lock:
            pass  # wait for the in-flight step to finish
        yield {"status": "paused",
               "in_flight": self.engine.scheduler.num_running()}

    async def resume(self, payload, ctx):
        self._paused.set()
        self._work.set()
        yield {"status": "running"}

    # -- RL weight-update surface (reference: lib/rl/src/lib.rs:4-16 —
    # frontend discovers workers' rl admin endpoints for weight pushes).
    # Applies a deterministic in-place delta (seeded noise scaled by
    # `scale`, modeling an RL policy push without checkpoint files),
    # invalidates captured graphs and flushes the KV/prefix cache (old
    # cache entries were computed under the old weights).
    async def update_weights(self, payload, ctx):
        seed = int(payload.get("seed", 0))
        scale = float(payload.get("scale", 0.01))
        # a weight push invalidates in-flight work: drain briefly, then
        # abort whatever is still running (old-weight KV must not mix with
        # new-weight decode steps)
        if not await self._drain_idle(float(payload.get("drain_timeout",
                                                        5.0))):
            async with self._engine_lock:
                self._abort_inflight("abort")
        async with self._engine_lock:
            n = await asyncio.to_thread(self.engine.apply_weight_delta,
                                        seed, scale)
        yield {"status": "ok", "tensors_updated": n, "seed": seed}

    # -- LoRA endpoints (reference: vllm/worker_factory.py:1378-1413) ----
    async def load_lora(self, payload, ctx):
        async with self._engine_lock:
            await asyncio.to_thread(
                self.engine.load_lora, payload["name"],
                payload.get("path"), int(payload.get("rank", 8)),
                float(payload.get("alpha", 16.0)), int(payload.get("seed", 0)))
            self.comp.update_metadata(loras=self.engine.list_loras())
        yield {"status": "ok", "loras": self.engine.list_loras()}

    async def unload_lora(self, payload, ctx):
        async with self._engine_lock:
            self.engine.unload_lora(payload["name"])
            self.comp.update_metadata(loras=self.engine.list_loras())
        yield {"status": "ok", "loras": self.engine.list_loras()}

    async def list_loras(self, payload, ctx):
        yield {"loras": self.engine.list_loras()}
