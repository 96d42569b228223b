"""OpenAI-compatible HTTP frontend (FastAPI).

Re-creates the serving surface of the reference's Axum frontend
(ai-dynamo/dynamo lib/llm/src/http/service/openai.rs:3875-4165):
/v1/models, /v1/completions, /v1/chat/completions (SSE streaming and
unary), /health, /metrics (Prometheus). Client disconnects propagate as
stream cancellation (http/service/disconnect.rs parity).
"""
from __future__ import annotations

import asyncio
import json
import logging
import time
import uuid
from typing import AsyncIterator, List, Optional, Union

from fastapi import FastAPI, HTTPException, Request, WebSocket
from fastapi.responses import JSONResponse, StreamingResponse
from pydantic import BaseModel, Field

from prometheus_client import (CONTENT_TYPE_LATEST, Counter, Histogram,
                               generate_latest)

from .service import ModelManager

log = logging.getLogger("dynamo_amd.frontend")

REQS = Counter("dynamo_amd_requests_total", "requests", ["model", "route"])
TTFT = Histogram("dynamo_amd_ttft_seconds", "time to first token", ["model"])
LATENCY = Histogram("dynamo_amd_request_seconds", "request latency", ["model"])
# inter-token latency per streamed chunk gap (reference parity:
# http/service/metrics.rs per-model latency depth)
ITL = Histogram("dynamo_amd_itl_seconds", "inter-token latency", ["model"],
                buckets=(.005, .01, .02, .03, .05, .075, .1, .15, .25, .5,
                         1.0, 2.5))
OUT_TOKENS = Counter("dynamo_amd_output_tokens_total", "output tokens",
                     ["model"])


class CompletionRequest(BaseModel):
    model: str = ""
    prompt: Union[str, List[int]] = ""
    max_tokens: Optional[int] = None
    temperature: Optional[float] = None
    top_p: float = 1.0
    top_k: int = 0
    stream: bool = False
    stop: Optional[Union[str, List[str]]] = None
    seed: int = 0
    ignore_eos: bool = False
    logprobs: Optional[int] = None  # top-N logprobs per token
    # PreprocessedRequest prompt_embeds parity: {"b64", "shape", "dtype"}
    prompt_embeds: Optional[dict] = None
    user: Optional[str] = None      # sticky-session key


class ChatMessage(BaseModel):
    role: str
    # plain text, or OpenAI content parts:
    # [{"type": "text", "text": ...},
    #  {"type": "image_url", "image_url": {"url": "data:image/png;base64,..."}}]
    content: Union[str, List[dict]] = ""


class EmbeddingRequest(BaseModel):
    model: str = ""
    input: Union[str, List[str], List[int], List[List[int]]] = ""
    encoding_format: str = "float"


class ChatRequest(BaseModel):
    model: str = ""
    messages: List[ChatMessage] = Field(default_factory=list)
    tools: Optional[List[dict]] = None   # enables tool-call parsing
    max_tokens: Optional[int] = None
    temperature: Optional[float] = None
    top_p: float = 1.0
    top_k: int = 0
    stream: bool = False
    stop: Optional[Union[str, List[str]]] = None
    seed: int = 0
    ignore_eos: bool = False
    user: Optional[str] = None      # sticky-session key
    # next-turn KV warming (reference parity: preprocessor/
    # speculative_prefill.rs nvext.agent_hints.speculative_prefill): after
    # the response completes, fire a background max_tokens=1 request with
    # the re-rendered conversation (assistant turn included, no generation
    # prompt) so the NEXT user turn hits a warm prefix.
    speculative_prefill: Optional[bool] = None
    nvext: Optional[dict] = None


class AnthropicMessagesRequest(BaseModel):
    """Anthropic Messages API shape (reference parity:
    lib/llm/src/http/service/anthropic.rs /v1/messages)."""
    model: str = ""
    max_tokens: Optional[int] = None
    messages: List[ChatMessage] = Field(default_factory=list)
    system: Optional[str] = None
    temperature: Optional[float] = None
    top_p: float = 1.0
    top_k: int = 0
    stream: bool = False
    ignore_eos: bool = False
    seed: int = 0
    user: Optional[str] = None


class ResponsesRequest(BaseModel):
    """OpenAI Responses API shape (reference parity:
    lib/llm/src/http/service/openai.rs:4158 handler_responses)."""
    model: str = ""
    input: Union[str, List[dict]] = ""
    instructions: Optional[str] = None
    max_output_tokens: int = 128
    temperature: Optional[float] = None
    top_p: float = 1.0
    stream: bool = False
    user: Optional[str] = None


def _session_of(req, raw: Request) -> Optional[str]:
    return raw.headers.get("x-session-id") or getattr(req, "user", None)


def _stop_list(req) -> List[str]:
    stop = getattr(req, "stop", None)
    if stop is None:
        return []
    return [stop] if isinstance(stop, str) else list(stop)


def _find_stop(text: str, stops: List[str]) -> int:
    """Earliest stop-string position in text, or -1 (Backend-operator
    stop-condition parity: the reference trims the stop string from the
    returned text)."""
    best = -1
    for st in stops:
        if st:
            i = text.find(st)
            if i >= 0 and (best < 0 or i < best):
                best = i
    return best


def build_app(manager: ModelManager,
              request_template: Optional[dict] = None) -> FastAPI:
    """request_template (reference parity: request_template.rs + the
    frontend --request-template flag): {"model", "temperature",
    "max_completion_tokens"} defaults applied when a request omits the
    field; an explicit value always wins."""
    app = FastAPI(title="dynamo_amd", version="0.1.0")
    app.state.manager = manager
    tmpl = request_template or {}

    def _model_of(req) -> str:
        return req.model or tmpl.get("model", "")

    def _temp_of(req) -> float:
        if getattr(req, "temperature", None) is not None:
            return req.temperature
        return float(tmpl.get("temperature", 0.0))

    def _max_tokens_of(req) -> int:
        if getattr(req, "max_tokens", None) is not None:
            return req.max_tokens
        return int(tmpl.get("max_completion_tokens", 128))

    @app.get("/health")
    async def health():
        return {"status": "ok", "models": list(manager.models)}

    @app.get("/metrics")
    async def metrics():
        from fastapi import Response
        return Response(generate_latest(), media_type=CONTENT_TYPE_LATEST)

    @app.get("/v1/models")
    async def models():
        return {"object": "list", "data": [
            {"id": name, "object": "model", "owned_by": "dynamo_amd",
             "created": int(e.card.get("registered_at", time.time()))
             if isinstance(e.card, dict) else int(time.time())}
            for name, e in manager.models.items()]}

    async def _run(entry, token_ids, req, rid,
                   session_id=None, extra=None) -> AsyncIterator[dict]:
        sampling = {"temperature": _temp_of(req), "top_p": req.top_p,
                    "top_k": getattr(req, "top_k", 0),
                    "seed": getattr(req, "seed", 0),
                    "logprobs": getattr(req, "logprobs", None) or 0}
        # checkpoint generation_config eos ids (possibly several, e.g.
        # llama-3) win over the tokenizer's probed eos token
        eos_ids = (entry.card or {}).get("eos_token_ids")
        if not eos_ids:
            eos = getattr(entry.tokenizer, "eos_id", None)
            eos_ids = [eos] if eos is not None else []
        stop = {"max_tokens": _max_tokens_of(req),
                "ignore_eos": getattr(req, "ignore_eos", False),
                "stop_token_ids": list(eos_ids)}
        from dynamo_amd.router.kv_router import AllWorkersBusy
        pe = getattr(req, "prompt_embeds", None)
        if pe:
            extra = dict(extra or {}, prompt_embeds=pe)
        try:
            async for chunk in manager.generate_tokens(
                    entry, token_ids, sampling, stop, request_id=rid,
                    session_id=session_id, extra=extra):
                yield chunk
        except AllWorkersBusy as e:
            raise HTTPException(503, str(e))

    @app.post("/internal/kv_overlap")
    async def kv_overlap(raw: Request):
        """Prefix-overlap digest for cross-pool (global) routing —
        kv_dc_relay parity-lite: a global router asks each pool how many
        KV blocks of a token prefix it already holds and routes to the
        warmest pool. Returns the best per-worker overlap in blocks."""
        payload = await raw.json()
        token_ids = payload.get("token_ids", [])
        model = payload.get("model", "")
        try:
            entry = manager.get(model)
        except KeyError:
            return {"overlap_blocks": 0, "block_size": 0}
        r = entry.router
        if r is None or not token_ids:
            return {"overlap_blocks": 0, "block_size": 0}
        from dynamo_amd import _core
        hashes = _core.chain_hashes(list(token_ids), r.cfg.block_size,
                                    r.cfg.block_salt)
        matches = r.indexer.find_matches(hashes)
        best = max(matches.values()) if matches else 0
        if r.cfg.host_overlap_weight > 0:
            hm = r.host_indexer.find_matches(hashes)
            if hm:
                best = max(best, int(r.cfg.host_overlap_weight
                                     * max(hm.values())))
        return {"overlap_blocks": best,
                "block_size": r.cfg.block_size,
                "total_blocks": len(hashes)}

    @app.post("/internal/kv_digest")
    async def kv_digest(raw: Request):
        """Cuckoo-filter digest of this pool's cached KV blocks
        (kv_dc_relay parity: lib/llm/src/kv_dc_relay builds cuckoo digests
        pools publish for cross-DC routing). A global router polls this
        periodically and ranks pools per request LOCALLY via max_prefix —
        no per-request overlap RPC."""
        import base64 as _b64
        payload = await raw.json()
        model = payload.get("model", "")
        try:
            entry = manager.get(model)
        except KeyError:
            return {"count": 0}
        r = entry.router
        if r is None:
            return {"count": 0}
        from dynamo_amd import _core
        hashes = r.indexer.all_hashes()
        # host-tier (G2/G3) blocks are onboardable prefix warmth too -
        # include them so cross-pool ranking sees the full cached set
        host_hashes = r.host_indexer.all_hashes()
        cf = _core.CuckooFilter(max(1024, (len(hashes)
                                           + len(host_hashes)) * 2))
        for h in hashes:
            cf.insert(h)
        for h in host_hashes:
            cf.insert(h)
        return {"b64": _b64.b64encode(cf.to_bytes()).decode(),
                "count": cf.count(), "block_size": r.cfg.block_size,
                "salt": r.cfg.block_salt}

    @app.websocket("/v1/realtime")
    async def realtime(ws: WebSocket):
        """Realtime bidirectional session, TEXT modality (the reference's
        realtime path is audio-centric via vllm/omni realtime_handler.py;
        this build has no audio models, so the OpenAI realtime event
        protocol is served for text): session.update ->
        session.updated; conversation.item.create (ack'd) accumulates the
        dialogue; response.create streams response.created ->
        response.output_text.delta* -> response.output_text.done ->
        response.done. Unlike the reference MVP the session is STATEFUL -
        each response re-renders the full conversation (and appends the
        assistant turn for later ones)."""
        await ws.accept()
        import uuid as _uuid
        session = {"id": f"sess_{_uuid.uuid4().hex[:16]}",
                   "model": ws.query_params.get("model", ""),
                   "modalities": ["text"]}
        conversation: List[dict] = []
        await ws.send_json({"type": "session.created", "session": session})
        try:
            while True:
                ev = await ws.receive_json()
                t = ev.get("type")
                if t == "session.update":
                    patch = ev.get("session") or {}
                    session.update({k: v for k, v in patch.items()
                                    if k in ("model", "modalities",
                                             "instructions")})
                    await ws.send_json({"type": "session.updated",
                                        "session": session})
                elif t == "conversation.item.create":
                    item = ev.get("item") or {}
                    text = "".join(
                        c.get("text", "") for c in item.get("content", [])
                        if isinstance(c, dict))
                    conversation.append({"role": item.get("role", "user"),
                                         "content": text})
                    await ws.send_json({"type": "conversation.item.created",
                                        "item": {"id": f"item_{len(conversation)}",
                                                 "role": item.get("role", "user")}})
                elif t == "response.create":
                    try:
                        entry = manager.get(session.get("model") or "")
                    except KeyError:
                        await ws.send_json({"type": "error", "error": {
                            "message": "unknown model"}})
                        continue
                    msgs = list(conversation)
                    if session.get("instructions"):
                        msgs = [{"role": "system",
                                 "content": session["instructions"]}] + msgs
                    prompt = entry.templater.render(msgs)
                    toks = entry.tokenizer.encode(prompt)
                    rid = f"resp_{_uuid.uuid4().hex[:16]}"
                    await ws.send_json({"type": "response.created",
                                        "response": {"id": rid}})
                    opts = ((ev.get("response") or {}).get(
                        "max_output_tokens") or 128)
                    produced: List[int] = []
                    acc = ""
                    async for chunk in manager.generate_tokens(
                            entry, toks, {"temperature": 0.0},
                            {"max_tokens": int(opts)}):
                        prev = len(produced)
                        produced.extend(chunk.get("token_ids", []))
                        d = entry.tokenizer.decode_incremental(produced, prev)
                        if d:
                            acc += d
                            await ws.send_json({
                                "type": "response.output_text.delta",
                                "response_id": rid, "delta": d})
                    conversation.append({"role": "assistant", "content": acc})
                    await ws.send_json({"type": "response.output_text.done",
                                        "response_id": rid, "text": acc})
                    await ws.send_json({"type": "response.done",
                                        "response": {"id": rid,
                                                     "status": "completed"}})
        except Exception:
            return

    @app.get("/config")
    async def config_dump():
        """Reproducibility config dump (reference parity:
        components common/config_dump)."""
        import dataclasses
        return {
            "router": dataclasses.asdict(manager.router_cfg),
            "namespace": manager.namespace,
            "models": {name: {"card": e.card,
                              "workers": [i.instance_id for i in
                                          e.router.client.instances()]
                              if e.router else []}
                       for name, e in manager.models.items()},
        }

    def _entry_or_404(model):
        try:
            return manager.get(model)
        except KeyError as e:
            raise HTTPException(404, str(e))

    # -- OpenAI Responses API (openai.rs:4158 handler_responses) --------
    @app.post("/v1/responses")
    async def responses(req: ResponsesRequest, raw: Request):
        entry = _entry_or_404(_model_of(req))
        REQS.labels(entry.name, "responses").inc()
        if isinstance(req.input, str):
            msgs = [{"role": "user", "content": req.input}]
        else:
            msgs = [{"role": m.get("role", "user"),
                     "content": (m.get("content") if isinstance(
                         m.get("content"), str) else "".join(
                             c.get("text", "") for c in m.get("content", [])
                             if isinstance(c, dict)))}
                    for m in req.input]
        if req.instructions:
            msgs = [{"role": "system", "content": req.instructions}] + msgs
        prompt = entry.templater.render(msgs)
        token_ids = entry.tokenizer.encode(prompt)
        rid = f"resp_{uuid.uuid4().hex[:24]}"
        t0 = time.time()
        shim = CompletionRequest(model=req.model, max_tokens=req.max_output_tokens,
                                 temperature=req.temperature, top_p=req.top_p)

        if req.stream:
            async def sse():
                produced: List[int] = []
                first = True
                try:
                    yield ("event: response.created\ndata: " + json.dumps(
                        {"type": "response.created",
                         "response": {"id": rid, "object": "response",
                                      "status": "in_progress",
                                      "model": entry.name}}) + "\n\n")
                    async for chunk in _run(entry, token_ids, shim, rid,
                                            _session_of(req, raw)):
                        if await raw.is_disconnected():
                            break
                        if first:
                            TTFT.labels(entry.name).observe(time.time() - t0)
                            first = False
                        prev = len(produced)
                        produced.extend(chunk.get("token_ids", []))
                        delta = entry.tokenizer.decode_incremental(produced,
                                                                   prev)
                        if delta:
                            yield ("event: response.output_text.delta\n"
                                   "data: " + json.dumps(
                                       {"type": "response.output_text.delta",
                                        "delta": delta}) + "\n\n")
                    text = entry.tokenizer.decode(produced)
                    yield ("event: response.completed\ndata: " + json.dumps(
                        {"type": "response.completed",
                         "response": _response_body(rid, entry.name, text,
                                                    len(token_ids),
                                                    len(produced))}) + "\n\n")
                finally:
                    LATENCY.labels(entry.name).observe(time.time() - t0)
            return StreamingResponse(sse(), media_type="text/event-stream")

        produced: List[int] = []
        async for chunk in _run(entry, token_ids, shim, rid,
                                _session_of(req, raw)):
            produced.extend(chunk.get("token_ids", []))
        LATENCY.labels(entry.name).observe(time.time() - t0)
        return _response_body(rid, entry.name, entry.tokenizer.decode(produced),
                              len(token_ids), len(produced))

    def _response_body(rid, model, text, in_toks, out_toks):
        return {
            "id": rid, "object": "response", "status": "completed",
            "created_at": int(time.time()), "model": model,
            "output": [{"type": "message", "id": f"msg_{rid[5:]}",
                        "role": "assistant", "status": "completed",
                        "content": [{"type": "output_text", "text": text,
                                     "annotations": []}]}],
            "output_text": text,
            "usage": {"input_tokens": in_toks, "output_tokens": out_toks,
                      "total_tokens": in_toks + out_toks},
        }

    # -- Files + Batches (openai.rs:3984-3987 batch routes) --------------
    files: dict = {}     # file_id -> {"bytes": b..., "filename": ...}
    batches: dict = {}   # batch_id -> status dict

    @app.post("/v1/files")
    async def create_file(raw: Request):
        # multipart when python-multipart is installed; otherwise accept
        # the raw body as the file content (purpose/filename via query)
        filename = raw.query_params.get("filename", "upload.jsonl")
        purpose = raw.query_params.get("purpose", "batch")
        data = None
        ctype = raw.headers.get("content-type", "")
        if ctype.startswith("multipart/"):
            try:
                form = await raw.form()
                up = form.get("file")
                if up is not None:
                    data = await up.read()
                    filename = up.filename or filename
                    purpose = form.get("purpose", purpose)
            except Exception:
                pass
        if data is None:
            data = await raw.body()
        if not data:
            raise HTTPException(400, "empty file upload")
        fid = f"file-{uuid.uuid4().hex[:24]}"
        files[fid] = {"bytes": data, "filename": filename,
                      "purpose": purpose}
        return {"id": fid, "object": "file", "bytes": len(data),
                "filename": filename, "purpose": purpose,
                "created_at": int(time.time())}

    @app.get("/v1/files/{fid}/content")
    async def file_content(fid: str):
        if fid not in files:
            raise HTTPException(404, f"file {fid} not found")
        from fastapi import Response
        return Response(files[fid]["bytes"],
                        media_type="application/octet-stream")

    async def _run_batch(bid: str):
        b = batches[bid]
        b["status"] = "in_progress"
        out_lines = []
        nerr = 0
        for line in files[b["input_file_id"]]["bytes"].splitlines():
            if not line.strip():
                continue
            item = None
            try:
                item = json.loads(line)
                body = item.get("body", {})
                entry = manager.get(body.get("model", ""))
                if b["endpoint"] == "/v1/chat/completions":
                    prompt = entry.templater.render(body.get("messages", []))
                else:
                    prompt = body.get("prompt", "")
                toks = entry.tokenizer.encode(prompt)
                produced = []
                async for ch in manager.generate_tokens(
                        entry, toks,
                        {"temperature": body.get("temperature", 0.0)},
                        {"max_tokens": body.get("max_tokens", 128)}):
                    produced.extend(ch.get("token_ids", []))
                text = entry.tokenizer.decode(produced)
                out_lines.append(json.dumps({
                    "id": f"batch_req_{uuid.uuid4().hex[:16]}",
                    "custom_id": item.get("custom_id"),
                    "response": {"status_code": 200, "body": {
                        "choices": [{"index": 0, "message": {
                            "role": "assistant", "content": text},
                            "finish_reason": "stop"}]}},
                    "error": None}))
                b["request_counts"]["completed"] += 1
            except Exception as e:
                nerr += 1
                b["request_counts"]["failed"] += 1
                out_lines.append(json.dumps({
                    "custom_id": item.get("custom_id") if item else None,
                    "response": None,
                    "error": {"message": str(e)}}))
        ofid = f"file-{uuid.uuid4().hex[:24]}"
        files[ofid] = {"bytes": "\n".join(out_lines).encode(),
                       "filename": f"{bid}_output.jsonl", "purpose": "batch_output"}
        b["output_file_id"] = ofid
        b["status"] = "completed" if nerr == 0 else "completed"
        b["completed_at"] = int(time.time())

    @app.post("/v1/batches")
    async def create_batch(raw: Request):
        body = await raw.json()
        fid = body.get("input_file_id")
        if fid not in files:
            raise HTTPException(404, f"input file {fid} not found")
        bid = f"batch_{uuid.uuid4().hex[:24]}"
        batches[bid] = {
            "id": bid, "object": "batch", "status": "validating",
            "endpoint": body.get("endpoint", "/v1/chat/completions"),
            "input_file_id": fid, "output_file_id": None,
            "created_at": int(time.time()), "completed_at": None,
            "request_counts": {"total": sum(
                1 for ln in files[fid]["bytes"].splitlines() if ln.strip()),
                "completed": 0, "failed": 0},
        }
        asyncio.get_running_loop().create_task(_run_batch(bid))
        return batches[bid]

    @app.get("/v1/batches/{bid}")
    async def get_batch(bid: str):
        if bid not in batches:
            raise HTTPException(404, f"batch {bid} not found")
        return batches[bid]

    @app.post("/v1/completions")
    async def completions(req: CompletionRequest, raw: Request):
        entry = _entry_or_404(_model_of(req))
        REQS.labels(entry.name, "completions").inc()
        if isinstance(req.prompt, list):
            token_ids = list(req.prompt)
        elif len(req.prompt) > 4096:
            # long prompts: tokenize off the event loop (reference parity:
            # runtime/src/compute rayon pool for CPU-bound tokenization)
            token_ids = await asyncio.to_thread(entry.tokenizer.encode,
                                                req.prompt)
        else:
            token_ids = entry.tokenizer.encode(req.prompt)
        rid = f"cmpl-{uuid.uuid4().hex[:24]}"
        t0 = time.time()

        if req.stream:
            stops = _stop_list(req)

            async def sse():
                produced: List[int] = []
                acc = ""     # decoded text so far (stable incremental path)
                sent = 0     # chars of acc already emitted
                # a stop may span chunk boundaries: withhold its max length
                # minus one trailing chars until they are cleared
                hold = max((len(s) for s in stops), default=1) - 1
                first = True

                def event(text, finish):
                    data = {"id": rid, "object": "text_completion",
                            "model": entry.name, "choices": [{
                                "index": 0, "text": text,
                                "finish_reason": finish}]}
                    return f"data: {json.dumps(data)}\n\n"

                try:
                    finish = None
                    last_t = None
                    async for chunk in _run(entry, token_ids, req, rid, _session_of(req, raw)):
                        if await raw.is_disconnected():
                            break
                        now = time.time()
                        if first:
                            TTFT.labels(entry.name).observe(now - t0)
                            first = False
                        elif last_t is not None:
                            ITL.labels(entry.name).observe(now - last_t)
                        last_t = now
                        prev = len(produced)
                        produced.extend(chunk.get("token_ids", []))
                        OUT_TOKENS.labels(entry.name).inc(
                            len(produced) - prev)
                        acc += entry.tokenizer.decode_incremental(produced,
                                                                  prev)
                        finish = chunk.get("finish_reason")
                        if stops:
                            cut = _find_stop(acc, stops)
                            if cut >= 0:
                                yield event(acc[sent:cut], "stop")
                                sent = cut
                                finish = "stop"
                                break
                            safe = max(sent, len(acc) - hold)
                            if safe > sent or finish:
                                tail = len(acc) if finish else safe
                                yield event(acc[sent:tail], finish)
                                sent = tail
                        else:
                            yield event(acc[sent:], finish)
                            sent = len(acc)
                    if stops and finish is None and sent < len(acc):
                        yield event(acc[sent:], None)   # disconnect flush
                    yield "data: [DONE]\n\n"
                finally:
                    LATENCY.labels(entry.name).observe(time.time() - t0)
            return StreamingResponse(sse(), media_type="text/event-stream")

        produced: List[int] = []
        lps: List[dict] = []
        finish = None
        stops = _stop_list(req)
        text_cut = None
        acc = ""
        async for chunk in _run(entry, token_ids, req, rid, _session_of(req, raw)):
            prev = len(produced)
            produced.extend(chunk.get("token_ids", []))
            lps.extend(chunk.get("logprobs", []))
            finish = chunk.get("finish_reason") or finish
            if stops:
                acc += entry.tokenizer.decode_incremental(produced, prev)
                cut = _find_stop(acc, stops)
                if cut >= 0:
                    text_cut = acc[:cut]
                    finish = "stop"
                    break
        LATENCY.labels(entry.name).observe(time.time() - t0)
        logprobs_out = None
        if req.logprobs and lps:
            logprobs_out = {
                "tokens": [entry.tokenizer.decode([t]) for t in produced],
                "token_logprobs": [d["token_logprob"] for d in lps],
                "top_logprobs": [
                    {entry.tokenizer.decode([t]): v for t, v in d["top"]}
                    for d in lps],
            }
        return {
            "id": rid, "object": "text_completion", "created": int(t0),
            "model": entry.name,
            "choices": [{"index": 0,
                         "text": (text_cut if text_cut is not None
                                  else entry.tokenizer.decode(produced)),
                         "finish_reason": finish or "stop",
                         "logprobs": logprobs_out,
                         "token_ids": produced}],
            "usage": {"prompt_tokens": len(token_ids),
                      "completion_tokens": len(produced),
                      "total_tokens": len(token_ids) + len(produced)},
        }

    @app.post("/v1/messages")
    async def anthropic_messages(req: AnthropicMessagesRequest, raw: Request):
        """Anthropic Messages API over the same engine pipeline."""
        entry = _entry_or_404(_model_of(req))
        REQS.labels(entry.name, "messages").inc()
        msgs = [m.model_dump() for m in req.messages]
        if req.system:
            msgs = [{"role": "system", "content": req.system}] + msgs
        prompt = entry.templater.render(msgs)
        token_ids = (await asyncio.to_thread(entry.tokenizer.encode, prompt)
                     if len(prompt) > 4096 else
                     entry.tokenizer.encode(prompt))
        rid = f"msg_{uuid.uuid4().hex[:24]}"
        t0 = time.time()

        if req.stream:
            async def sse():
                yield ("event: message_start\ndata: " + json.dumps(
                    {"type": "message_start", "message": {
                        "id": rid, "type": "message", "role": "assistant",
                        "model": entry.name, "content": [],
                        "usage": {"input_tokens": len(token_ids)}}}) + "\n\n")
                yield ("event: content_block_start\ndata: " + json.dumps(
                    {"type": "content_block_start", "index": 0,
                     "content_block": {"type": "text", "text": ""}}) + "\n\n")
                produced: List[int] = []
                finish = None
                try:
                    async for chunk in _run(entry, token_ids, req, rid,
                                            _session_of(req, raw)):
                        if await raw.is_disconnected():
                            break
                        prev = len(produced)
                        produced.extend(chunk.get("token_ids", []))
                        finish = chunk.get("finish_reason") or finish
                        text = entry.tokenizer.decode_incremental(produced,
                                                                  prev)
                        if text:
                            yield ("event: content_block_delta\ndata: "
                                   + json.dumps(
                                       {"type": "content_block_delta",
                                        "index": 0, "delta": {
                                            "type": "text_delta",
                                            "text": text}}) + "\n\n")
                    yield ("event: content_block_stop\ndata: " + json.dumps(
                        {"type": "content_block_stop", "index": 0}) + "\n\n")
                    stop = ("max_tokens" if finish == "length"
                            else "end_turn")
                    yield ("event: message_delta\ndata: " + json.dumps(
                        {"type": "message_delta",
                         "delta": {"stop_reason": stop},
                         "usage": {"output_tokens": len(produced)}}) + "\n\n")
                    yield ("event: message_stop\ndata: " + json.dumps(
                        {"type": "message_stop"}) + "\n\n")
                finally:
                    LATENCY.labels(entry.name).observe(time.time() - t0)
            return StreamingResponse(sse(), media_type="text/event-stream")

        produced: List[int] = []
        finish = None
        async for chunk in _run(entry, token_ids, req, rid,
                                _session_of(req, raw)):
            produced.extend(chunk.get("token_ids", []))
            finish = chunk.get("finish_reason") or finish
        LATENCY.labels(entry.name).observe(time.time() - t0)
        return {
            "id": rid, "type": "message", "role": "assistant",
            "model": entry.name,
            "content": [{"type": "text",
                         "text": entry.tokenizer.decode(produced)}],
            "stop_reason": ("max_tokens" if finish == "length"
                            else "end_turn"),
            "usage": {"input_tokens": len(token_ids),
                      "output_tokens": len(produced)},
        }

    @app.post("/v1/embeddings")
    async def embeddings(req: EmbeddingRequest):
        """Mean-pooled last-hidden-state embeddings (reference parity:
        lib/llm/src/http embeddings route)."""
        entry = _entry_or_404(_model_of(req))
        REQS.labels(entry.name, "embeddings").inc()
        raw_inputs = req.input
        if isinstance(raw_inputs, str):
            raw_inputs = [raw_inputs]
        elif raw_inputs and isinstance(raw_inputs[0], int):
            raw_inputs = [raw_inputs]      # single pre-tokenized prompt
        data, total_tokens = [], 0
        for idx, item in enumerate(raw_inputs):
            token_ids = (list(item) if isinstance(item, list)
                         else entry.tokenizer.encode(item))
            total_tokens += len(token_ids)
            rid = f"embd-{uuid.uuid4().hex[:24]}"
            sampling = {"embed": True}
            stop = {"max_tokens": 1, "ignore_eos": True}
            vec = None
            async for chunk in manager.generate_tokens(entry, token_ids,
                                                       sampling, stop,
                                                       request_id=rid):
                if chunk.get("embedding") is not None:
                    vec = chunk["embedding"]
            if vec is None:
                raise HTTPException(500, "worker returned no embedding")
            data.append({"object": "embedding", "index": idx,
                         "embedding": vec})
        return {"object": "list", "model": entry.name, "data": data,
                "usage": {"prompt_tokens": total_tokens,
                          "total_tokens": total_tokens}}

    _IMG_SENT = "\x00img\x00"

    def _image_payload(part) -> dict:
        """Media fetch/decode (reference preprocessor.rs:2248): data: URIs
        only — there is no egress for http(s) media in this environment."""
        url = part.get("image_url")
        if isinstance(url, dict):
            url = url.get("url", "")
        if not isinstance(url, str) or not url.startswith("data:"):
            raise HTTPException(
                400, "only data: image URLs are supported (no egress)")
        head, _, b64 = url.partition(",")
        if "tensor" in head:   # data:application/x-tensor;shape=3x32x32;base64,
            shape = [int(x) for x in
                     head.split("shape=")[1].split(";")[0].split("x")]
            return {"b64_pixels": b64, "shape": shape, "dtype": "float32"}
        return {"b64_image": b64}

    async def _mm_prepare(entry, req):
        """Multimodal chat: flatten content parts, route images to an
        encode worker, splice embeddings over placeholder tokens.
        Returns (token_ids, extra) or None for text-only requests."""
        if not any(isinstance(m.content, list) for m in req.messages):
            return None
        msgs, images = [], []
        for m in req.messages:
            if isinstance(m.content, str):
                msgs.append({"role": m.role, "content": m.content})
                continue
            text = []
            for p in m.content:
                if p.get("type") == "image_url":
                    images.append(_image_payload(p))
                    text.append(_IMG_SENT)
                else:
                    text.append(p.get("text", ""))
            msgs.append({"role": m.role, "content": "".join(text)})
        prompt = entry.templater.render(msgs)
        if not images:
            return entry.tokenizer.encode(prompt), None
        embs = await manager.encode_images(entry.name, images)
        segs = prompt.split(_IMG_SENT)
        token_ids: List[int] = []
        mm = []
        for i, seg in enumerate(segs):
            token_ids.extend(entry.tokenizer.encode(seg))
            if i < len(segs) - 1:
                emb = embs[i]
                mm.append({"offset": len(token_ids), **emb})
                token_ids.extend([0] * emb["shape"][0])
        return token_ids, {"mm_embeds": mm}

    def _spec_prefill_enabled(req) -> bool:
        v = getattr(req, "speculative_prefill", None)
        if v is not None:
            return bool(v)
        hints = (getattr(req, "nvext", None) or {}).get("agent_hints") or {}
        return bool(hints.get("speculative_prefill"))

    async def _spec_prefill(entry, messages: List[dict], reply_text: str):
        """Warm the next turn's prefix: render the conversation WITH the
        completed assistant message and no generation prompt (the exact
        prefix the next user turn extends), then run a 1-token request
        through the normal routed path - the KV router lands it on the
        warmest worker and the prefix cache keeps the blocks."""
        try:
            convo = messages + [{"role": "assistant", "content": reply_text}]
            prefix = entry.templater.render(convo,
                                            add_generation_prompt=False)
            toks = await asyncio.to_thread(entry.tokenizer.encode, prefix)
            sp = {"temperature": 0.0}
            opts = {"max_tokens": 1, "ignore_eos": True}
            async for _ in manager.generate_tokens(entry, toks, sp, opts):
                pass
        except Exception:
            log.debug("speculative next-turn prefill failed", exc_info=True)

    @app.post("/v1/chat/completions")
    async def chat(req: ChatRequest, raw: Request):
        entry = _entry_or_404(_model_of(req))
        REQS.labels(entry.name, "chat").inc()
        mm_extra = None
        mm = await _mm_prepare(entry, req)
        if mm is not None:
            token_ids, mm_extra = mm
        else:
            prompt = entry.templater.render(
                [m.model_dump() for m in req.messages])
            token_ids = (await asyncio.to_thread(entry.tokenizer.encode,
                                                 prompt)
                         if len(prompt) > 4096 else
                         entry.tokenizer.encode(prompt))
        rid = f"chatcmpl-{uuid.uuid4().hex[:24]}"
        t0 = time.time()

        if req.stream:
            stops = _stop_list(req)

            async def sse():
                produced: List[int] = []
                acc = ""     # decoded text so far
                sent = 0     # chars of acc already emitted
                # stop-string enforcement (same discipline as streaming
                # /v1/completions): a stop may span chunk boundaries, so
                # withhold its max length minus one trailing chars
                hold = max((len(s) for s in stops), default=1) - 1
                first = True

                def event(text, finish, role=False):
                    delta = {}
                    if role:
                        delta["role"] = "assistant"
                    delta["content"] = text
                    data = {"id": rid, "object": "chat.completion.chunk",
                            "model": entry.name, "choices": [{
                                "index": 0, "delta": delta,
                                "finish_reason": finish}]}
                    return f"data: {json.dumps(data)}\n\n"

                try:
                    finish = None
                    async for chunk in _run(entry, token_ids, req, rid, _session_of(req, raw), extra=mm_extra):
                        if await raw.is_disconnected():
                            break
                        role = first
                        if first:
                            TTFT.labels(entry.name).observe(time.time() - t0)
                            first = False
                        prev = len(produced)
                        produced.extend(chunk.get("token_ids", []))
                        acc += entry.tokenizer.decode_incremental(produced,
                                                                  prev)
                        finish = chunk.get("finish_reason")
                        if stops:
                            cut = _find_stop(acc, stops)
                            if cut >= 0:
                                yield event(acc[sent:cut], "stop", role)
                                sent = cut
                                finish = "stop"
                                break
                            safe = max(sent, len(acc) - hold)
                            if safe > sent or finish or role:
                                tail = len(acc) if finish else safe
                                yield event(acc[sent:tail], finish, role)
                                sent = tail
                        else:
                            yield event(acc[sent:], finish, role)
                            sent = len(acc)
                    yield "data: [DONE]\n\n"
                    if _spec_prefill_enabled(req):
                        asyncio.get_running_loop().create_task(_spec_prefill(
                            entry, [m.model_dump() for m in req.messages],
                            acc))
                finally:
                    LATENCY.labels(entry.name).observe(time.time() - t0)
            return StreamingResponse(sse(), media_type="text/event-stream")

        produced: List[int] = []
        finish = None
        stops = _stop_list(req)
        text_cut = None
        acc = ""
        async for chunk in _run(entry, token_ids, req, rid, _session_of(req, raw), extra=mm_extra):
            prev = len(produced)
            produced.extend(chunk.get("token_ids", []))
            finish = chunk.get("finish_reason") or finish
            if stops:
                acc += entry.tokenizer.decode_incremental(produced, prev)
                cut = _find_stop(acc, stops)
                if cut >= 0:
                    text_cut = acc[:cut]
                    finish = "stop"
                    break
        LATENCY.labels(entry.name).observe(time.time() - t0)
        text_out = (text_cut if text_cut is not None
                    else entry.tokenizer.decode(produced))
        # postprocessing parity (preprocessor.rs:3953/:4577): reasoning
        # split always; tool-call extraction only when tools were offered
        from .parsers import parse_reasoning, parse_tool_calls
        text_out, reasoning = parse_reasoning(text_out)
        tool_calls = []
        if req.tools:
            text_out, tool_calls = parse_tool_calls(text_out)
        message = {"role": "assistant", "content": text_out or None}
        if reasoning:
            message["reasoning_content"] = reasoning
        if tool_calls:
            message["tool_calls"] = tool_calls
        if _spec_prefill_enabled(req):
            asyncio.get_running_loop().create_task(_spec_prefill(
                entry, [m.model_dump() for m in req.messages],
                text_out or ""))
        return {
            "id": rid, "object": "chat.completion", "created": int(t0),
            "model": entry.name,
            "choices": [{"index": 0, "message": message,
                "finish_reason": ("tool_calls" if tool_calls
                                  else finish or "stop")}],
            "usage": {"prompt_tokens": len(token_ids),
                      "completion_tokens": len(produced),
                      "total_tokens": len(token_ids) + len(produced)},
        }

    return app
