"""Global (cross-pool) router: one HTTP endpoint over multiple frontends.

Reference parity: components/src/dynamo/global_router — hierarchical
routing across pools/deployments. Each downstream pool is a full
dynamo_amd frontend (with its own KV-aware router); this layer picks a
pool per request by model availability + in-flight load, proxies the
request (streaming passthrough), and fails over to another pool when one
dies mid-connect.

  python -m dynamo_amd.frontend.global_router \
      --pools http://host1:8000,http://host2:8000 --port 9000
"""
from __future__ import annotations

import argparse
import asyncio
import json
import logging
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set

import httpx
import uvicorn
from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import Response, StreamingResponse

log = logging.getLogger("dynamo_amd.global_router")


@dataclass
class PoolState:
    url: str
    models: Set[str] = field(default_factory=set)
    healthy: bool = False
    inflight: int = 0
    last_check: float = 0.0
    # cuckoo digests of the pool's cached KV blocks, per model
    # (kv_dc_relay parity: digest stream instead of per-request RPC)
    digests: Dict[str, object] = field(default_factory=dict)
    digest_cfg: Dict[str, tuple] = field(default_factory=dict)  # (bs, salt)


class GlobalRouter:
    def __init__(self, pool_urls: List[str], check_interval: float = 2.0):
        self.pools: Dict[str, PoolState] = {u: PoolState(u)
                                            for u in pool_urls}
        self.check_interval = check_interval
        self.client = httpx.AsyncClient(timeout=None)
        self._task: Optional[asyncio.Task] = None

    async def start(self):
        await self._refresh()
        self._task = asyncio.create_task(self._watch())
        return self

    async def stop(self):
        if self._task:
            self._task.cancel()
        await self.client.aclose()

    async def _refresh(self):
        import base64 as _b64
        for p in self.pools.values():
            try:
                r = await self.client.get(p.url + "/health", timeout=3.0)
                body = r.json()
                p.healthy = r.status_code == 200
                p.models = set(body.get("models", []))
            except Exception:
                p.healthy = False
            p.last_check = time.time()
            if not p.healthy:
                continue
            # pull each model's KV digest (best-effort; stale-tolerant)
            for model in p.models:
                try:
                    r = await self.client.post(
                        p.url + "/internal/kv_digest",
                        json={"model": model}, timeout=1.0)
                    d = r.json()
                    if d.get("count", 0) and d.get("b64"):
                        from dynamo_amd import _core
                        p.digests[model] = _core.CuckooFilter.from_bytes(
                            _b64.b64decode(d["b64"]), d["count"])
                        p.digest_cfg[model] = (d["block_size"],
                                               d.get("salt", 0))
                    else:
                        p.digests.pop(model, None)
                except Exception:
                    pass

    async def _watch(self):
        while True:
            await asyncio.sleep(self.check_interval)
            try:
                await self._refresh()
            except Exception:
                log.exception("pool refresh failed")

    def candidates(self, model: str) -> List[PoolState]:
        pools = [p for p in self.pools.values() if p.healthy
                 and (not model or model in p.models or not p.models)]
        return sorted(pools, key=lambda p: p.inflight)

    # -- cross-pool KV awareness (kv_dc_relay parity-lite) ---------------
    async def _kv_rank(self, pools: List[PoolState], model: str,
                       token_ids) -> List[PoolState]:
        """Re-rank candidate pools by prefix overlap: ask each pool's
        /internal/kv_overlap digest (short timeout, best-effort) and sort
        by (-overlap_blocks, inflight)."""
        if not token_ids or len(pools) < 2:
            return pools

        # fast path: rank from the polled cuckoo digests (no RPC) when
        # every candidate pool has one
        if all(model in p.digests for p in pools):
            from dynamo_amd import _core
            scores = []
            for p in pools:
                bs, salt = p.digest_cfg[model]
                chain = _core.chain_hashes(list(token_ids), bs, salt)
                scores.append(int(p.digests[model].max_prefix(chain)))
            ranked = sorted(zip(pools, scores),
                            key=lambda po: (-po[1], po[0].inflight))
            return [p for p, _ in ranked]

        async def ask(p: PoolState) -> int:
            try:
                r = await self.client.post(
                    p.url + "/internal/kv_overlap",
                    json={"model": model, "token_ids": list(token_ids)},
                    timeout=0.25)
                return int(r.json().get("overlap_blocks", 0))
            except Exception:
                return 0

        overlaps = await asyncio.gather(*[ask(p) for p in pools])
        ranked = sorted(zip(pools, overlaps),
                        key=lambda po: (-po[1], po[0].inflight))
        return [p for p, _ in ranked]

    # -- proxying --------------------------------------------------------
    async def proxy(self, path: str, payload: dict, stream: bool):
        model = payload.get("model", "")
        cands = self.candidates(model)
        if not cands:
            raise HTTPException(503, f"no healthy pool serves {model!r}")
        prompt = payload.get("prompt")
        if isinstance(prompt, list) and prompt and isinstance(prompt[0], int):
            cands = await self._kv_rank(cands, model, prompt)
        last_err: Optional[Exception] = None
        for pool in cands:                       # failover across pools
            pool.inflight += 1
            try:
                if stream:
                    return await self._proxy_stream(pool, path, payload)
                r = await self.client.post(pool.url + path, json=payload)
                pool.inflight -= 1
                return Response(content=r.content, status_code=r.status_code,
                                media_type=r.headers.get("content-type"))
            except (httpx.ConnectError, httpx.ReadError,
                    httpx.RemoteProtocolError) as e:
                pool.inflight -= 1
                pool.healthy = False
                last_err = e
        raise HTTPException(502, f"all pools failed: {last_err}")

    async def _proxy_stream(self, pool: PoolState, path: str, payload: dict):
        req = self.client.build_request("POST", pool.url + path,
                                        json=payload)
        resp = await self.client.send(req, stream=True)

        async def body():
            try:
                async for chunk in resp.aiter_bytes():
                    yield chunk
            finally:
                await resp.aclose()
                pool.inflight -= 1
        return StreamingResponse(body(), status_code=resp.status_code,
                                 media_type=resp.headers.get("content-type"))


def build_global_app(router: GlobalRouter) -> FastAPI:
    app = FastAPI(title="dynamo_amd global router")

    @app.get("/health")
    async def health():
        return {"status": "ok", "pools": {
            p.url: {"healthy": p.healthy, "models": sorted(p.models),
                    "inflight": p.inflight}
            for p in router.pools.values()}}

    @app.get("/v1/models")
    async def models():
        seen = {}
        for p in router.pools.values():
            if p.healthy:
                for m in p.models:
                    seen.setdefault(m, p.url)
        return {"object": "list", "data": [
            {"id": m, "object": "model", "owned_by": "dynamo_amd"}
            for m in sorted(seen)]}

    for route in ("/v1/completions", "/v1/chat/completions",
                  "/v1/embeddings", "/v1/messages"):
        def make(path):
            async def handler(raw: Request):
                payload = await raw.json()
                return await router.proxy(path, payload,
                                          bool(payload.get("stream")))
            return handler
        app.add_api_route(route, make(route), methods=["POST"])

    return app


async def async_main(args):
    logging.basicConfig(level=logging.INFO)
    router = GlobalRouter(args.pools.split(","))
    await router.start()
    app = build_global_app(router)
    config = uvicorn.Config(app, host=args.host, port=args.port,
                            log_level="warning")
    server = uvicorn.Server(config)
    print(f"GLOBAL_ROUTER_READY http://{args.host}:{args.port}", flush=True)
    await server.serve()
    await router.stop()


def main():
    ap = argparse.ArgumentParser("dynamo_amd.frontend.global_router")
    ap.add_argument("--pools", required=True,
                    help="comma-separated downstream frontend URLs")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=9000)
    asyncio.run(async_main(ap.parse_args()))


if __name__ == "__main__":
    main()
