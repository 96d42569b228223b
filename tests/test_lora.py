"""LoRA tests (CPU): load/activate/deactivate semantics, output effects,
worker endpoints, save/load roundtrip."""
import asyncio

import pytest
import torch

from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
from dynamo_amd.engine.config import PRESETS


def make_engine():
    cfg = EngineConfig(model=PRESETS["tiny-llama"], device="cpu",
                       max_num_seqs=4, max_batched_tokens=256,
                       max_model_len=256, kv_pool_pages=64, page_size=16,
                       enable_prefix_caching=False)
    return LLMEngine(cfg, seed=7)


def gen(engine, rid, prompt, n=5):
    engine.add_request(rid, prompt, SamplingParams(max_tokens=n))
    out = []
    while engine.has_work():
        for so in engine.step():
            out.append(so.new_token)
    return out


def test_lora_changes_and_restores_outputs():
    e = make_engine()
    p = list(range(40))
    base = gen(e, "a", p)
    e.load_lora("ad1", rank=4, seed=11)
    with_lora = gen(e, "b", p)
    assert with_lora != base, "adapter had no effect"
    # same adapter again -> deterministic
    assert gen(e, "c", p) == with_lora
    e.unload_lora("ad1")
    assert gen(e, "d", p) == base, "unload did not restore base weights"


def test_lora_list_and_multiple():
    e = make_engine()
    e.load_lora("x", rank=4, seed=1, activate=False)
    e.load_lora("y", rank=8, seed=2, activate=False)
    assert e.list_loras() == ["x", "y"]
    p = list(range(30))
    e.lora.activate("x")
    e._invalidate_graphs()
    ox = gen(e, "a", p)
    e.lora.activate("y")
    e._invalidate_graphs()
    oy = gen(e, "b", p)
    assert ox != oy


def test_lora_save_load_roundtrip(tmp_path):
    e = make_engine()
    e.load_lora("r", rank=4, seed=3)
    ad = e.lora.adapters["r"]
    sd = {}
    for k, (A, B) in ad.weights.items():
        sd[f"{k}.A"] = A
        sd[f"{k}.B"] = B
    path = str(tmp_path / "ad.pt")
    torch.save(sd, path)
    e2 = make_engine()
    e2.load_lora("r2", path=path, rank=4)
    p = list(range(40))
    assert gen(e, "a", p) == gen(e2, "b", p)


def test_lora_worker_endpoints():
    async def main():
        from dynamo_amd.runtime import DistributedRuntime, MemoryDiscovery
        from dynamo_amd.workers import WorkerService
        shared = MemoryDiscovery()
        rt = DistributedRuntime(shared)
        ws = WorkerService(make_engine(), rt)
        await ws.start()
        addr = rt.server.address
        r = await rt.client.call(addr, "backend.load_lora",
                                 {"name": "ep1", "rank": 4, "seed": 9})
        assert r["loras"] == ["ep1"]
        r = await rt.client.call(addr, "backend.list_loras", {})
        assert r["loras"] == ["ep1"]
        # metadata updated for lora-aware routing
        inst = shared.list("dynamo", "backend")[0]
        assert inst.metadata.get("loras") == ["ep1"]
        r = await rt.client.call(addr, "backend.unload_lora", {"name": "ep1"})
        assert r["loras"] == []
        await ws.stop()
        await rt.shutdown(drain=False)
    asyncio.new_event_loop().run_until_complete(main())
