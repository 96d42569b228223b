"""GPU host-tier test: pinned-host offload/onboard with the page-copy
kernels and stream fencing, output-exact across the demote/promote cycle."""
import pytest

from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
from dynamo_amd.engine.config import PRESETS

pytestmark = pytest.mark.gpu


def generate(engine, rid, prompt, max_tokens=4):
    engine.add_request(rid, prompt, SamplingParams(max_tokens=max_tokens))
    out = []
    while engine.has_work():
        for so in engine.step():
            if so.req_id == rid:
                out.append(so.new_token)
    return out


def test_gpu_offload_onboard_exact():
    cfg = EngineConfig(model=PRESETS["tiny-llama-gpu"], device="cuda:0",
                       max_num_seqs=4, max_batched_tokens=1024,
                       max_model_len=2048, kv_pool_pages=12, page_size=64,
                       host_cache_pages=32)
    eng = LLMEngine(cfg, seed=7)
    p1 = list(range(256))  # 4 full pages of 64
    o1 = generate(eng, "a", p1)
    for i in range(3):
        generate(eng, f"c{i}", [(300 + i * 320 + j) % 1000 for j in range(320)])
    assert eng.host_tier.stats["offloaded"] > 0
    before = eng.host_tier.stats["onboarded"]
    o2 = generate(eng, "a2", p1)
    assert eng.host_tier.stats["onboarded"] > before
    assert o2 == o1


def test_gpu_disk_tier_exact(tmp_path):
    """G3 on GPU: device -> pinned host -> disk and all the way back,
    output-exact (exercises the stream-sync before a host page is spilled)."""
    cfg = EngineConfig(model=PRESETS["tiny-llama-gpu"], device="cuda:0",
                       max_num_seqs=4, max_batched_tokens=1024,
                       max_model_len=2048, kv_pool_pages=12, page_size=64,
                       host_cache_pages=4, disk_cache_pages=64,
                       disk_cache_path=str(tmp_path / "g3.bin"))
    eng = LLMEngine(cfg, seed=7)
    p1 = list(range(256))
    o1 = generate(eng, "a", p1)
    for i in range(6):
        generate(eng, f"c{i}", [(300 + i * 320 + j) % 1000 for j in range(320)])
    assert eng.host_tier.stats["spilled_disk"] > 0
    o2 = generate(eng, "a2", p1)
    assert eng.host_tier.stats["onboarded_disk"] > 0
    assert o2 == o1


def test_gpu_embedding_request():
    """Embed request on the native GPU path returns a pooled hidden state
    consistent between chunked and unchunked prefill."""
    import torch
    from dynamo_amd.engine.scheduler import SamplingParams as SP

    def embed_with(batched):
        cfg = EngineConfig(model=PRESETS["tiny-llama-gpu"], device="cuda:0",
                           max_num_seqs=4, max_batched_tokens=batched,
                           max_model_len=2048, kv_pool_pages=32, page_size=64,
                           enable_prefix_caching=False)
        eng = LLMEngine(cfg, seed=7)
        eng.add_request("e", list(range(200)), SP(embed=True))
        vec = None
        while eng.has_work():
            for so in eng.step():
                if so.finish_reason == "embed":
                    vec = so.embedding
        assert vec is not None and len(vec) == 512
        return torch.tensor(vec)

    v1 = embed_with(1024)
    v2 = embed_with(64)
    assert torch.allclose(v1, v2, rtol=3e-2, atol=3e-3), \
        (v1 - v2).abs().max().item()


def test_gpu_object_tier_cross_engine(tmp_path):
    """G4 on GPU: pages published by one engine onboard into a second
    engine via the shared object store, output-exact."""
    import time
    store = str(tmp_path / "store")

    def make(seed=7):
        cfg = EngineConfig(model=PRESETS["tiny-llama-gpu"], device="cuda:0",
                           max_num_seqs=4, max_batched_tokens=1024,
                           max_model_len=2048, kv_pool_pages=12, page_size=64,
                           host_cache_pages=8, object_cache_dir=store)
        return LLMEngine(cfg, seed=seed)

    p1 = list(range(256))
    e_a = make()
    o_a = generate(e_a, "a", p1)
    for i in range(3):
        generate(e_a, f"c{i}", [(300 + i * 320 + j) % 1000 for j in range(320)])
    assert e_a.host_tier.stats["published_object"] > 0
    for _ in range(100):
        if e_a.host_tier.objects.stats["put"] >= \
                e_a.host_tier.stats["published_object"]:
            break
        time.sleep(0.02)
    e_b = make()
    o_b = generate(e_b, "b", p1)
    assert e_b.host_tier.stats["onboarded_object"] > 0
    assert o_b == o_a
