"""KVBM G2 host tier: offload on device eviction, onboard on prefix hit,
with output correctness across the demote/promote cycle (CPU engine)."""
from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
from dynamo_amd.engine.config import PRESETS


def make_engine(host_pages=64):
    cfg = EngineConfig(model=PRESETS["tiny-llama"], device="cpu",
                       max_num_seqs=4, max_batched_tokens=512,
                       max_model_len=512, kv_pool_pages=24, page_size=16,
                       host_cache_pages=host_pages)
    return LLMEngine(cfg, seed=7)


def generate(engine, rid, prompt, max_tokens=4):
    engine.add_request(rid, prompt, SamplingParams(max_tokens=max_tokens))
    out = []
    while engine.has_work():
        for so in engine.step():
            if so.req_id == rid:
                out.append(so.new_token)
    return out


def test_offload_and_onboard_roundtrip():
    eng = make_engine()
    p1 = list(range(64))          # 4 full pages
    o1 = generate(eng, "a", p1)
    # churn: force eviction of p1's cached pages (pool is 24 pages)
    for i in range(4):
        generate(eng, f"churn{i}", [(100 + 80 * i + j) % 500 for j in range(80)])
    assert eng.host_tier.stats["offloaded"] > 0, "nothing offloaded to G2"
    # re-run p1: pages should onboard from the host tier
    before = eng.host_tier.stats["onboarded"]
    o2 = generate(eng, "a2", p1)
    assert eng.host_tier.stats["onboarded"] > before, "no G2 onboard hit"
    assert o2 == o1, "outputs diverged after offload/onboard cycle"


def test_host_tier_lru_eviction():
    eng = make_engine(host_pages=4)
    generate(eng, "a", list(range(64)))
    for i in range(6):
        generate(eng, f"c{i}", [(200 + 64 * i + j) % 500 for j in range(64)])
    st = eng.host_tier.stats
    assert st["offloaded"] > 4
    assert st["evicted_host"] > 0  # tiny host pool had to evict
    assert len(eng.host_tier.map) <= 4


def test_host_events_emitted():
    eng = make_engine()
    generate(eng, "a", list(range(64)))
    for i in range(4):
        generate(eng, f"c{i}", [(300 + 80 * i + j) % 500 for j in range(80)])
    kinds = {e.kind for e in eng.drain_kv_events()}
    assert "stored_host" in kinds
