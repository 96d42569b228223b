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
