"""GPU disaggregation E2E: two real-engine worker processes on one MI355X
(prefill + decode), KV handoff via hipIpc + page-copy kernel over the
request-plane protocol; outputs must match an aggregated run bit-for-bit
(reference parity: tests/kvbm_integration/test_determinism_disagg.py)."""
import asyncio
import random

import pytest
import torch

from tests.proc_utils import ManagedProcess, worker_cmd

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(600)
def test_disagg_matches_aggregated(tmp_path):
    disc = f"file:{tmp_path}/disc"
    rng = random.Random(7)
    prompt = [rng.randrange(1024) for _ in range(2200)]  # > bypass threshold

    common = dict(model="tiny-llama-gpu", discovery=disc, kv_pool_pages=512,
                  max_batched_tokens=4096, max_model_len=8192)
    decode_w = ManagedProcess(worker_cmd(worker_type="aggregated", **common),
                              ready_marker="WORKER_READY").start()
    prefill_w = ManagedProcess(worker_cmd(worker_type="prefill", **common),
                               ready_marker="WORKER_READY").start()
    try:
        from dynamo_amd.frontend.service import ModelManager
        from dynamo_amd.runtime import DistributedRuntime

        async def run_disagg():
            rt = DistributedRuntime(disc)
            mgr = ModelManager(rt)
            await mgr.start(watch_interval=0.2)
            entry = None
            for _ in range(100):
                try:
                    entry = mgr.get("tiny-llama-gpu")
                    if entry.prefill_router is not None and \
                            entry.prefill_router.has_prefill_pool() and \
                            entry.router.client.instances():
                        break
                except KeyError:
                    pass
                await asyncio.sleep(0.2)
            assert entry is not None and entry.prefill_router is not None
            toks = []
            async for chunk in mgr.generate_tokens(
                    entry, prompt, {"temperature": 0.0},
                    {"max_tokens": 12, "ignore_eos": True}):
                toks.extend(chunk.get("token_ids", []))
            await mgr.stop()
            await rt.shutdown(drain=False)
            return toks

        disagg_tokens = asyncio.new_event_loop().run_until_complete(
            run_disagg())
        assert len(disagg_tokens) == 12
    finally:
        decode_w.stop()
        prefill_w.stop()

    # aggregated reference in this process (same seed -> same weights)
    from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from dynamo_amd.engine.config import PRESETS
    cfg = EngineConfig(model=PRESETS["tiny-llama-gpu"], device="cuda:0",
                       kv_pool_pages=512, max_batched_tokens=4096,
                       max_model_len=8192, max_num_seqs=8, page_size=64)
    eng = LLMEngine(cfg, seed=0)
    eng.add_request("ref", prompt, SamplingParams(max_tokens=12,
                                                  ignore_eos=True))
    agg_tokens = []
    while eng.has_work():
        for so in eng.step():
            agg_tokens.append(so.new_token)
    assert disagg_tokens == agg_tokens, (
        f"disagg {disagg_tokens} != aggregated {agg_tokens}")
