"""Multimodal serving path (CPU): chat content parts -> media decode ->
encode worker -> embedding splice -> LLM engine (embed_spans rows replace
token-table embeddings). Reference parity: preprocessor.rs:2248 media,
encoder-disaggregation.md E/PD flow."""
import asyncio
import base64
import io

import httpx
import pytest

from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
from dynamo_amd.engine.config import PRESETS
from dynamo_amd.frontend.openai import build_app
from dynamo_amd.frontend.service import ModelManager
from dynamo_amd.models.vision import VISION_PRESETS, VisionEncoder
from dynamo_amd.runtime import DistributedRuntime, MemoryDiscovery
from dynamo_amd.workers import WorkerService
from dynamo_amd.workers.encoder import EncoderService


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def png_data_uri(color):
    from PIL import Image
    img = Image.new("RGB", (32, 32), color)
    buf = io.BytesIO()
    img.save(buf, format="PNG")
    return ("data:image/png;base64,"
            + base64.b64encode(buf.getvalue()).decode())


async def _stack():
    shared = MemoryDiscovery()
    rt = DistributedRuntime(shared)
    cfg = EngineConfig(model=PRESETS["tiny-llama"], device="cpu",
                       max_num_seqs=4, max_batched_tokens=256,
                       max_model_len=512, kv_pool_pages=64, page_size=16)
    eng = LLMEngine(cfg, seed=7)
    ws = WorkerService(eng, rt)
    await ws.start()
    ert = DistributedRuntime(shared)
    import dataclasses
    vcfg = dataclasses.replace(VISION_PRESETS["tiny-vit"],
                               out_hidden_size=256)
    enc = EncoderService(VisionEncoder(vcfg, "cpu", seed=3), ert,
                         model_name="tiny-llama")
    await enc.start()
    mgr_rt = DistributedRuntime(shared)
    mgr = ModelManager(mgr_rt)
    await mgr.start(watch_interval=0.2)
    transport = httpx.ASGITransport(app=build_app(mgr))
    client = httpx.AsyncClient(transport=transport, base_url="http://t")
    return (ws, rt, enc, ert, mgr, client)


async def _teardown(ws, rt, enc, ert, mgr, client):
    await client.aclose()
    await mgr.stop()
    await enc.stop()
    await ws.stop()
    await rt.shutdown(drain=False)
    await ert.shutdown(drain=False)


def _chat_body(img_uri):
    return {"model": "tiny-llama", "max_tokens": 6, "ignore_eos": True,
            "messages": [{"role": "user", "content": [
                {"type": "text", "text": "what is in "},
                {"type": "image_url", "image_url": {"url": img_uri}},
                {"type": "text", "text": " this image?"}]}]}


@pytest.mark.timeout(180)
def test_multimodal_chat_end_to_end():
    async def main():
        stack = await _stack()
        ws, rt, enc, ert, mgr, client = stack
        r1 = await client.post("/v1/chat/completions",
                               json=_chat_body(png_data_uri("red")))
        assert r1.status_code == 200, r1.text
        out1 = r1.json()["choices"][0]["message"]
        assert r1.json()["usage"]["completion_tokens"] == 6
        assert enc.count == 1, "encode worker was not used"
        # a DIFFERENT image must change the model's output (embeddings
        # really reach the forward pass)
        r2 = await client.post("/v1/chat/completions",
                               json=_chat_body(png_data_uri("blue")))
        assert r2.status_code == 200
        out2 = r2.json()["choices"][0]["message"]
        assert out1 != out2, "image content did not affect generation"
        # same image -> deterministic same output
        r3 = await client.post("/v1/chat/completions",
                               json=_chat_body(png_data_uri("red")))
        assert r3.json()["choices"][0]["message"] == out1
        # text-only requests still work alongside
        r4 = await client.post("/v1/chat/completions", json={
            "model": "tiny-llama", "max_tokens": 4, "ignore_eos": True,
            "messages": [{"role": "user", "content": "hello"}]})
        assert r4.status_code == 200
        await _teardown(*stack)
    run(main())


@pytest.mark.timeout(120)
def test_embed_spans_engine_level():
    """Sparse embed spans: placeholder rows take provided embeddings and
    change the logits; surrounding text rows still use the token table."""
    import torch
    cfg = EngineConfig(model=PRESETS["tiny-llama"], device="cpu",
                       max_num_seqs=2, max_batched_tokens=128,
                       max_model_len=256, kv_pool_pages=32, page_size=16)

    def gen(spans):
        eng = LLMEngine(cfg, seed=7)
        req = eng.add_request("r", list(range(30)),
                              SamplingParams(max_tokens=4, ignore_eos=True))
        req.embed_spans = spans
        out = []
        while eng.has_work():
            for so in eng.step():
                out.append(so.new_token)
        return out

    g = torch.Generator().manual_seed(5)
    emb_a = torch.randn(8, 256, generator=g) * 0.02
    emb_b = torch.randn(8, 256, generator=g) * 0.02
    base = gen(None)
    with_a = gen([(10, emb_a)])
    with_b = gen([(10, emb_b)])
    assert with_a != base
    assert with_a != with_b
    assert gen([(10, emb_a)]) == with_a   # deterministic
