"""HF-checkpoint loading: config.json parsing + safetensors mapping.

Round-trip discipline: export a random-init model's weights under HF
names (models/loader.export_hf), load them into a FRESH model, and the
two models must compute bit-identical logits. TP slicing is checked by
comparing rank shards against the exported full tensors.
"""
import dataclasses

import pytest
import torch

from dynamo_amd.engine.config import PRESETS
from dynamo_amd.models.loader import (config_from_hf, export_hf,
                                      load_weights)
from dynamo_amd.models.registry import build_model, resolve_model_config


def _logits(model, cfg):
    from dynamo_amd.engine.kv_cache import KVCachePool
    from dynamo_amd.models.layers import AttnMetadata
    pool = KVCachePool(cfg.num_layers, 8, cfg.num_kv_heads, 16,
                       cfg.head_dim, "cpu", torch.float32)
    T = 12
    meta = AttnMetadata(
        slot_mapping=torch.arange(T, dtype=torch.int64),
        positions=torch.arange(T, dtype=torch.int32),
        num_decode=0, num_prefill_tokens=T,
        prefill_page_table=torch.arange(8, dtype=torch.int32).view(1, -1),
        seq_q_start=torch.tensor([0], dtype=torch.int32),
        seq_q_len=torch.tensor([T], dtype=torch.int32),
        seq_ctx_len=torch.tensor([T], dtype=torch.int32),
    )
    ids = torch.arange(T, dtype=torch.int32) % cfg.vocab_size
    h = model.forward(ids, pool, meta)
    return model.compute_logits(h)


@pytest.mark.parametrize("preset", ["tiny-llama", "tiny-qwen",
                                    "tiny-mixtral"])
def test_hf_roundtrip_bit_exact(preset, tmp_path):
    cfg = dataclasses.replace(PRESETS[preset])
    src = build_model(cfg, "cpu", torch.float32, None, seed=3)
    export_hf(src, str(tmp_path))
    # config.json parses back to the same architecture
    parsed = config_from_hf(str(tmp_path))
    for f in ("hidden_size", "intermediate_size", "num_layers",
              "num_q_heads", "num_kv_heads", "head_dim", "vocab_size",
              "num_experts", "attn_bias", "tie_embeddings"):
        assert getattr(parsed, f) == getattr(cfg, f), f
    dst = build_model(cfg, "cpu", torch.float32, None, seed=99)  # different init
    n = load_weights(dst, str(tmp_path))
    assert n > 0
    a = _logits(src, cfg)
    b = _logits(dst, cfg)
    assert torch.equal(a, b), (a - b).abs().max()


def test_tp_shard_slicing_matches_full():
    """TP2 load: each rank's shard equals the corresponding slice of the
    full checkpoint tensors."""
    from dynamo_amd.models.layers import TPContext
    cfg = dataclasses.replace(PRESETS["tiny-llama"])
    import tempfile
    with tempfile.TemporaryDirectory() as d:
        full = build_model(cfg, "cpu", torch.float32, None, seed=5)
        export_hf(full, d)
        for rank in (0, 1):
            tp = TPContext(2, rank, group=None)
            m = build_model(cfg, "cpu", torch.float32, tp, seed=77)
            load_weights(m, d)
            at = full.layers[0].attn
            mt = m.layers[0].attn
            hd = cfg.head_dim
            qr = mt.hq * hd
            # q rows: rank slice of the full q block
            assert torch.equal(mt.wqkv[:qr],
                               at.wqkv[rank * qr:(rank + 1) * qr])
            # o columns
            oc = at.wo.shape[1] // 2
            assert torch.equal(mt.wo, at.wo[:, rank * oc:(rank + 1) * oc])
            # mlp gate rows
            i_l = m.layers[0].mlp.I
            assert torch.equal(m.layers[0].mlp.w_gate_up[:i_l],
                               full.layers[0].mlp.w_gate_up[:2 * i_l]
                               [rank * i_l:(rank + 1) * i_l])


def test_resolve_model_config_from_dir(tmp_path):
    cfg = dataclasses.replace(PRESETS["tiny-llama"])
    src = build_model(cfg, "cpu", torch.float32, None, seed=1)
    export_hf(src, str(tmp_path))
    r = resolve_model_config(str(tmp_path))
    assert r.weights_path == str(tmp_path)
    assert r.hidden_size == cfg.hidden_size


def test_engine_with_checkpoint(tmp_path):
    """End-to-end: an engine built from a checkpoint DIR generates the
    same tokens as one built from the original random-init model."""
    from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
    cfg = dataclasses.replace(PRESETS["tiny-llama"])
    src = build_model(cfg, "cpu", torch.float32, None, seed=11)
    export_hf(src, str(tmp_path))

    def gen(model_ref):
        e = LLMEngine(EngineConfig(model=model_ref, device="cpu",
                                   dtype="float32",
                                   kv_pool_pages=64, max_model_len=512,
                                   page_size=16), seed=11)
        e.add_request("x", list(range(50)),
                      SamplingParams(max_tokens=6, ignore_eos=True))
        toks = []
        while e.has_work():
            for so in e.step():
                toks.append(so.new_token)
        return toks
    a = gen(cfg)                                   # random init, seed 11
    b = gen(resolve_model_config(str(tmp_path)))   # loaded checkpoint
    assert a == b


def test_worker_card_advertises_checkpoint_tokenizer(tmp_path):
    """A worker built from a checkpoint dir with tokenizer.json publishes
    an hf tokenizer spec + chat template in its model card, and the
    frontend's make_tokenizer loads it."""
    import asyncio
    import json

    from dynamo_amd.engine import EngineConfig, LLMEngine
    from dynamo_amd.frontend.tokenizer import make_tokenizer
    from dynamo_amd.runtime import DistributedRuntime, MemoryDiscovery
    from dynamo_amd.workers import WorkerService

    cfg = dataclasses.replace(PRESETS["tiny-llama"])
    src = build_model(cfg, "cpu", torch.float32, None, seed=2)
    export_hf(src, str(tmp_path))
    # minimal real tokenizers JSON (WordLevel with a tiny vocab)
    vocab = {"<unk>": 0, "</s>": 1, "hello": 2, "world": 3}
    tok = {"version": "1.0", "truncation": None, "padding": None,
           "added_tokens": [], "normalizer": None,
           "pre_tokenizer": {"type": "Whitespace"},
           "post_processor": None, "decoder": None,
           "model": {"type": "WordLevel", "vocab": vocab, "unk_token": "<unk>"}}
    (tmp_path / "tokenizer.json").write_text(json.dumps(tok))
    (tmp_path / "tokenizer_config.json").write_text(json.dumps(
        {"chat_template": "{% for m in messages %}[{{ m.role }}]{{ m.content }}{% endfor %}"}))
    (tmp_path / "generation_config.json").write_text(json.dumps(
        {"eos_token_id": [1, 3]}))

    async def main():
        rt = DistributedRuntime(MemoryDiscovery())
        eng = LLMEngine(EngineConfig(
            model=resolve_model_config(str(tmp_path)), device="cpu",
            dtype="float32", kv_pool_pages=32, max_model_len=256,
            page_size=16))
        ws = WorkerService(eng, rt)
        card = ws._card_body()
        assert card["tokenizer"]["type"] == "hf"
        assert "chat_template" in card
        t = make_tokenizer(card["tokenizer"])
        assert t.encode("hello world") == [2, 3]
        assert t.eos_id == 1
        assert card["eos_token_ids"] == [1, 3]
        await rt.shutdown(drain=False)
    asyncio.new_event_loop().run_until_complete(main())


def test_multifile_checkpoint(tmp_path):
    """Sharded checkpoints (model-00001-of-0000N.safetensors) load the
    same as single-file ones."""
    from safetensors import safe_open
    from safetensors.torch import save_file
    cfg = dataclasses.replace(PRESETS["tiny-llama"])
    src = build_model(cfg, "cpu", torch.float32, None, seed=6)
    export_hf(src, str(tmp_path))
    # split model.safetensors into two shards
    tensors = {}
    with safe_open(str(tmp_path / "model.safetensors"),
                   framework="pt") as f:
        for k in f.keys():
            tensors[k] = f.get_tensor(k)
    (tmp_path / "model.safetensors").unlink()
    keys = sorted(tensors)
    mid = len(keys) // 2
    save_file({k: tensors[k] for k in keys[:mid]},
              str(tmp_path / "model-00001-of-00002.safetensors"))
    save_file({k: tensors[k] for k in keys[mid:]},
              str(tmp_path / "model-00002-of-00002.safetensors"))
    dst = build_model(cfg, "cpu", torch.float32, None, seed=77)
    assert load_weights(dst, str(tmp_path)) == len(keys)
    assert torch.equal(_logits(src, cfg), _logits(dst, cfg))


def test_mixtral_ep_expert_slicing(tmp_path):
    """Under expert parallelism each rank loads only its expert window,
    at full intermediate width."""
    from dynamo_amd.models.layers import TPContext
    cfg = dataclasses.replace(PRESETS["tiny-mixtral"], moe_ep=True)
    full = build_model(dataclasses.replace(cfg, moe_ep=False),
                       "cpu", torch.float32, None, seed=8)
    export_hf(full, str(tmp_path))
    E = cfg.num_experts
    for rank in (0, 1):
        tp = TPContext(2, rank, group=None)
        m = build_model(cfg, "cpu", torch.float32, tp, seed=55)
        load_weights(m, str(tmp_path))
        moe = m.layers[0].moe
        assert moe.ep and moe.El == E // 2
        fmoe = full.layers[0].moe
        for el in range(moe.El):
            e = moe.e0 + el
            assert torch.equal(moe.w_gate_up[el], fmoe.w_gate_up[e])
            assert torch.equal(moe.w_down[el], fmoe.w_down[e])
