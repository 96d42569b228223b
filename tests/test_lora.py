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


def test_peft_adapter_exact_vs_dense_delta(tmp_path):
    """PEFT-format adapter (separate q/k/v/o + gate/up/down targets) loads
    onto the fused projections as a block-form LoRA; activating it must
    equal applying the dense deltas W += scale*B@A directly."""
    import dataclasses
    import json

    import torch
    from safetensors.torch import save_file

    from dynamo_amd.engine.config import PRESETS
    from dynamo_amd.lora.manager import LoRAManager
    from dynamo_amd.models.registry import build_model

    cfg = dataclasses.replace(PRESETS["tiny-llama"])
    torch.manual_seed(0)
    model = build_model(cfg, "cpu", torch.float32, None, seed=4)
    r, alpha = 4, 8.0
    g = torch.Generator().manual_seed(9)
    sd = {}
    hd = cfg.head_dim
    D = cfg.hidden_size
    dims = {"q_proj": cfg.num_q_heads * hd, "k_proj": cfg.num_kv_heads * hd,
            "v_proj": cfg.num_kv_heads * hd, "o_proj": D,
            "gate_proj": cfg.intermediate_size,
            "up_proj": cfg.intermediate_size, "down_proj": D}
    ins = {"q_proj": D, "k_proj": D, "v_proj": D,
           "o_proj": cfg.num_q_heads * hd, "gate_proj": D, "up_proj": D,
           "down_proj": cfg.intermediate_size}
    for li in range(cfg.num_layers):
        for proj, outf in dims.items():
            mod = "self_attn" if "proj" in proj and proj[0] in "qkvo" else "mlp"
            stem = f"base_model.model.model.layers.{li}.{mod}.{proj}"
            sd[f"{stem}.lora_A.weight"] = torch.randn(r, ins[proj],
                                                      generator=g) * 0.1
            sd[f"{stem}.lora_B.weight"] = torch.randn(outf, r,
                                                      generator=g) * 0.1
    save_file(sd, str(tmp_path / "adapter_model.safetensors"))
    (tmp_path / "adapter_config.json").write_text(json.dumps(
        {"r": r, "lora_alpha": alpha,
         "target_modules": list(dims)}))

    x = torch.randn(5, D)
    lm = LoRAManager(model)
    lm.load("a", str(tmp_path))
    lm.activate("a")
    at = model.layers[0].attn
    from dynamo_amd.models.layers import linear_lora
    y_lora = linear_lora(x, at.wqkv, at.lora, "qkv")
    # dense reference: per-projection deltas applied to the fused weight
    scale = alpha / r
    wq = at.wqkv.clone()
    qr, kr = at.hq * hd, at.hkv * hd
    pre = "base_model.model.model.layers.0.self_attn."
    wq[:qr] += scale * (sd[pre + "q_proj.lora_B.weight"]
                        @ sd[pre + "q_proj.lora_A.weight"])
    wq[qr:qr + kr] += scale * (sd[pre + "k_proj.lora_B.weight"]
                               @ sd[pre + "k_proj.lora_A.weight"])
    wq[qr + kr:] += scale * (sd[pre + "v_proj.lora_B.weight"]
                             @ sd[pre + "v_proj.lora_A.weight"])
    y_ref = x @ wq.t()
    assert torch.allclose(y_lora, y_ref, rtol=1e-4, atol=1e-4)
    # mlp fused gate_up too
    mlp = model.layers[0].mlp
    y2 = linear_lora(x, mlp.w_gate_up, mlp.lora, "gate_up")
    wg = mlp.w_gate_up.clone()
    pm = "base_model.model.model.layers.0.mlp."
    wg[:mlp.I] += scale * (sd[pm + "gate_proj.lora_B.weight"]
                           @ sd[pm + "gate_proj.lora_A.weight"])
    wg[mlp.I:] += scale * (sd[pm + "up_proj.lora_B.weight"]
                           @ sd[pm + "up_proj.lora_A.weight"])
    assert torch.allclose(y2, x @ wg.t(), rtol=1e-4, atol=1e-4)
    # partial adapter (q only) activates without KeyError
    sd2 = {k: v for k, v in sd.items() if "q_proj" in k}
    save_file(sd2, str(tmp_path / "adapter_model.safetensors"))
    lm.load("partial", str(tmp_path))
    lm.activate("partial")
    assert "o" not in (model.layers[0].attn.lora or {})
