"""Tensor-parallel engine tests on CPU (gloo, world_size=2).

Verifies the TP lockstep protocol and that a TP-2 engine computes the same
function as TP-1 (deterministic full-weight init + sharding; fp32 so
reduction-order noise cannot flip greedy argmax)."""
import multiprocessing as mp
import os

import pytest
import torch

from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
from dynamo_amd.engine.config import PRESETS

PROMPTS = [list(range(40, 90)), [7, 8, 9] * 20]


def _gen_tp1():
    cfg = EngineConfig(model=PRESETS["tiny-llama"], device="cpu",
                       dtype="float32", max_num_seqs=4,
                       max_batched_tokens=256, max_model_len=512,
                       kv_pool_pages=128, page_size=16)
    eng = LLMEngine(cfg, seed=7)
    outs = {i: [] for i in range(len(PROMPTS))}
    for i, p in enumerate(PROMPTS):
        eng.add_request(f"r{i}", p, SamplingParams(max_tokens=6))
    while eng.has_work():
        for so in eng.step():
            outs[int(so.req_id[1:])].append(so.new_token)
    return [outs[i] for i in range(len(PROMPTS))]


def _tp_rank(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from dynamo_amd.parallel import TPEngineGroup, follower_loop
    from dynamo_amd.models.layers import TPContext
    tp = TPContext(world, rank, group=None)
    tp.control_group = dist.new_group(backend="gloo")
    cfg = EngineConfig(model=PRESETS["tiny-llama"], device="cpu",
                       dtype="float32", max_num_seqs=4,
                       max_batched_tokens=256, max_model_len=512,
                       kv_pool_pages=128, page_size=16,
                       tp_size=world, tp_rank=rank)
    eng = LLMEngine(cfg, tp=tp, seed=7)
    if rank == 0:
        group = TPEngineGroup(eng, tp)
        outs = {i: [] for i in range(len(PROMPTS))}
        for i, p in enumerate(PROMPTS):
            group.add_request(f"r{i}", p, SamplingParams(max_tokens=6))
        steps = 0
        while group.has_work():
            for so in group.step():
                outs[int(so.req_id[1:])].append(so.new_token)
            steps += 1
            assert steps < 200
        group.shutdown()
        q.put([outs[i] for i in range(len(PROMPTS))])
    else:
        follower_loop(eng, tp)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp2_matches_tp1():
    tp1 = _gen_tp1()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29611
    procs = [ctx.Process(target=_tp_rank, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    tp2 = q.get(timeout=240)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    assert tp2 == tp1, f"TP2 {tp2} != TP1 {tp1}"


MOE_PROMPTS = [list(range(30, 70)), [5, 6, 7] * 15]


def _gen_moe_single():
    import dataclasses
    mc = dataclasses.replace(PRESETS["tiny-mixtral"])
    cfg = EngineConfig(model=mc, device="cpu", dtype="float32",
                       max_num_seqs=4, max_batched_tokens=256,
                       max_model_len=512, kv_pool_pages=128, page_size=16)
    eng = LLMEngine(cfg, seed=5)
    outs = {i: [] for i in range(len(MOE_PROMPTS))}
    for i, p in enumerate(MOE_PROMPTS):
        eng.add_request(f"r{i}", p, SamplingParams(max_tokens=5))
    while eng.has_work():
        for so in eng.step():
            outs[int(so.req_id[1:])].append(so.new_token)
    return [outs[i] for i in range(len(MOE_PROMPTS))]


def _ep_rank(rank, world, port, q):
    import dataclasses
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from dynamo_amd.parallel import TPEngineGroup, follower_loop
    from dynamo_amd.models.layers import TPContext
    tp = TPContext(world, rank, group=None)
    tp.control_group = dist.new_group(backend="gloo")
    mc = dataclasses.replace(PRESETS["tiny-mixtral"], moe_ep=True)
    cfg = EngineConfig(model=mc, device="cpu", dtype="float32",
                       max_num_seqs=4, max_batched_tokens=256,
                       max_model_len=512, kv_pool_pages=128, page_size=16,
                       tp_size=world, tp_rank=rank)
    eng = LLMEngine(cfg, tp=tp, seed=5)
    if rank == 0:
        group = TPEngineGroup(eng, tp)
        outs = {i: [] for i in range(len(MOE_PROMPTS))}
        for i, p in enumerate(MOE_PROMPTS):
            group.add_request(f"r{i}", p, SamplingParams(max_tokens=5))
        while group.has_work():
            for so in group.step():
                outs[int(so.req_id[1:])].append(so.new_token)
        group.shutdown()
        q.put([outs[i] for i in range(len(MOE_PROMPTS))])
    else:
        follower_loop(eng, tp)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_moe_expert_parallel_matches_single():
    """EP-2 (experts sharded, partial-sum all-reduce) == single rank.
    Attention stays TP-sharded — the combined TP-attn + EP-MoE deployment."""
    single = _gen_moe_single()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_ep_rank, args=(r, 2, 29612, q))
             for r in range(2)]
    for p in procs:
        p.start()
    ep = q.get(timeout=240)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    assert ep == single, f"EP {ep} != single {single}"


@pytest.mark.timeout(300)
def test_tp4_matches_tp1():
    """TP-4 lockstep == TP-1 (the 70B disagg pools run TP4 groups)."""
    tp1 = _gen_tp1()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_tp_rank, args=(r, 4, 29613, q))
             for r in range(4)]
    for p in procs:
        p.start()
    tp4 = q.get(timeout=240)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    assert tp4 == tp1, f"TP4 {tp4} != TP1 {tp1}"
