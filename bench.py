"""Flagship benchmark: Llama-3-70B serving throughput on MI355X.

Measures the BASELINE.json metric — output tok/s (whole node) with p50
TTFT/ITL for Llama-3-70B at ISL 8192 / OSL 1024, concurrency 16 per GPU
(the reference's Llama-3.3-70B recipe discipline,
docs/.../llama-3-3-70b-topology.mdx:18-22) — on synthetic data with
random-init weights (no network for checkpoints).

N=1 (default): one aggregated engine.
N>1 (driver launches via torch.distributed.run, one rank per GPU over
RCCL): default mode is DISAGGREGATED prefill/decode — ranks [0, N/2) form
a TP prefill group, ranks [N/2, N) a TP decode group (BASELINE config #4:
TP4 prefill + TP4 decode at N=8). All requests prefill on the prefill
group (untimed; records TTFT), KV pages move prefill->decode rank-to-rank
over xGMI via RCCL send/recv, then the decode group runs the timed region.
--mode dp runs data-parallel aggregated engines instead.

Timed region: exactly K decode steps at full concurrency, bracketed by
barrier + torch.cuda.synchronize on both sides; value = aggregate output
tokens / elapsed (max over ranks).
"""
from __future__ import annotations

import argparse
import json
import os
import random
import statistics
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", default="llama-3-70b")
    p.add_argument("--isl", type=int, default=8192)
    p.add_argument("--osl", type=int, default=1024)
    p.add_argument("--conc-per-gpu", type=int, default=16)
    p.add_argument("--page-size", type=int, default=64)
    p.add_argument("--mode", choices=["auto", "dp", "disagg"], default="auto")
    p.add_argument("--kv-pool-pages", type=int, default=0)
    p.add_argument("--kv-cache-dtype", default="auto",
                   choices=["auto", "fp8"],
                   help="fp8 KV measurement runs are SUPPLEMENTARY: the "
                        "flagship headline stays bf16 (BASELINE config)")
    p.add_argument("--kv-v-layout", default="auto", choices=["auto", "never"],
                   help="V-page layout: auto = d-major on GPU when supported")
    p.add_argument("--device", default=None, help="cpu for gloo testing")
    p.add_argument("--max-batched-tokens", type=int, default=0)
    p.add_argument("--moe-ep", action="store_true",
                   help="expert parallelism for MoE models (auto for N>1)")
    return p.parse_args()


def make_cfg(args, mc, device, world=1, rank=0, worker_type="aggregated",
             max_seqs=None):
    from dynamo_amd.engine import EngineConfig
    if mc.num_experts and (world > 1 or args.moe_ep):
        # MoE at N>1 defaults to expert parallelism (BASELINE config #5)
        import dataclasses
        mc = dataclasses.replace(mc, moe_ep=True)
    return EngineConfig(
        model=mc, device=device, page_size=args.page_size,
        max_num_seqs=max_seqs or args.conc_per_gpu,
        max_batched_tokens=args.max_batched_tokens or args.isl,
        max_model_len=args.isl + args.osl + 64,
        kv_pool_pages=args.kv_pool_pages,
        enable_prefix_caching=False,  # synthetic distinct prompts
        kv_cache_dtype=args.kv_cache_dtype,
        kv_v_layout=args.kv_v_layout,
        dtype="bfloat16" if device.startswith("cuda") else "float32",
        worker_type=worker_type, tp_size=world, tp_rank=rank)


def make_prompts(args, mc, n, seed):
    rng = random.Random(seed)
    return [[rng.randrange(mc.vocab_size) for _ in range(args.isl)]
            for _ in range(n)]


def emit(args, world, value, ms_per_step, ttft_p50, prefill_time, mode, conc):
    # BASELINE.json names the Llama-3-70B config; non-default models keep
    # the same metric shape but say what was actually run
    names = {"llama-3-70b": "Llama-3-70B", "llama-3-8b": "Llama-3-8B",
             "mixtral-8x7b": "Mixtral-8x7B", "qwen2-7b": "Qwen2-7B"}
    result = {
        "metric": (f"output tok/s (node), "
                   f"{names.get(args.model, args.model)} serving, "
                   f"ISL{args.isl}/OSL{args.osl}, conc {conc // world}/GPU"),
        "value": round(value, 2),
        "unit": "tok/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if (args.device or "cuda").startswith("cuda") else "fp32",
        "data": "synthetic",
        "config": {
            "model": args.model,
            "global_batch": conc,
            "seq_len": args.isl,
            "osl": args.osl,
            "parallelism": mode,
            "kv_cache_dtype": args.kv_cache_dtype,
            "ttft_p50_s": round(ttft_p50, 3) if ttft_p50 else None,
            "itl_p50_ms": round(ms_per_step, 3),
            "prefill_time_s": round(prefill_time, 2),
        },
    }
    print(json.dumps(result), flush=True)


def dev_sync(device):
    if device.startswith("cuda"):
        torch.cuda.synchronize()


# ---------------------------------------------------------------------------
def run_single_or_dp(args, world, rank, local_rank, device):
    from dynamo_amd.engine import LLMEngine, SamplingParams
    from dynamo_amd.models.registry import resolve_model_config
    mc = resolve_model_config(args.model)
    cfg = make_cfg(args, mc, device)
    t0 = time.monotonic()
    eng = LLMEngine(cfg, seed=0)
    dev_sync(device)
    if rank == 0:
        print(f"# engine init {time.monotonic() - t0:.1f}s "
              f"({eng.runner.num_pages} pages)", flush=True)

    conc = args.conc_per_gpu
    prompts = make_prompts(args, mc, conc, 1234 + rank)
    submit_t = {}
    ttfts = []
    for i, prompt in enumerate(prompts):
        rid = f"r{rank}-{i}"
        eng.add_request(rid, prompt, SamplingParams(
            max_tokens=args.osl, temperature=0.0, ignore_eos=True))
        submit_t[rid] = time.monotonic()

    t_pre0 = time.monotonic()
    while any(not r.output_tokens for r in eng.requests.values()):
        for so in eng.step():
            if so.num_output_tokens == 1:
                ttfts.append(time.monotonic() - submit_t[so.req_id])
    dev_sync(device)
    prefill_time = time.monotonic() - t_pre0

    for _ in range(args.warmup):
        eng.step()
    dev_sync(device)
    if world > 1:
        dist.barrier()
    t0 = time.monotonic()
    gen = 0
    for _ in range(args.steps):
        gen += len(eng.step())
    dev_sync(device)
    elapsed = time.monotonic() - t0
    if world > 1:
        dist.barrier()
        t = torch.tensor([elapsed, float(gen)])
        dist.all_reduce(t[:1], op=dist.ReduceOp.MAX)
        dist.all_reduce(t[1:])
        elapsed, gen = float(t[0]), int(t[1])

    if rank == 0:
        emit(args, world, gen / elapsed, elapsed / args.steps * 1000,
             statistics.median(ttfts) if ttfts else None, prefill_time,
             f"dp{world}" if world > 1 else "agg1", conc * world)


# ---------------------------------------------------------------------------
# Disaggregated P/D: side-local lockstep command protocol. All ranks of a
# side apply identical command lists, so the deterministic engines
# (scheduler + allocator) stay in lockstep and page ids agree across ranks.
def side_apply(cmd, eng, args, side_rank, peer_global, staging, device):
    """Execute one command batch on this rank's engine."""
    from dynamo_amd.engine.scheduler import SamplingParams
    from dynamo_amd.engine.kv_cache import SequenceKV
    from dynamo_amd import ops
    outs = []
    for r in cmd.get("new", []):
        sp = SamplingParams(**r["sampling"])
        req = eng.add_request(r["request_id"], r["token_ids"], sp)
        req.hold_kv = r.get("hold_kv", False)
    for t in cmd.get("transfer_out", []):
        # gather this rank's shard of the pages and send to peer
        req = eng._held[t["request_id"]]
        pages = req.kv.pages[:t["npages"]]
        buf = _pages_to_staging(eng, pages, staging, device)
        dist.send(buf, dst=peer_global)
    for a in cmd.get("attach", []):
        kv = SequenceKV(eng.alloc, eng.cfg.block_salt)
        kv.ensure_capacity(a["num_tokens"])
        n = a["npages"]
        nelem = n * _page_plane_elems(eng)
        buf = staging[:nelem].view(-1)
        dist.recv(buf, src=peer_global)
        _staging_to_pages(eng, kv.pages[:n], buf, device)
        sp = SamplingParams(**a["sampling"])
        req = eng.add_request(a["request_id"], a["token_ids"], sp)
        req.kv = kv
        req.num_computed = a["num_tokens"]
        req.output_tokens.append(a["first_token"])
    for rid in cmd.get("release", []):
        eng.release_held(rid)
    if cmd.get("step"):
        outs = eng.step()
    return outs


def _page_plane_elems(eng):
    pool = eng.runner.kv_pool
    L, two, P = pool.shape[0], pool.shape[1], pool.shape[2]
    return L * two * (pool.buffer.numel() // (L * two * P))


def _pages_to_staging(eng, pages, staging, device):
    pool = eng.runner.kv_pool
    L, two, P = pool.shape[0], pool.shape[1], pool.shape[2]
    pe = pool.buffer.numel() // (L * two * P)
    flat = pool.buffer.reshape(L * two * P, pe)
    n = len(pages)
    buf = staging[: n * L * two * pe].view(n * L * two, pe)
    # page p of plane (l,j) -> staging row ((l*2+j)*n + i)
    idx = []
    for k in range(L * two):
        for p in pages:
            idx.append(k * P + p)
    if device.startswith("cuda"):
        from dynamo_amd import ops
        ids = torch.tensor(idx, dtype=torch.int32, device=flat.device)
        ops.hip().gather_pages(buf.view(-1), flat, ids)
    else:
        buf.copy_(flat[torch.tensor(idx, dtype=torch.long)])
    return buf.view(-1)


def _staging_to_pages(eng, pages, buf, device):
    pool = eng.runner.kv_pool
    L, two, P = pool.shape[0], pool.shape[1], pool.shape[2]
    pe = pool.buffer.numel() // (L * two * P)
    flat = pool.buffer.reshape(L * two * P, pe)
    n = len(pages)
    idx = []
    for k in range(L * two):
        for p in pages:
            idx.append(k * P + p)
    if device.startswith("cuda"):
        from dynamo_amd import ops
        ids = torch.tensor(idx, dtype=torch.int32, device=flat.device)
        ops.hip().scatter_pages(buf, flat, ids)
    else:
        flat[torch.tensor(idx, dtype=torch.long)] = buf.view(n * L * two, pe)


def run_disagg(args, world, rank, local_rank, device):
    from dynamo_amd.engine import LLMEngine, SamplingParams
    from dynamo_amd.models.registry import resolve_model_config
    from dynamo_amd.models.layers import TPContext

    mc = resolve_model_config(args.model)
    PN = world // 2
    is_prefill = rank < PN
    side_rank = rank if is_prefill else rank - PN
    peer_global = rank + PN if is_prefill else rank - PN
    backend = "nccl" if device.startswith("cuda") else "gloo"
    # new_group is collective: every rank constructs every group
    pf_ranks = list(range(PN))
    dc_ranks = list(range(PN, world))
    pf_group = dist.new_group(pf_ranks, backend=backend)
    dc_group = dist.new_group(dc_ranks, backend=backend)
    pf_ctl = dist.new_group(pf_ranks, backend="gloo")
    dc_ctl = dist.new_group(dc_ranks, backend="gloo")
    leaders_ctl = dist.new_group([0, PN], backend="gloo")
    side_ranks = pf_ranks if is_prefill else dc_ranks
    side_group = pf_group if is_prefill else dc_group
    side_ctl = pf_ctl if is_prefill else dc_ctl

    tp = TPContext(PN, side_rank, group=side_group)
    conc_total = args.conc_per_gpu * world
    cfg = make_cfg(args, mc, device, world=PN, rank=side_rank,
                   worker_type="prefill" if is_prefill else "decode",
                   max_seqs=conc_total)
    t0 = time.monotonic()
    eng = LLMEngine(cfg, tp=tp, seed=0)
    dev_sync(device)
    if side_rank == 0:
        print(f"# {'prefill' if is_prefill else 'decode'} engine init "
              f"{time.monotonic() - t0:.1f}s ({eng.runner.num_pages} pages)",
              flush=True)

    # staging buffer: one request's worth of this rank's KV shard
    max_pages_req = (args.isl + args.page_size) // args.page_size + 1
    pe = _page_plane_elems(eng)
    staging = torch.empty(max_pages_req * pe, dtype=eng.runner.kv_pool.dtype,
                          device=device)

    def side_cmd(cmd):
        """Leader broadcasts, everyone applies."""
        box = [cmd]
        dist.broadcast_object_list(box, src=side_ranks[0], group=side_ctl)
        return side_apply(box[0], eng, args, side_rank, peer_global, staging,
                          device)

    is_leader = side_rank == 0
    ttfts = []
    prefill_time = 0.0

    if is_prefill:
        prompts = make_prompts(args, mc, conc_total, 1234)
        t_pre0 = time.monotonic()
        if is_leader:
            submit = {}
            finished = []
            pending = []
            for i, p in enumerate(prompts):
                pending.append({
                    "request_id": f"d{i}", "token_ids": p, "hold_kv": True,
                    "sampling": {"max_tokens": 1, "temperature": 0.0,
                                 "ignore_eos": True}})
                submit[f"d{i}"] = time.monotonic()
            # submit in chunks; step until all prefills finish
            outs_meta = {}
            while len(finished) < conc_total:
                cmd = {"new": pending[:4], "step": True}
                pending = pending[4:]
                outs = side_cmd(cmd)
                now = time.monotonic()
                for so in outs:
                    if so.finished:
                        finished.append(so.req_id)
                        ttfts.append(now - submit[so.req_id])
                        outs_meta[so.req_id] = so.new_token
            prefill_time = time.monotonic() - t_pre0
            # hand off all requests to the decode leader, then transfer
            for i in range(0, conc_total, 4):
                batch = []
                for rid in finished[i:i + 4]:
                    req = eng._held[rid]
                    npages = (req.num_computed + args.page_size - 1) // args.page_size
                    batch.append({
                        "request_id": rid,
                        "token_ids": req.prompt_tokens,
                        "first_token": outs_meta[rid],
                        "num_tokens": req.num_computed,
                        "npages": npages,
                        "sampling": {"max_tokens": args.osl,
                                     "temperature": 0.0, "ignore_eos": True}})
                dist.send_object_list([batch], dst=PN, group=leaders_ctl)
                side_cmd({"transfer_out": [
                    {"request_id": b["request_id"], "npages": b["npages"]}
                    for b in batch]})
                side_cmd({"release": [b["request_id"] for b in batch]})
            dist.send_object_list([{"prefill_time": prefill_time,
                                    "ttfts": ttfts}], dst=PN,
                                  group=leaders_ctl)
            side_cmd({"shutdown": True})
        else:
            while True:
                box = [None]
                dist.broadcast_object_list(box, src=side_ranks[0],
                                           group=side_ctl)
                if box[0].get("shutdown"):
                    break
                side_apply(box[0], eng, args, side_rank, peer_global, staging,
                           device)
        # two global barriers bracket the decode side's timed region
        dev_sync(device)
        dist.barrier()
        dist.barrier()
        if is_leader:
            # rank 0 prints the result line (driver contract); the decode
            # leader measured it and ships it over the leaders ctl group
            box = [None]
            dist.recv_object_list(box, src=PN, group=leaders_ctl)
            r = box[0]
            emit(args, world, r["value"], r["ms_per_step"], r["ttft_p50"],
                 r["prefill_time"], r["mode"], r["conc"])
        return

    # ---- decode side ----
    if is_leader:
        received = 0
        while received < conc_total:
            box = [None]
            dist.recv_object_list(box, src=0, group=leaders_ctl)
            batch = box[0]
            side_cmd({"attach": batch})
            received += len(batch)
        box = [None]
        dist.recv_object_list(box, src=0, group=leaders_ctl)
        info = box[0]
        prefill_time = info["prefill_time"]
        ttfts = info["ttfts"]
        # warmup
        for _ in range(args.warmup):
            side_cmd({"step": True})
        side_cmd({"barrier": True})   # followers: dev_sync + global barrier
        dev_sync(device)
        dist.barrier()
        t0 = time.monotonic()
        gen = 0
        for _ in range(args.steps):
            gen += len(side_cmd({"step": True}))
        side_cmd({"barrier": True})
        dev_sync(device)
        dist.barrier()
        elapsed = time.monotonic() - t0
        side_cmd({"shutdown": True})
        dist.send_object_list([{
            "value": gen / elapsed,
            "ms_per_step": elapsed / args.steps * 1000,
            "ttft_p50": statistics.median(ttfts) if ttfts else None,
            "prefill_time": prefill_time,
            "mode": f"disagg_p{PN}tp{PN}_d{world - PN}tp{world - PN}",
            "conc": conc_total}], dst=0, group=leaders_ctl)
    else:
        while True:
            box = [None]
            dist.broadcast_object_list(box, src=side_ranks[0], group=side_ctl)
            cmd = box[0]
            if cmd.get("shutdown"):
                break
            if cmd.get("barrier"):
                dev_sync(device)
                dist.barrier()
                continue
            side_apply(cmd, eng, args, side_rank, peer_global, staging,
                       device)


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    device = args.device or (f"cuda:{local_rank}"
                             if torch.cuda.is_available() else "cpu")
    if device.startswith("cuda"):
        torch.cuda.set_device(local_rank)
        from dynamo_amd.utils import enable_tunableop
        enable_tunableop(tuning=False)
    if world > 1:
        dist.init_process_group("nccl" if device.startswith("cuda") else "gloo")

    mode = args.mode
    if mode == "auto":
        mode = "disagg" if world > 1 else "dp"
    if world == 1 or mode == "dp":
        run_single_or_dp(args, world, rank, local_rank, device)
    else:
        run_disagg(args, world, rank, local_rank, device)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
