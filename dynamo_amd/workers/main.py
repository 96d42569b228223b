"""Worker CLI: `python -m dynamo_amd.workers [--mock] --model ...`

The native analog of the reference's engine-worker entrypoints
(`python -m dynamo.vllm`, components/src/dynamo/vllm/main.py:149): builds
the engine (native CDNA4 or mock), registers the model card in discovery,
and serves the worker endpoint contract on the request plane.
"""
from __future__ import annotations

import argparse
import asyncio
import logging
import os
import signal
import tempfile

from dynamo_amd.engine import EngineConfig, LLMEngine
from dynamo_amd.models.registry import resolve_model_config
from dynamo_amd.runtime import DistributedRuntime
from .service import WorkerService


def build_parser():
    p = argparse.ArgumentParser("dynamo_amd.workers")
    p.add_argument("--model", default="tiny-llama")
    p.add_argument("--mock", action="store_true", help="GPU-free mock engine")
    p.add_argument("--device", default=None, help="cuda:0 | cpu (default auto)")
    p.add_argument("--discovery", default="memory",
                   help="memory | file:/path (DYN_DISCOVERY_BACKEND analog)")
    p.add_argument("--namespace", default="dynamo")
    p.add_argument("--component", default=None,
                   help="default: backend, or prefill for prefill workers")
    p.add_argument("--worker-type", default="aggregated",
                   choices=["aggregated", "prefill", "decode"])
    p.add_argument("--page-size", type=int, default=64)
    p.add_argument("--max-num-seqs", type=int, default=64)
    p.add_argument("--max-batched-tokens", type=int, default=8192)
    p.add_argument("--max-model-len", type=int, default=16384)
    p.add_argument("--kv-pool-pages", type=int, default=0)
    p.add_argument("--gpu-mem-fraction", type=float, default=0.9)
    p.add_argument("--no-prefix-caching", action="store_true")
    p.add_argument("--no-hip-graphs", action="store_true")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--kv-cache-dtype", default="auto",
                   choices=["auto", "fp8"],
                   help="fp8 stores the paged KV cache as OCP e4m3 "
                        "(half the decode HBM bytes; GPU only)")
    p.add_argument("--kv-v-layout", default="auto",
                   choices=["auto", "never"],
                   help="V-page layout: auto = d-major on GPU when supported")
    p.add_argument("--dtype", default=None,
                   help="bfloat16 | float32 (default: bf16 on GPU, "
                        "fp32 on CPU)")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--host-cache-pages", type=int, default=0,
                   help="KVBM G2 pinned-host tier size in pages")
    p.add_argument("--host-cache-policy", default="lru",
                   choices=["lru", "tinylfu"],
                   help="G2 eviction: lru, or tinylfu admission filter")
    p.add_argument("--disk-cache-pages", type=int, default=0,
                   help="KVBM G3 disk tier size in pages")
    p.add_argument("--disk-cache-path", default="",
                   help="G3 backing file (default <tmp>/dynamo_kv_g3.bin)")
    p.add_argument("--object-cache-dir", default="",
                   help="G4 shared object-store directory (cross-worker "
                        "KV reuse; disabled when empty)")
    p.add_argument("--gms", action="store_true",
                   help="import weights zero-copy from a GMS weight server")
    p.add_argument("--status-port", type=int, default=0,
                   help="serve a JSON system-status HTTP endpoint on this "
                        "port (reference DYN_SYSTEM_PORT parity; 0 = off)")
    p.add_argument("--moe-ep", action="store_true",
                   help="MoE models at tp>1: expert parallelism over the "
                        "TP group (default: auto-on for MoE at tp>1; "
                        "reference flag moe_ep_size)")
    p.add_argument("--no-moe-ep", action="store_true",
                   help="force MoE tensor-parallel sharding instead of EP")
    p.add_argument("--tp-size", type=int, default=1,
                   help="tensor parallelism: launch via torchrun "
                        "--nproc-per-node N, one rank per GPU over RCCL; "
                        "rank 0 serves the request plane, followers run "
                        "the lockstep loop (reference parity: engine "
                        "--tensor-parallel-size, recipes/llama-3-70b/"
                        "vllm/disagg-single-node/deploy.yaml:57,100)")
    return p


def make_engine_from_args(args, tp=None) -> LLMEngine:
    mc = resolve_model_config(args.model)
    if mc.num_experts and tp is not None and tp.size > 1:
        # expert parallelism over the TP group (config #5 deployment);
        # --no-moe-ep keeps MoE-TP sharding instead
        import dataclasses
        mc = dataclasses.replace(mc, moe_ep=not args.no_moe_ep)
    if args.mock:
        from dynamo_amd.mocker import make_mock_engine
        return make_mock_engine(
            model=mc, page_size=args.page_size, max_num_seqs=args.max_num_seqs,
            max_batched_tokens=args.max_batched_tokens,
            max_model_len=args.max_model_len,
            num_pages=args.kv_pool_pages or 1024,
            worker_type=args.worker_type)
    import torch
    if tp is not None and torch.cuda.is_available():
        device = f"cuda:{int(os.environ.get('LOCAL_RANK', tp.rank))}"
        torch.cuda.set_device(device)
    else:
        device = args.device or ("cuda:0" if torch.cuda.is_available()
                                 else "cpu")
    if device.startswith("cuda"):
        from dynamo_amd.utils import enable_tunableop
        enable_tunableop(tuning=False)
    weight_pool = None
    if args.gms:
        from dynamo_amd.gms import WeightPool
        from dynamo_amd.runtime import make_discovery
        disc = make_discovery(args.discovery)
        metas = [i for i in disc.list(args.namespace, "gms")
                 if i.metadata.get("model") == args.model]
        if not metas:
            raise RuntimeError(f"no GMS server found for {args.model}")
        weight_pool = WeightPool.open(metas[0].metadata["gms"], device)
        logging.info("imported %d weight bytes zero-copy from GMS %s",
                     weight_pool.buffer.numel(), metas[0].instance_id)
    cfg = EngineConfig(
        model=mc, device=device,
        dtype=(args.dtype or
               ("bfloat16" if device.startswith("cuda") else "float32")),
        page_size=args.page_size,
        max_num_seqs=args.max_num_seqs,
        max_batched_tokens=args.max_batched_tokens,
        max_model_len=args.max_model_len,
        kv_pool_pages=args.kv_pool_pages,
        gpu_mem_fraction=args.gpu_mem_fraction,
        kv_cache_dtype=args.kv_cache_dtype,
        kv_v_layout=args.kv_v_layout,
        enable_prefix_caching=not args.no_prefix_caching,
        enable_hip_graphs=not args.no_hip_graphs,
        worker_type=args.worker_type,
        tp_size=tp.size if tp is not None else 1,
        tp_rank=tp.rank if tp is not None else 0,
        # CPU workers in a P/D split export the pool as a shared mapping
        # so decode processes can pull (GPU uses hipIpc)
        cpu_shm_pool=(device == "cpu"
                      and args.worker_type in ("prefill", "decode")),
        host_cache_pages=args.host_cache_pages,
        host_cache_policy=args.host_cache_policy,
        disk_cache_pages=args.disk_cache_pages,
        object_cache_dir=args.object_cache_dir,
        disk_cache_path=(args.disk_cache_path or
                         (os.path.join(tempfile.gettempdir(),
                                       f"dynamo_kv_g3_{os.getpid()}.bin")
                          if args.disk_cache_pages else "")))
    return LLMEngine(cfg, tp=tp, seed=args.seed, weight_pool=weight_pool)


async def async_main(args, engine=None, kv_meta=None):
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s %(levelname)s %(message)s")
    if engine is None:
        engine = make_engine_from_args(args)
    rt = DistributedRuntime(args.discovery, host=args.host)
    component = args.component or (
        "prefill" if args.worker_type == "prefill" else "backend")
    ws = WorkerService(engine, rt, namespace=args.namespace,
                       component=component, kv_transfer_meta=kv_meta)
    await ws.start()
    status_runner = None
    if args.status_port:
        from .status import start_status_server
        status_runner = await start_status_server(ws, args.host,
                                                  args.status_port)
        print(f"STATUS_READY http://{args.host}:{args.status_port}",
              flush=True)
    logging.info("worker %s (%s) serving %s on %s", ws.instance_id,
                 args.worker_type, args.model, rt.server.address)
    print(f"WORKER_READY {ws.instance_id} {rt.server.address}", flush=True)

    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        loop.add_signal_handler(sig, stop.set)
    await stop.wait()
    if status_runner is not None:
        await status_runner()
    await ws.stop()
    await rt.shutdown()


def run_tp(args):
    """TP>1 worker: torchrun launches one process per GPU; rank 0 serves
    the request plane behind a TPEngineGroup facade, ranks 1..N-1 apply
    the lockstep command broadcast (parallel/tp.follower_loop)."""
    import uuid

    import torch.distributed as dist

    from dynamo_amd.disagg.transfer import tp_transfer_metadata
    from dynamo_amd.parallel.tp import (TPEngineGroup, follower_loop,
                                        init_tp)
    tp = init_tp()
    if tp.size != args.tp_size:
        raise RuntimeError(f"torchrun world size {tp.size} != "
                           f"--tp-size {args.tp_size}")
    engine = make_engine_from_args(args, tp=tp)
    kv_meta = None
    if engine.runner.kv_pool is not None:
        box = [uuid.uuid4().hex if tp.rank == 0 else None]
        dist.broadcast_object_list(box, src=0, group=tp.control_group)
        kv_meta = tp_transfer_metadata(box[0], engine.runner.kv_pool, tp)
    if tp.rank == 0:
        group = TPEngineGroup(engine, tp)
        try:
            asyncio.run(async_main(args, engine=group, kv_meta=kv_meta))
        finally:
            group.shutdown()
    else:
        logging.basicConfig(level=logging.INFO)
        follower_loop(engine, tp)
    dist.destroy_process_group()


def main():
    args = build_parser().parse_args()
    if args.tp_size > 1:
        run_tp(args)
    else:
        asyncio.run(async_main(args))


if __name__ == "__main__":
    main()
