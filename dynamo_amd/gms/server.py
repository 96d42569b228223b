"""GMS weight server CLI: `python -m dynamo_amd.gms --model llama-3-8b ...`

Owns the model weights in a shareable hipMalloc arena and publishes the
hipIpc handle + manifest in discovery; workers started with --gms map the
arena and reconstruct every weight tensor zero-copy (instant warm start;
weights survive worker crashes). Reference parity: lib/gpu_memory_service.
"""
from __future__ import annotations

import argparse
import asyncio
import logging
import signal

import torch

from dynamo_amd.models.registry import build_model, resolve_model_config
from dynamo_amd.models.layers import TPContext
from dynamo_amd.runtime import DistributedRuntime
from .pool import WeightPool, estimate_pool_bytes, weight_allocator


def build_parser():
    p = argparse.ArgumentParser("dynamo_amd.gms")
    p.add_argument("--model", default="llama-3-8b")
    p.add_argument("--discovery", default="memory")
    p.add_argument("--namespace", default="dynamo")
    p.add_argument("--device", default="cuda:0")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--tp-size", type=int, default=1)
    p.add_argument("--tp-rank", type=int, default=0)
    p.add_argument("--host", default="127.0.0.1")
    return p


async def async_main(args):
    logging.basicConfig(level=logging.INFO)
    mc = resolve_model_config(args.model)
    nbytes = estimate_pool_bytes(mc, args.tp_size)
    pool = WeightPool(nbytes, args.device)
    tp = TPContext(args.tp_size, args.tp_rank)
    with weight_allocator(pool):
        model = build_model(mc, args.device, torch.bfloat16, tp, args.seed)
    del model  # views die; the pool owns the memory
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()
        torch.cuda.empty_cache()

    rt = DistributedRuntime(args.discovery, host=args.host)
    comp = rt.namespace(args.namespace).component("gms")

    async def ping(payload, ctx):
        yield {"status": "ok", "used": pool.offset}
    comp.serve_endpoint("ping", ping)
    await comp.register(metadata={
        "gms": pool.export_meta(),
        "model": args.model, "seed": args.seed,
        "tp_size": args.tp_size, "tp_rank": args.tp_rank,
    })
    print(f"GMS_READY {comp.instance_id} used={pool.offset} of {nbytes}",
          flush=True)

    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        loop.add_signal_handler(sig, stop.set)
    await stop.wait()
    await rt.shutdown()


def main():
    asyncio.run(async_main(build_parser().parse_args()))


if __name__ == "__main__":
    main()
