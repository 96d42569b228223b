"""Standalone KV-router service: `python -m dynamo_amd.router ...`

Parity with the reference's standalone router process
(components/src/dynamo/router/__main__.py): runs a KvRouter over a worker
component and serves selection + state endpoints on the request plane so
external frontends / gateways (the Envoy EPP role) can query placement.
"""
from __future__ import annotations

import argparse
import asyncio
import logging
import signal

from dynamo_amd import envs
from dynamo_amd.router import KvRouter, RouterConfig
from dynamo_amd.runtime import DistributedRuntime


def build_parser():
    p = argparse.ArgumentParser("dynamo_amd.router")
    p.add_argument("--discovery", default=envs.discovery())
    p.add_argument("--namespace", default=envs.namespace())
    p.add_argument("--component", default="backend")
    p.add_argument("--mode", default=envs.get("DYN_ROUTER_MODE", "kv"))
    p.add_argument("--block-size", type=int,
                   default=envs.get("DYN_KV_BLOCK_SIZE", 64, int))
    p.add_argument("--temperature", type=float,
                   default=envs.get("DYN_ROUTER_TEMPERATURE", 0.0, float))
    p.add_argument("--host", default="127.0.0.1")
    return p


async def async_main(args):
    logging.basicConfig(level=logging.INFO)
    rt = DistributedRuntime(args.discovery, host=args.host)
    router = KvRouter(rt, args.namespace, args.component,
                      RouterConfig(mode=args.mode, block_size=args.block_size,
                                   router_temperature=args.temperature))
    await router.start()
    comp = rt.namespace(args.namespace).component("router")

    async def select(payload, ctx):
        iid = router.select(payload["token_ids"])
        yield {"instance_id": iid}

    async def state(payload, ctx):
        yield {
            "workers": {iid: {"active_requests": ws.active_requests,
                              "active_blocks": ws.active_blocks,
                              "kv_usage": ws.kv_usage}
                        for iid, ws in router.workers.items()},
            "indexed_blocks": router.indexer.size(),
        }
    comp.serve_endpoint("select", select)
    comp.serve_endpoint("state", state)
    await comp.register()
    print(f"ROUTER_READY {comp.instance_id} {rt.server.address}", flush=True)

    stop = asyncio.Event()
    loop = asyncio.get_running_loop()
    for sig in (signal.SIGINT, signal.SIGTERM):
        loop.add_signal_handler(sig, stop.set)
    await stop.wait()
    await router.stop()
    await rt.shutdown()


def main():
    asyncio.run(async_main(build_parser().parse_args()))


main()
