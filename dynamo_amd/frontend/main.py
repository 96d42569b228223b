"""Frontend CLI: `python -m dynamo_amd.frontend --discovery file:/run/dyn`

The analog of `python -m dynamo.frontend`
(components/src/dynamo/frontend/main.py:33-40): watches discovery for
model cards and serves the OpenAI HTTP API.
"""
from __future__ import annotations

import argparse
import asyncio
import logging

import uvicorn

from dynamo_amd.router import RouterConfig
from dynamo_amd.runtime import DistributedRuntime
from .openai import build_app
from .service import ModelManager


def build_parser():
    p = argparse.ArgumentParser("dynamo_amd.frontend")
    p.add_argument("--discovery", default="memory")
    p.add_argument("--namespace", default="dynamo")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--router-mode", default="kv",
                   choices=["kv", "round_robin", "random", "least_loaded",
                            "p2c"])
    p.add_argument("--router-temperature", type=float, default=0.0)
    p.add_argument("--busy-threshold", type=float, default=0.0,
                   help="reject with 503 when every worker's load exceeds "
                        "this (0 = disabled)")
    p.add_argument("--record", default=None, metavar="FILE",
                   help="record requests + response chunks to a JSONL file "
                        "(replay with python -m dynamo_amd.tools.replay)")
    p.add_argument("--request-template", default=None, metavar="FILE",
                   help="JSON {model, temperature, max_completion_tokens} "
                        "defaults applied when a request omits the field "
                        "(reference request_template.rs parity)")
    p.add_argument("--grpc-port", type=int, default=0,
                   help="also serve the KServe v2 gRPC inference protocol "
                        "on this port (0 = disabled)")
    return p


async def async_main(args):
    logging.basicConfig(level=logging.INFO)
    rt = DistributedRuntime(args.discovery, host=args.host)
    mgr = ModelManager(rt, namespace=args.namespace,
                       router_cfg=RouterConfig(
                           mode=args.router_mode,
                           router_temperature=args.router_temperature,
                           busy_threshold=args.busy_threshold),
                       record_path=args.record)
    await mgr.start()
    tmpl = None
    if args.request_template:
        import json as _json
        tmpl = _json.loads(open(args.request_template).read())
    app = build_app(mgr, request_template=tmpl)
    config = uvicorn.Config(app, host=args.host, port=args.port,
                            log_level="warning")
    server = uvicorn.Server(config)
    grpc_server = None
    if args.grpc_port:
        from .kserve import make_grpc_server
        grpc_server, bound = make_grpc_server(mgr, args.host, args.grpc_port)
        await grpc_server.start()
        print(f"GRPC_READY {args.host}:{bound}", flush=True)
    print(f"FRONTEND_READY http://{args.host}:{args.port}", flush=True)
    await server.serve()
    if grpc_server is not None:
        await grpc_server.stop(grace=1.0)
    await mgr.stop()
    await rt.shutdown()


def main():
    asyncio.run(async_main(build_parser().parse_args()))


if __name__ == "__main__":
    main()
