"""Worker system-status HTTP server.

Reference parity: the per-process system status server exposed on
DYN_SYSTEM_PORT (ai-dynamo/dynamo lib/runtime/src/system_status_server.rs):
liveness + engine/scheduler state as JSON for ops tooling, independent of
the request plane.
"""
from __future__ import annotations

import json


async def start_status_server(ws, host: str, port: int):
    """Tiny stdlib-only HTTP/1.0 responder; returns an async closer."""
    import asyncio

    def payload() -> bytes:
        eng = ws.engine
        m = eng.last_metrics
        body = json.dumps({
            "status": "ok",
            "instance_id": ws.instance_id,
            "model": ws.model_name,
            "worker_type": ws.worker_type,
            "step": m.step,
            "num_running": eng.scheduler.num_running(),
            "num_waiting": eng.scheduler.num_waiting(),
            "kv_usage": m.kv_usage,
            "total_kv_pages": eng.alloc.num_pages,
            "paused": not ws._paused.is_set(),
            "loras": eng.list_loras(),
        }).encode()
        return (b"HTTP/1.0 200 OK\r\nContent-Type: application/json\r\n"
                b"Content-Length: " + str(len(body)).encode() +
                b"\r\n\r\n" + body)

    async def handle(reader, writer):
        try:
            await reader.readline()       # request line; rest ignored
            writer.write(payload())
            await writer.drain()
        except Exception:
            pass
        finally:
            writer.close()

    server = await asyncio.start_server(handle, host, port)

    async def close():
        server.close()
        await server.wait_closed()

    return close
