"""Request plane: multiplexed streaming RPC over raw TCP.

The transport role of the reference's TCP request plane
(ai-dynamo/dynamo lib/runtime/src/pipeline/network/{egress,ingress},
transports/tcp.rs): workers serve endpoints behind an accept loop
(PushEndpoint::start, ingress/push_endpoint.rs:21), clients keep pooled
multiplexed connections (egress/tcp_client.rs). Frames use the two-part
codec; responses stream until a `final` frame; `cancel` frames propagate
client-side cancellation (AsyncEngineContext::stop_generating parity,
runtime/src/engine.rs:117).
"""
from __future__ import annotations

import asyncio
import itertools
import logging
from typing import Any, AsyncIterator, Awaitable, Callable, Dict, Optional

from .codec import encode_frame, read_frame

log = logging.getLogger("dynamo_amd.request_plane")

Handler = Callable[[Any, "RequestContext"], AsyncIterator[Any]]


class RequestContext:
    """Per-request context handed to endpoint handlers (cancellation)."""

    def __init__(self, rid: int):
        self.rid = rid
        self._cancelled = asyncio.Event()
        self._callbacks = []

    @property
    def cancelled(self) -> bool:
        return self._cancelled.is_set()

    def on_cancel(self, cb):
        """Register a callback fired when a cancel frame arrives (lets
        handlers wake event-driven waits instead of polling `cancelled`)."""
        if self._cancelled.is_set():
            cb()
        else:
            self._callbacks.append(cb)

    def cancel(self):
        self._cancelled.set()
        cbs, self._callbacks = self._callbacks, []
        for cb in cbs:
            try:
                cb()
            except Exception:
                pass


class RequestPlaneServer:
    """Serves named endpoints; handlers are async generators of chunks."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0):
        # host "unix:/path" serves on a Unix domain socket (second
        # transport next to TCP — reference has a pluggable transport
        # matrix, runtime/src/transports)
        self.host = host
        self.port = port
        self.uds_path = host[5:] if host.startswith("unix:") else None
        self.handlers: Dict[str, Handler] = {}
        self._server: Optional[asyncio.AbstractServer] = None
        self._inflight: Dict[tuple, asyncio.Task] = {}
        self._contexts: Dict[tuple, RequestContext] = {}
        self._conn_counter = itertools.count()
        self._writers: set = set()

    def add_endpoint(self, name: str, handler: Handler):
        self.handlers[name] = handler

    @property
    def address(self) -> str:
        if self.uds_path is not None:
            return f"unix:{self.uds_path}"
        return f"{self.host}:{self.port}"

    async def start(self):
        if self.uds_path is not None:
            self._server = await asyncio.start_unix_server(self._on_conn,
                                                           self.uds_path)
            return self.address
        self._server = await asyncio.start_server(self._on_conn, self.host,
                                                  self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.address

    async def stop(self, drain: bool = True):
        if self._server:
            self._server.close()
            await self._server.wait_closed()
        if drain and self._inflight:
            await asyncio.gather(*self._inflight.values(),
                                 return_exceptions=True)
        for t in self._inflight.values():
            t.cancel()
        # close established connections (clients see EOF and fail over)
        for w in list(self._writers):
            try:
                w.close()
            except Exception:
                pass

    async def _on_conn(self, reader: asyncio.StreamReader,
                       writer: asyncio.StreamWriter):
        cid = next(self._conn_counter)
        wlock = asyncio.Lock()
        self._writers.add(writer)
        try:
            while True:
                frame = await read_frame(reader)
                if frame is None:
                    break
                header, body = frame
                t = header.get("type")
                rid = header.get("rid")
                if t == "req":
                    key = (cid, rid)
                    ctx = RequestContext(rid)
                    self._contexts[key] = ctx
                    task = asyncio.create_task(self._run_handler(
                        header.get("endpoint", ""), body, ctx, rid, writer,
                        wlock, key))
                    self._inflight[key] = task
                elif t == "cancel":
                    ctx = self._contexts.get((cid, rid))
                    if ctx:
                        ctx.cancel()
        except (ConnectionResetError, BrokenPipeError, asyncio.CancelledError):
            pass
        finally:
            # connection gone: cancel its in-flight work
            for key in [k for k in self._inflight if k[0] == cid]:
                ctx = self._contexts.get(key)
                if ctx:
                    ctx.cancel()
            self._writers.discard(writer)
            writer.close()

    async def _run_handler(self, endpoint, body, ctx, rid, writer, wlock, key):
        async def send(header, payload):
            async with wlock:
                writer.write(encode_frame(header, payload))
                await writer.drain()

        try:
            handler = self.handlers.get(endpoint)
            if handler is None:
                await send({"type": "rsp", "rid": rid, "final": True,
                            "error": f"no such endpoint {endpoint!r}"}, None)
                return
            agen = handler(body, ctx)
            try:
                async for chunk in agen:
                    if ctx.cancelled:
                        break
                    await send({"type": "rsp", "rid": rid, "final": False},
                               chunk)
            finally:
                # close the handler promptly so its cleanup (e.g. engine
                # abort on cancellation) runs NOW, not at GC time
                if hasattr(agen, "aclose"):
                    await agen.aclose()
            await send({"type": "rsp", "rid": rid, "final": True}, None)
        except (ConnectionResetError, BrokenPipeError):
            pass
        except Exception as e:  # handler error -> error frame
            log.exception("handler %s failed", endpoint)
            try:
                await send({"type": "rsp", "rid": rid, "final": True,
                            "error": f"{type(e).__name__}: {e}"}, None)
            except Exception:
                pass
        finally:
            self._inflight.pop(key, None)
            self._contexts.pop(key, None)


class EndpointError(RuntimeError):
    pass


class _Conn:
    def __init__(self, reader, writer):
        self.reader = reader
        self.writer = writer
        self.wlock = asyncio.Lock()
        self.queues: Dict[int, asyncio.Queue] = {}
        self.rid_counter = itertools.count(1)
        self.reader_task = asyncio.create_task(self._read_loop())
        self.closed = False

    async def _read_loop(self):
        try:
            while True:
                frame = await read_frame(self.reader)
                if frame is None:
                    break
                header, body = frame
                q = self.queues.get(header.get("rid"))
                if q is not None:
                    q.put_nowait((header, body))
        except Exception:
            pass
        finally:
            self.closed = True
            for q in self.queues.values():
                q.put_nowait(({"type": "rsp", "final": True,
                               "error": "connection lost"}, None))

    async def send(self, header, body):
        # Explicit acquire/release: if this coroutine is GC'd mid-drain at
        # interpreter teardown (GeneratorExit), a plain `async with` would
        # release the lock via call_soon on a CLOSED loop and raise an
        # unraisable RuntimeError; guard the release instead.
        await self.wlock.acquire()
        try:
            self.writer.write(encode_frame(header, body))
            await self.writer.drain()
        finally:
            try:
                self.wlock.release()
            except RuntimeError:
                pass  # event loop already closed


class RequestPlaneClient:
    """Pooled multiplexed client; one connection per remote address."""

    def __init__(self):
        self._conns: Dict[str, _Conn] = {}
        self._lock = asyncio.Lock()

    async def _conn(self, address: str) -> _Conn:
        async with self._lock:
            c = self._conns.get(address)
            if c is None or c.closed:
                if address.startswith("unix:"):
                    reader, writer = await asyncio.open_unix_connection(
                        address[5:])
                else:
                    host, port = address.rsplit(":", 1)
                    reader, writer = await asyncio.open_connection(
                        host, int(port))
                c = _Conn(reader, writer)
                self._conns[address] = c
            return c

    async def call_stream(self, address: str, endpoint: str,
                          payload: Any) -> AsyncIterator[Any]:
        """Async generator of response chunks; raises EndpointError on
        handler error; propagates cancellation as a cancel frame."""
        c = await self._conn(address)
        rid = next(c.rid_counter)
        q: asyncio.Queue = asyncio.Queue()
        c.queues[rid] = q
        try:
            await c.send({"type": "req", "rid": rid, "endpoint": endpoint},
                         payload)
            while True:
                header, body = await q.get()
                if header.get("error"):
                    raise EndpointError(header["error"])
                if header.get("final"):
                    break
                yield body
        except (asyncio.CancelledError, GeneratorExit):
            # CancelledError: the consuming task was cancelled.
            # GeneratorExit: the consumer broke out of its `async for` and
            # this generator is being aclose()d — awaiting a last send here
            # is legal as long as we don't yield again.
            try:
                if not c.closed:
                    await c.send({"type": "cancel", "rid": rid}, None)
            except Exception:
                pass
            raise
        finally:
            c.queues.pop(rid, None)

    async def call(self, address: str, endpoint: str, payload: Any) -> Any:
        """Unary helper: returns the single (or last) chunk."""
        last = None
        async for chunk in self.call_stream(address, endpoint, payload):
            last = chunk
        return last

    async def close(self):
        for c in self._conns.values():
            c.reader_task.cancel()
            c.writer.close()
        self._conns.clear()
