"""Runtime-layer tests: codec, request plane streaming RPC, discovery,
endpoint registration + push-client routing, cancellation, worker death."""
import asyncio
import time

import pytest

from dynamo_amd.runtime import (DistributedRuntime, EndpointError,
                                FileDiscovery, Instance, MemoryDiscovery,
                                NoInstancesError)
from dynamo_amd.runtime.codec import decode_frame, decode_prefix, encode_frame


def test_codec_roundtrip():
    h = {"type": "req", "rid": 3, "endpoint": "gen"}
    b = {"tokens": list(range(100)), "text": "héllo", "b": b"\x00\x01"}
    frame = encode_frame(h, b)
    hlen, blen, csum = decode_prefix(frame[:24])
    h2, b2 = decode_frame(frame[24:24 + hlen], frame[24 + hlen:], csum)
    assert h2 == h
    assert b2["tokens"] == b["tokens"]
    assert b2["b"] == b["b"]
    with pytest.raises(ValueError):
        bad = frame[:30] + bytes([frame[30] ^ 1]) + frame[31:]
        decode_frame(bad[24:24 + hlen], bad[24 + hlen:], csum)


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_request_plane_streaming():
    async def main():
        shared = MemoryDiscovery()
        rt = DistributedRuntime(shared)
        comp = rt.namespace("ns").component("backend")

        async def gen(payload, ctx):
            for i in range(payload["n"]):
                yield {"i": i}
        comp.serve_endpoint("generate", gen)
        await comp.register()

        client = rt.namespace("ns").component("backend").endpoint("generate").client()
        client.runtime = rt  # same-process client
        chunks = [c async for c in client.generate({"n": 5})]
        assert chunks == [{"i": i} for i in range(5)]
        await rt.shutdown()
    run(main())


def test_request_plane_error_propagation():
    async def main():
        rt = DistributedRuntime(MemoryDiscovery())
        comp = rt.namespace("ns").component("backend")

        async def bad(payload, ctx):
            yield {"ok": 1}
            raise RuntimeError("boom")
        comp.serve_endpoint("generate", bad)
        await comp.register()
        client = comp.endpoint("generate").client()
        with pytest.raises(EndpointError, match="boom"):
            async for _ in client.generate({}):
                pass
        await rt.shutdown()
    run(main())


def test_cancellation_propagates():
    async def main():
        rt = DistributedRuntime(MemoryDiscovery())
        comp = rt.namespace("ns").component("backend")
        seen = {"max": 0, "cancelled": False}

        async def slow(payload, ctx):
            for i in range(1000):
                if ctx.cancelled:
                    seen["cancelled"] = True
                    return
                seen["max"] = i
                yield {"i": i}
                await asyncio.sleep(0.001)
        comp.serve_endpoint("generate", slow)
        await comp.register()
        client = comp.endpoint("generate").client()

        async def consume():
            async for c in client.generate({}):
                if c["i"] >= 3:
                    raise asyncio.CancelledError
        with pytest.raises(asyncio.CancelledError):
            await consume()
        await asyncio.sleep(0.1)
        assert seen["max"] < 999  # handler stopped early
        await rt.shutdown()
    run(main())


def test_round_robin_across_instances():
    async def main():
        shared = MemoryDiscovery()
        rts = []
        for w in range(3):
            rt = DistributedRuntime(shared)
            comp = rt.namespace("ns").component("backend")

            async def gen(payload, ctx, w=w):
                yield {"worker": w}
            comp.serve_endpoint("generate", gen)
            await comp.register()
            rts.append(rt)
        client_rt = DistributedRuntime(shared)
        client = client_rt.namespace("ns").component("backend").endpoint(
            "generate").client()
        seen = set()
        for _ in range(9):
            r = await client.call({})
            seen.add(r["worker"])
        assert seen == {0, 1, 2}
        for rt in rts:
            await rt.shutdown()
        await client_rt.shutdown()
    run(main())


def test_dead_worker_inhibited_and_failover():
    async def main():
        shared = MemoryDiscovery()
        rt1 = DistributedRuntime(shared)
        c1 = rt1.namespace("ns").component("backend")

        async def gen1(payload, ctx):
            yield {"worker": 1}
        c1.serve_endpoint("generate", gen1)
        await c1.register()

        rt2 = DistributedRuntime(shared)
        c2 = rt2.namespace("ns").component("backend")

        async def gen2(payload, ctx):
            yield {"worker": 2}
        c2.serve_endpoint("generate", gen2)
        await c2.register()

        # kill worker 1's server but leave it in discovery (stale instance)
        await rt1.server.stop(drain=False)

        client_rt = DistributedRuntime(shared)
        client = client_rt.namespace("ns").component("backend").endpoint(
            "generate").client()
        results = []
        for _ in range(4):
            try:
                results.append((await client.call({}))["worker"])
            except EndpointError:
                pass  # first hit on the dead worker inhibits it
        assert results and all(r == 2 for r in results)
        await rt2.shutdown()
        await client_rt.shutdown()
    run(main())


def test_file_discovery_lease(tmp_path):
    d = FileDiscovery(str(tmp_path), ttl=0.2)
    inst = Instance("ns", "backend", "abc", "127.0.0.1:1", ["generate"])
    d.register(inst)
    assert len(d.list("ns", "backend")) == 1
    time.sleep(0.3)
    assert d.list("ns", "backend") == []  # lease expired
    d.refresh(inst)
    assert len(d.list("ns", "backend")) == 1
    d.deregister(inst)
    assert d.list("ns", "backend") == []


def test_request_plane_unix_socket(tmp_path):
    """UDS transport: the same codec/endpoints over a Unix domain socket."""
    import asyncio

    from dynamo_amd.runtime.request_plane import (RequestPlaneClient,
                                                  RequestPlaneServer)

    async def main():
        server = RequestPlaneServer(host=f"unix:{tmp_path}/rp.sock")

        async def echo(payload, ctx):
            for i in range(3):
                yield {"i": i, "got": payload}

        server.add_endpoint("echo", echo)
        addr = await server.start()
        assert addr.startswith("unix:")
        client = RequestPlaneClient()
        chunks = []
        async for ch in client.call_stream(addr, "echo", {"x": 42}):
            chunks.append(ch)
        assert [c["i"] for c in chunks] == [0, 1, 2]
        assert chunks[0]["got"] == {"x": 42}
        await client.close()
        await server.stop(drain=False)
    asyncio.new_event_loop().run_until_complete(main())


def test_codec_fuzz_roundtrip():
    """Codec survives arbitrary nested payloads + the UNCHECKED sentinel."""
    import random

    from dynamo_amd.runtime.codec import (UNCHECKED, decode_frame,
                                          decode_prefix, encode_frame)

    rng = random.Random(42)

    def rand_value(depth=0):
        k = rng.randrange(7 if depth < 3 else 4)
        if k == 0:
            return rng.randrange(-2**40, 2**40)
        if k == 1:
            return "".join(chr(rng.randrange(32, 0x2FF))
                           for _ in range(rng.randrange(0, 40)))
        if k == 2:
            return rng.random()
        if k == 3:
            return rng.choice([None, True, False])
        if k == 4:
            return bytes(rng.randrange(256) for _ in range(rng.randrange(64)))
        if k == 5:
            return [rand_value(depth + 1) for _ in range(rng.randrange(5))]
        return {f"k{i}": rand_value(depth + 1)
                for i in range(rng.randrange(5))}

    for _ in range(100):
        header = {"type": "rsp", "rid": rng.randrange(1 << 30),
                  "final": rng.random() < 0.5}
        body = rand_value()
        frame = encode_frame(header, body)
        hlen, blen, csum = decode_prefix(frame[:24])
        h2, b2 = decode_frame(frame[24:24 + hlen],
                              frame[24 + hlen:24 + hlen + blen], csum)
        assert h2 == header and b2 == body
        # thin-client sentinel skips verification
        h3, b3 = decode_frame(frame[24:24 + hlen],
                              frame[24 + hlen:24 + hlen + blen], UNCHECKED)
        assert b3 == body
        # corrupted checksum is rejected
        try:
            decode_frame(frame[24:24 + hlen],
                         frame[24 + hlen:24 + hlen + blen], csum ^ 1)
            assert False, "corruption not detected"
        except ValueError:
            pass


def test_file_discovery_ttl_expiry(tmp_path):
    """Instances whose lease file goes stale past the TTL disappear."""
    import os
    import time

    from dynamo_amd.runtime.discovery import FileDiscovery, Instance
    d = FileDiscovery(str(tmp_path), ttl=0.3)
    d.register(Instance(namespace="dynamo", component="backend",
                        instance_id="abc123", address="127.0.0.1:1"))
    assert [i.instance_id for i in d.list("dynamo")] == ["abc123"]
    # age the lease file beyond the TTL without touching it
    for root, _, files in os.walk(tmp_path):
        for f in files:
            p = os.path.join(root, f)
            old = time.time() - 5
            os.utime(p, (old, old))
    assert d.list("dynamo") == []


def test_otlp_exporter_posts_spans():
    """OTLP/HTTP export hook: spans emitted through RequestTracer reach a
    local collector endpoint as OTLP JSON (reference parity:
    observability-architecture.md OTEL_EXPORT_ENABLED + request_trace
    OTLP sink)."""
    import http.server
    import json as js
    import threading

    got = []

    class H(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            got.append((self.path, js.loads(self.rfile.read(n))))
            self.send_response(200)
            self.end_headers()
            self.wfile.write(b"{}")

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), H)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        from dynamo_amd.observability import OtlpExporter, RequestTracer
        from dynamo_amd.observability import set_tracer, span, trace_event
        exp = OtlpExporter(
            endpoint=f"http://127.0.0.1:{srv.server_port}",
            flush_interval=30.0)
        tracer = RequestTracer(path=None, otlp=exp)
        set_tracer(tracer)
        with span("http-request", request_id="req-42", model="m"):
            trace_event("handle_payload", worker_id="w1")
        set_tracer(None)
        exp.flush()
        assert exp.exported >= 2 and exp.errors == 0
        path, body = got[0]
        assert path == "/v1/traces"
        spans = body["resourceSpans"][0]["scopeSpans"][0]["spans"]
        names = {s["name"] for s in spans}
        assert "http-request" in names and "handle_payload" in names
        rs = body["resourceSpans"][0]["resource"]["attributes"]
        assert any(a["key"] == "service.name" for a in rs)
        # parent/child share the trace id
        tid = {s["traceId"] for s in spans}
        assert len(tid) == 1
        tracer.close()
    finally:
        srv.shutdown()
