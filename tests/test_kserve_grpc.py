"""KServe v2 gRPC frontend: ServerLive/ModelMetadata/ModelInfer over a real
grpc channel against mock workers (reference parity:
lib/llm/src/grpc/service/kserve.rs)."""
import asyncio

import pytest

from dynamo_amd.engine.config import ModelConfig
from dynamo_amd.frontend.kserve import MSG, SERVICE, make_grpc_server
from dynamo_amd.frontend.service import ModelManager
from dynamo_amd.mocker import make_mock_engine
from dynamo_amd.runtime import DistributedRuntime, MemoryDiscovery
from dynamo_amd.workers import WorkerService


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


MODEL = ModelConfig(name="mock-model", vocab_size=512)


@pytest.mark.timeout(120)
def test_kserve_grpc_infer():
    import grpc

    async def main():
        shared = MemoryDiscovery()
        rt = DistributedRuntime(shared)
        eng = make_mock_engine(model=MODEL)
        ws = WorkerService(eng, rt)
        await ws.start()
        mgr_rt = DistributedRuntime(shared)
        mgr = ModelManager(mgr_rt)
        await mgr.start(watch_interval=0.2)
        server, port = make_grpc_server(mgr)
        await server.start()
        try:
            async with grpc.aio.insecure_channel(f"127.0.0.1:{port}") as ch:
                def rpc(method, req_cls, resp_cls):
                    return ch.unary_unary(
                        f"/{SERVICE}/{method}",
                        request_serializer=lambda m: m.SerializeToString(),
                        response_deserializer=resp_cls.FromString)

                live = await rpc("ServerLive", MSG["ServerLiveRequest"],
                                 MSG["ServerLiveResponse"])(
                    MSG["ServerLiveRequest"]())
                assert live.live
                md = await rpc("ModelMetadata", MSG["ModelMetadataRequest"],
                               MSG["ModelMetadataResponse"])(
                    MSG["ModelMetadataRequest"](name="mock-model"))
                assert md.platform == "dynamo_amd"
                assert md.inputs[0].name == "text_input"

                req = MSG["ModelInferRequest"](model_name="mock-model",
                                               id="rq1")
                t = req.inputs.add()
                t.name, t.datatype = "text_input", "BYTES"
                t.shape.append(1)
                t.contents.bytes_contents.append(b"hello kserve")
                mt = req.inputs.add()
                mt.name, mt.datatype = "max_tokens", "INT32"
                mt.shape.append(1)
                mt.contents.int_contents.append(6)
                resp = await rpc("ModelInfer", MSG["ModelInferRequest"],
                                 MSG["ModelInferResponse"])(req)
                outs = {o.name: o for o in resp.outputs}
                assert resp.id == "rq1"
                assert len(outs["token_ids"].contents.int_contents) == 6
                assert outs["text_output"].contents.bytes_contents[0]
                # unknown model -> NOT_FOUND... only if >1 model registered;
                # with a single model the manager falls back to it
                resp2 = await rpc("ModelInfer", MSG["ModelInferRequest"],
                                  MSG["ModelInferResponse"])(
                    MSG["ModelInferRequest"](model_name=""))
                # missing inputs -> INVALID_ARGUMENT
                assert False, "expected INVALID_ARGUMENT"
        except grpc.aio.AioRpcError as e:
            assert e.code() == grpc.StatusCode.INVALID_ARGUMENT
        finally:
            await server.stop(grace=0.2)
            await mgr.stop()
            await ws.stop()
            await rt.shutdown(drain=False)
            await mgr_rt.shutdown(drain=False)
    run(main())
