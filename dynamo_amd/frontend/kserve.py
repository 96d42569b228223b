"""KServe v2 gRPC inference frontend.

Reference parity: the reference's KServe gRPC service
(ai-dynamo/dynamo lib/llm/src/grpc/service/kserve.rs) serving LLMs over
the open-inference-protocol: GRPCInferenceService with
ServerLive/ServerReady/ModelReady/ModelMetadata/ModelInfer.

There is no protoc in this environment, so the protocol messages are
built AT RUNTIME from a hand-written FileDescriptorProto (same wire
format as kserve's grpc_predict_v2.proto; `parameters` maps are omitted —
unknown fields are skipped by protobuf parsing, so clients that set them
still interoperate).

LLM mapping (text in/out, matching the reference's ModelInput::Text):
  inputs:  "text_input" (BYTES, 1 string)  — the prompt
           "max_tokens" (INT32, optional), "temperature" (FP32, optional),
           "top_p" (FP32), "top_k" (INT32), "seed" (INT64)
  outputs: "text_output" (BYTES, 1 string), "token_ids" (INT32, [n])
"""
from __future__ import annotations

import logging
from typing import Optional

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

log = logging.getLogger("dynamo_amd.kserve")

_PKG = "inference"


def _build_pool():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "dynamo_amd_kserve.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"

    def msg(name):
        m = fdp.message_type.add()
        m.name = name
        return m

    def field(m, name, number, ftype, label=1, type_name=None):
        f = m.field.add()
        f.name = name
        f.number = number
        f.type = ftype
        f.label = label  # 1=optional, 3=repeated
        if type_name:
            f.type_name = type_name
        return f

    T = descriptor_pb2.FieldDescriptorProto
    msg("ServerLiveRequest")
    m = msg("ServerLiveResponse")
    field(m, "live", 1, T.TYPE_BOOL)
    msg("ServerReadyRequest")
    m = msg("ServerReadyResponse")
    field(m, "ready", 1, T.TYPE_BOOL)
    m = msg("ModelReadyRequest")
    field(m, "name", 1, T.TYPE_STRING)
    field(m, "version", 2, T.TYPE_STRING)
    m = msg("ModelReadyResponse")
    field(m, "ready", 1, T.TYPE_BOOL)
    m = msg("ModelMetadataRequest")
    field(m, "name", 1, T.TYPE_STRING)
    field(m, "version", 2, T.TYPE_STRING)
    m = msg("TensorMetadata")
    field(m, "name", 1, T.TYPE_STRING)
    field(m, "datatype", 2, T.TYPE_STRING)
    field(m, "shape", 3, T.TYPE_INT64, label=3)
    m = msg("ModelMetadataResponse")
    field(m, "name", 1, T.TYPE_STRING)
    field(m, "versions", 2, T.TYPE_STRING, label=3)
    field(m, "platform", 3, T.TYPE_STRING)
    field(m, "inputs", 4, T.TYPE_MESSAGE, label=3,
          type_name=f".{_PKG}.TensorMetadata")
    field(m, "outputs", 5, T.TYPE_MESSAGE, label=3,
          type_name=f".{_PKG}.TensorMetadata")
    m = msg("InferTensorContents")
    field(m, "bool_contents", 1, T.TYPE_BOOL, label=3)
    field(m, "int_contents", 2, T.TYPE_INT32, label=3)
    field(m, "int64_contents", 3, T.TYPE_INT64, label=3)
    field(m, "uint_contents", 4, T.TYPE_UINT32, label=3)
    field(m, "uint64_contents", 5, T.TYPE_UINT64, label=3)
    field(m, "fp32_contents", 6, T.TYPE_FLOAT, label=3)
    field(m, "fp64_contents", 7, T.TYPE_DOUBLE, label=3)
    field(m, "bytes_contents", 8, T.TYPE_BYTES, label=3)
    m = msg("InferInputTensor")
    field(m, "name", 1, T.TYPE_STRING)
    field(m, "datatype", 2, T.TYPE_STRING)
    field(m, "shape", 3, T.TYPE_INT64, label=3)
    field(m, "contents", 5, T.TYPE_MESSAGE,
          type_name=f".{_PKG}.InferTensorContents")
    m = msg("InferRequestedOutputTensor")
    field(m, "name", 1, T.TYPE_STRING)
    m = msg("ModelInferRequest")
    field(m, "model_name", 1, T.TYPE_STRING)
    field(m, "model_version", 2, T.TYPE_STRING)
    field(m, "id", 3, T.TYPE_STRING)
    field(m, "inputs", 5, T.TYPE_MESSAGE, label=3,
          type_name=f".{_PKG}.InferInputTensor")
    field(m, "outputs", 6, T.TYPE_MESSAGE, label=3,
          type_name=f".{_PKG}.InferRequestedOutputTensor")
    field(m, "raw_input_contents", 7, T.TYPE_BYTES, label=3)
    m = msg("InferOutputTensor")
    field(m, "name", 1, T.TYPE_STRING)
    field(m, "datatype", 2, T.TYPE_STRING)
    field(m, "shape", 3, T.TYPE_INT64, label=3)
    field(m, "contents", 5, T.TYPE_MESSAGE,
          type_name=f".{_PKG}.InferTensorContents")
    m = msg("ModelInferResponse")
    field(m, "model_name", 1, T.TYPE_STRING)
    field(m, "model_version", 2, T.TYPE_STRING)
    field(m, "id", 3, T.TYPE_STRING)
    field(m, "outputs", 5, T.TYPE_MESSAGE, label=3,
          type_name=f".{_PKG}.InferOutputTensor")
    field(m, "raw_output_contents", 6, T.TYPE_BYTES, label=3)

    pool = descriptor_pool.DescriptorPool()
    pool.Add(fdp)
    return pool


_pool = _build_pool()


def _cls(name):
    return message_factory.GetMessageClass(
        _pool.FindMessageTypeByName(f"{_PKG}.{name}"))


MSG = {n: _cls(n) for n in (
    "ServerLiveRequest", "ServerLiveResponse", "ServerReadyRequest",
    "ServerReadyResponse", "ModelReadyRequest", "ModelReadyResponse",
    "ModelMetadataRequest", "ModelMetadataResponse", "ModelInferRequest",
    "ModelInferResponse", "InferTensorContents")}

SERVICE = "inference.GRPCInferenceService"


def _scalar(tensor, default=None):
    c = tensor.contents
    for fieldname in ("int_contents", "int64_contents", "fp32_contents",
                      "fp64_contents", "uint_contents", "bool_contents"):
        vals = list(getattr(c, fieldname))
        if vals:
            return vals[0]
    return default


class KServeService:
    """grpc.aio service over a ModelManager (shares routing/migration with
    the HTTP frontend)."""

    def __init__(self, manager):
        self.manager = manager

    async def server_live(self, request, context):
        return MSG["ServerLiveResponse"](live=True)

    async def server_ready(self, request, context):
        return MSG["ServerReadyResponse"](ready=True)

    async def model_ready(self, request, context):
        ok = request.name in self.manager.models or (
            not request.name and bool(self.manager.models))
        return MSG["ModelReadyResponse"](ready=ok)

    async def model_metadata(self, request, context):
        resp = MSG["ModelMetadataResponse"](
            name=request.name or next(iter(self.manager.models), ""),
            platform="dynamo_amd")
        resp.versions.append("1")
        ti = resp.inputs.add()
        ti.name, ti.datatype = "text_input", "BYTES"
        ti.shape.append(1)
        to = resp.outputs.add()
        to.name, to.datatype = "text_output", "BYTES"
        to.shape.append(1)
        return resp

    async def model_infer(self, request, context):
        import grpc
        try:
            entry = self.manager.get(request.model_name)
        except KeyError as e:
            await context.abort(grpc.StatusCode.NOT_FOUND, str(e))
        tensors = {t.name: t for t in request.inputs}
        text = None
        token_ids = None
        if "text_input" in tensors:
            bc = list(tensors["text_input"].contents.bytes_contents)
            if bc:
                text = bc[0].decode("utf-8", "replace")
        elif request.raw_input_contents:
            raw = request.raw_input_contents[0]
            # KServe raw BYTES framing: u32 length prefix + payload
            if len(raw) >= 4:
                n = int.from_bytes(raw[:4], "little")
                text = raw[4:4 + n].decode("utf-8", "replace")
        if "input_ids" in tensors:
            token_ids = [int(v) for v in
                         tensors["input_ids"].contents.int_contents] or None
        if token_ids is None:
            if text is None:
                await context.abort(grpc.StatusCode.INVALID_ARGUMENT,
                                    "need text_input or input_ids")
            token_ids = entry.tokenizer.encode(text)
        sampling = {
            "temperature": float(_scalar(tensors.get("temperature"), 0.0)
                                 if "temperature" in tensors else 0.0),
            "top_p": float(_scalar(tensors.get("top_p"), 1.0)
                           if "top_p" in tensors else 1.0),
            "top_k": int(_scalar(tensors.get("top_k"), 0)
                         if "top_k" in tensors else 0),
            "seed": int(_scalar(tensors.get("seed"), 0)
                        if "seed" in tensors else 0),
        }
        max_tokens = int(_scalar(tensors.get("max_tokens"), 128)
                         if "max_tokens" in tensors else 128)
        eos = getattr(entry.tokenizer, "eos_id", None)
        stop = {"max_tokens": max_tokens,
                "stop_token_ids": [eos] if eos is not None else []}
        produced = []
        async for chunk in self.manager.generate_tokens(
                entry, token_ids, sampling, stop, request_id=request.id or None):
            produced.extend(chunk.get("token_ids", []))
        resp = MSG["ModelInferResponse"](model_name=entry.name,
                                         model_version="1", id=request.id)
        out = resp.outputs.add()
        out.name, out.datatype = "text_output", "BYTES"
        out.shape.append(1)
        out.contents.bytes_contents.append(
            entry.tokenizer.decode(produced).encode())
        tok = resp.outputs.add()
        tok.name, tok.datatype = "token_ids", "INT32"
        tok.shape.append(len(produced))
        tok.contents.int_contents.extend(produced)
        return resp


def make_grpc_server(manager, host: str = "127.0.0.1", port: int = 0):
    """Build a grpc.aio server (returns (server, bound_port))."""
    import grpc

    svc = KServeService(manager)

    def unary(fn, req_cls):
        return grpc.unary_unary_rpc_method_handler(
            fn, request_deserializer=req_cls.FromString,
            response_serializer=lambda m: m.SerializeToString())

    handlers = {
        "ServerLive": unary(svc.server_live, MSG["ServerLiveRequest"]),
        "ServerReady": unary(svc.server_ready, MSG["ServerReadyRequest"]),
        "ModelReady": unary(svc.model_ready, MSG["ModelReadyRequest"]),
        "ModelMetadata": unary(svc.model_metadata,
                               MSG["ModelMetadataRequest"]),
        "ModelInfer": unary(svc.model_infer, MSG["ModelInferRequest"]),
    }
    server = grpc.aio.server()
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(SERVICE, handlers),))
    bound = server.add_insecure_port(f"{host}:{port}")
    return server, bound
