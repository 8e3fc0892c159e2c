"""gRPC transport with msgpack (default) or protobuf message bodies.

The reference compiles .proto files with protoc (scheduler/runtime/protobuf,
Makefile rpc_stubs); this environment has grpcio but no protoc, so the same
three services are exposed through gRPC's generic-handler API.  Two body
codecs:

* ``msgpack`` (default) — dict messages keyed by the proto field names.
* ``proto``  (``SWQ_RPC_CODEC=proto``) — true protobuf wire format from
  runtime-built stubs (rpc/pb.py parses the committed .proto files and
  materializes message classes through the google.protobuf runtime;
  tests/test_pb.py proves byte-level interop against classes built from
  the REFERENCE's own .proto files).

Service and method names, field names and field semantics mirror the
reference protos (runtime/protobuf/{worker_to_scheduler,
scheduler_to_worker,iterator_to_scheduler}.proto) — that schema is the
compatibility surface in both codecs.
"""

from __future__ import annotations

import logging
from concurrent import futures
from typing import Callable, Dict

import os

import grpc
import msgpack

logger = logging.getLogger("shockwave_amd.rpc")


def _pack(obj) -> bytes:
    return msgpack.packb(obj, use_bin_type=True)


def _unpack(data: bytes):
    return msgpack.unpackb(data, raw=False)


def _codec() -> str:
    return os.environ.get("SWQ_RPC_CODEC", "msgpack")


def _proto_codecs(service: str, method: str):
    """(request_deserializer, response_serializer) server-side and
    (request_serializer, response_deserializer) client-side for the
    protobuf wire format, from the runtime-built stubs."""
    from .pb import our_schema

    schema = our_schema()
    req_name, resp_name = schema.method_types(service, method)

    def req_deser(data: bytes):
        m = schema.messages[req_name]()
        m.ParseFromString(data)
        return schema.to_dict(m)

    def resp_ser(payload) -> bytes:
        return schema.to_message(resp_name, payload).SerializeToString()

    def req_ser(payload) -> bytes:
        return schema.to_message(req_name, payload).SerializeToString()

    def resp_deser(data: bytes):
        m = schema.messages[resp_name]()
        m.ParseFromString(data)
        return schema.to_dict(m)

    return req_deser, resp_ser, req_ser, resp_deser


def make_server(
    port: int,
    services: Dict[str, Dict[str, Callable]],
    max_workers: int = 16,
) -> grpc.Server:
    """services: {service_name: {method_name: fn(request_dict) -> dict}}"""
    server = grpc.server(
        futures.ThreadPoolExecutor(max_workers=max_workers),
        options=[
            ("grpc.max_send_message_length", 64 * 1024 * 1024),
            ("grpc.max_receive_message_length", 64 * 1024 * 1024),
        ],
    )
    for service_name, methods in services.items():
        handlers = {}
        for method_name, fn in methods.items():
            def _wrap(f):
                def handler(request, context):
                    try:
                        return f(request) or {}
                    except Exception:
                        logger.exception("RPC handler failed")
                        raise

                return handler

            if _codec() == "proto":
                req_deser, resp_ser, _, _ = _proto_codecs(
                    service_name, method_name
                )
            else:
                req_deser, resp_ser = _unpack, _pack
            handlers[method_name] = grpc.unary_unary_rpc_method_handler(
                _wrap(fn),
                request_deserializer=req_deser,
                response_serializer=resp_ser,
            )
        server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler(service_name, handlers),)
        )
    server.add_insecure_port(f"[::]:{port}")
    server.start()
    return server


class RpcClient:
    """One insecure channel per client (the reference opens one per call;
    a persistent channel is cheaper and semantically identical)."""

    def __init__(self, addr: str, port: int):
        self._target = f"{addr}:{port}"
        self._channel = grpc.insecure_channel(self._target)

    def call(self, service: str, method: str, payload: dict, timeout=30):
        if _codec() == "proto":
            _, _, req_ser, resp_deser = _proto_codecs(service, method)
        else:
            req_ser, resp_deser = _pack, _unpack
        fn = self._channel.unary_unary(
            f"/{service}/{method}",
            request_serializer=req_ser,
            response_deserializer=resp_deser,
        )
        return fn(payload or {}, timeout=timeout)

    def close(self):
        self._channel.close()
