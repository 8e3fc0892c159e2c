"""gRPC transport with msgpack message bodies.

The reference compiles .proto files with protoc (scheduler/runtime/protobuf,
Makefile rpc_stubs); this environment has grpcio but no protoc, so the same
three services are exposed through gRPC's generic-handler API with
msgpack-serialized dict messages.  Service and method names, field names
and field semantics mirror the reference protos
(runtime/protobuf/{worker_to_scheduler,scheduler_to_worker,
iterator_to_scheduler}.proto) — that schema is the compatibility surface.
"""

from __future__ import annotations

import logging
from concurrent import futures
from typing import Callable, Dict

import grpc
import msgpack

logger = logging.getLogger("shockwave_amd.rpc")


def _pack(obj) -> bytes:
    return msgpack.packb(obj, use_bin_type=True)


def _unpack(data: bytes):
    return msgpack.unpackb(data, raw=False)


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

            handlers[method_name] = grpc.unary_unary_rpc_method_handler(
                _wrap(fn),
                request_deserializer=_unpack,
                response_serializer=_pack,
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
        fn = self._channel.unary_unary(
            f"/{service}/{method}",
            request_serializer=_pack,
            response_deserializer=_unpack,
        )
        return fn(payload or {}, timeout=timeout)

    def close(self):
        self._channel.close()
