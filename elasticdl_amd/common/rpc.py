"""Minimal gRPC service layer over the binary tensor codec.

The reference defines its control and PS protocols in protobuf
(elasticai_api/proto/elasticai_api.proto, elasticdl/proto/elasticdl.proto).
This rebuild keeps gRPC as the transport — HTTP/2 framing, multiplexing,
and deadline semantics are exactly what an elastic control plane needs —
but replaces protobuf message types with the zero-copy codec
(elasticdl_amd.common.codec): every method is ``bytes -> bytes`` at the
gRPC layer and ``dict -> dict`` (with embedded torch tensors) at the
application layer.

Usage::

    servicer = {"get_task": fn, ...}
    server = start_server("0.0.0.0:0", {"Master": servicer})
    client = RpcClient("127.0.0.1:12345")
    resp = client.call("Master", "get_task", {"worker_id": 0})
"""

import threading
from concurrent import futures
from typing import Any, Callable, Dict, Optional

import grpc

from elasticdl_amd.common import codec
from elasticdl_amd.common.constants import GRPC_CHANNEL_OPTIONS
from elasticdl_amd.common.log_utils import default_logger as logger

_identity = lambda b: b  # noqa: E731  raw bytes pass through grpc


class _GenericService(grpc.GenericRpcHandler):
    def __init__(self, service_name: str, methods: Dict[str, Callable]):
        self._prefix = f"/{service_name}/"
        self._handlers = {}
        for method_name, fn in methods.items():
            self._handlers[self._prefix + method_name] = (
                grpc.unary_unary_rpc_method_handler(
                    self._wrap(fn),
                    request_deserializer=_identity,
                    response_serializer=_identity,
                )
            )

    @staticmethod
    def _wrap(fn: Callable) -> Callable:
        def handler(request_bytes: bytes, context: grpc.ServicerContext) -> bytes:
            try:
                request = codec.decode(request_bytes)
                response = fn(request)
                return codec.encode(response if response is not None else {})
            except Exception:
                logger.exception("RPC handler %s failed", fn.__name__)
                context.abort(grpc.StatusCode.INTERNAL, "handler error")

        return handler

    def service(self, handler_call_details):
        return self._handlers.get(handler_call_details.method)


class RpcServer:
    def __init__(self, server: grpc.Server, port: int):
        self.server = server
        self.port = port

    def stop(self, grace: Optional[float] = None) -> None:
        self.server.stop(grace)


def start_server(
    bind_address: str,
    services: Dict[str, Dict[str, Callable]],
    max_workers: int = 64,
) -> RpcServer:
    """Start a gRPC server. ``bind_address`` like "0.0.0.0:0" (0 = pick a
    free port; the chosen port is returned on the RpcServer)."""
    server = grpc.server(
        futures.ThreadPoolExecutor(max_workers=max_workers),
        options=GRPC_CHANNEL_OPTIONS,
    )
    handlers = [_GenericService(name, methods) for name, methods in services.items()]
    server.add_generic_rpc_handlers(tuple(handlers))
    port = server.add_insecure_port(bind_address)
    server.start()
    return RpcServer(server, port)


class RpcClient:
    """Thread-safe client over one channel; lazily creates per-method stubs."""

    def __init__(self, address: str):
        self.address = address
        self._channel = grpc.insecure_channel(address, options=GRPC_CHANNEL_OPTIONS)
        self._stubs: Dict[str, Callable] = {}
        self._lock = threading.Lock()

    def _stub(self, service: str, method: str) -> Callable:
        key = f"/{service}/{method}"
        stub = self._stubs.get(key)
        if stub is None:
            with self._lock:
                stub = self._stubs.get(key)
                if stub is None:
                    stub = self._channel.unary_unary(
                        key,
                        request_serializer=_identity,
                        response_deserializer=_identity,
                    )
                    self._stubs[key] = stub
        return stub

    def call(
        self,
        service: str,
        method: str,
        message: Any = None,
        timeout: Optional[float] = None,
    ) -> Any:
        data = codec.encode(message if message is not None else {})
        return codec.decode(self._stub(service, method)(data, timeout=timeout))

    def call_future(self, service: str, method: str, message: Any = None,
                    timeout: Optional[float] = 300.0):
        """Async fan-out variant (reference: ps_client futures); returns a
        future whose .result() must be passed through codec.decode by
        ``resolve``. A default deadline keeps workers from hanging forever
        on a dead PS (the master's job-failure path then kicks in)."""
        data = codec.encode(message if message is not None else {})
        return self._stub(service, method).future(data, timeout=timeout)

    @staticmethod
    def resolve(future) -> Any:
        return codec.decode(future.result())

    def wait_ready(self, timeout: float = 30.0) -> None:
        grpc.channel_ready_future(self._channel).result(timeout=timeout)

    def close(self) -> None:
        self._channel.close()
