"""gRPC transport with msgpack message encoding.

The control plane speaks gRPC end-to-end (SDK <-> gateway <-> worker), same
macro-protocol as the reference (reference: proto/*.proto, pkg/gateway/
gateway.go:291-319 grpc server setup). This framework registers services as
generic gRPC handlers with msgpack-encoded messages instead of compiled
protobufs: the image has no protoc/grpcio-tools, and msgpack keeps the wire
contract introspectable from the typed pydantic models in beta9_amd/types.py.

Service surface: register async handlers by (service, method); unary and
server/client/bidi streaming are all supported. Auth is a bearer token in
metadata, validated by a server-side callback.
"""

from __future__ import annotations

import asyncio
import inspect
from typing import Any, AsyncIterator, Awaitable, Callable, Dict, Optional, Tuple

import grpc
import msgpack


def pack(obj: Any) -> bytes:
    return msgpack.packb(obj, use_bin_type=True, default=_default)


def _default(obj: Any) -> Any:
    # pydantic models and anything with model_dump() serialize transparently
    if hasattr(obj, "model_dump"):
        return obj.model_dump()
    if isinstance(obj, (set, tuple)):
        return list(obj)
    raise TypeError(f"unpackable type {type(obj)!r}")


def unpack(data: bytes) -> Any:
    return msgpack.unpackb(data, raw=False, strict_map_key=False)


def _identity(data: bytes) -> bytes:
    return data


class RpcError(Exception):
    def __init__(self, code: grpc.StatusCode, message: str):
        super().__init__(message)
        self.code = code
        self.message = message


def not_found(msg: str) -> RpcError:
    return RpcError(grpc.StatusCode.NOT_FOUND, msg)


def invalid(msg: str) -> RpcError:
    return RpcError(grpc.StatusCode.INVALID_ARGUMENT, msg)


def unauthenticated(msg: str = "invalid token") -> RpcError:
    return RpcError(grpc.StatusCode.UNAUTHENTICATED, msg)


def permission_denied(msg: str = "permission denied") -> RpcError:
    return RpcError(grpc.StatusCode.PERMISSION_DENIED, msg)


Handler = Callable[..., Any]
AuthFn = Callable[[str, str], Awaitable[Optional[Dict[str, Any]]]]
# AuthFn(token, full_method) -> auth context dict or None (reject)


class _GenericService(grpc.GenericRpcHandler):
    def __init__(self, registry: Dict[str, Tuple[str, Handler]], auth_fn: Optional[AuthFn], open_methods: set):
        self._registry = registry
        self._auth_fn = auth_fn
        self._open = open_methods

    def service(self, handler_call_details: grpc.HandlerCallDetails):
        full = handler_call_details.method  # "/Service/Method"
        entry = self._registry.get(full)
        if entry is None:
            return None
        kind, fn = entry
        wrapped = self._wrap(full, kind, fn)
        # responses are packed INSIDE the wrapper (identity serializer):
        # grpc.aio str-encodes bare-str responses before a serializer runs,
        # which would corrupt msgpack str payloads into bins
        ident = _identity
        if kind == "unary_unary":
            return grpc.unary_unary_rpc_method_handler(wrapped, unpack, ident)
        if kind == "unary_stream":
            return grpc.unary_stream_rpc_method_handler(wrapped, unpack, ident)
        if kind == "stream_unary":
            return grpc.stream_unary_rpc_method_handler(wrapped, unpack, ident)
        if kind == "stream_stream":
            return grpc.stream_stream_rpc_method_handler(wrapped, unpack, ident)
        return None

    def _wrap(self, full: str, kind: str, fn: Handler):
        auth_fn = self._auth_fn
        needs_auth = auth_fn is not None and full not in self._open

        async def _authctx(context) -> Dict[str, Any]:
            if not needs_auth:
                return {}
            md = dict(context.invocation_metadata() or ())
            token = md.get("authorization", "")
            if token.startswith("Bearer "):
                token = token[7:]
            ctx = await auth_fn(token, full)
            if ctx is None:
                await context.abort(grpc.StatusCode.UNAUTHENTICATED, "invalid token")
            return ctx

        if kind in ("unary_unary", "stream_unary"):

            async def handler(request, context):
                auth = await _authctx(context)
                try:
                    return pack(await fn(request, context, auth))
                except RpcError as e:
                    await context.abort(e.code, e.message)

            return handler

        async def stream_handler(request, context):
            auth = await _authctx(context)
            try:
                async for item in fn(request, context, auth):
                    yield pack(item)
            except RpcError as e:
                await context.abort(e.code, e.message)

        return stream_handler


class RpcServer:
    """Async gRPC server hosting msgpack generic services."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0, auth_fn: Optional[AuthFn] = None):
        self._host = host
        self.port = port
        self._auth_fn = auth_fn
        self._registry: Dict[str, Tuple[str, Handler]] = {}
        self._open_methods: set = set()
        self._server: Optional[grpc.aio.Server] = None

    def register(self, service: str, method: str, fn: Handler, kind: str = "unary_unary", open: bool = False) -> None:
        full = f"/{service}/{method}"
        self._registry[full] = (kind, fn)
        if open:
            self._open_methods.add(full)

    def register_service(self, service_obj: Any) -> None:
        """Register every rpc-decorated method of a service object.

        Methods opt in with the @rpc(...) decorator below; the gRPC service
        name is the class attribute `service_name`.
        """
        service = getattr(service_obj, "service_name", type(service_obj).__name__)
        for name in dir(service_obj):
            fn = getattr(service_obj, name)
            meta = getattr(fn, "_rpc_meta", None)
            if meta is not None:
                self.register(service, meta["name"] or name, fn, kind=meta["kind"], open=meta["open"])

    async def start(self) -> int:
        self._server = grpc.aio.server(
            options=[
                ("grpc.max_send_message_length", 256 * 1024 * 1024),
                ("grpc.max_receive_message_length", 256 * 1024 * 1024),
                ("grpc.keepalive_time_ms", 30000),
                ("grpc.keepalive_timeout_ms", 10000),
            ]
        )
        self._server.add_generic_rpc_handlers(
            (_GenericService(self._registry, self._auth_fn, self._open_methods),)
        )
        self.port = self._server.add_insecure_port(f"{self._host}:{self.port}")
        await self._server.start()
        return self.port

    async def stop(self, grace: float = 1.0) -> None:
        if self._server is not None:
            await self._server.stop(grace)
            self._server = None


def rpc(kind: str = "unary_unary", name: Optional[str] = None, open: bool = False):
    """Mark a service method as an RPC handler.

    Handler signature: async fn(request, context, auth) — for stream-input
    kinds, `request` is an async iterator of messages.
    """

    def deco(fn):
        fn._rpc_meta = {"kind": kind, "name": name, "open": open}
        return fn

    return deco


class RpcClient:
    """Client channel for msgpack generic services."""

    def __init__(self, target: str, token: str = "", private_pool: bool = False):
        self.target = target
        self._token = token
        self._private_pool = private_pool
        self._channel: Optional[grpc.aio.Channel] = None

    def _md(self):
        if self._token:
            return (("authorization", f"Bearer {self._token}"),)
        return ()

    async def connect(self) -> None:
        if self._channel is None:
            self._channel = grpc.aio.insecure_channel(
                self.target,
                options=[
                    ("grpc.max_send_message_length", 256 * 1024 * 1024),
                    ("grpc.max_receive_message_length", 256 * 1024 * 1024),
                    # cold-start path: a probe may race the runner's bind;
                    # the default ~1 s reconnect backoff would become the
                    # container's observed cold start
                    ("grpc.initial_reconnect_backoff_ms", 20),
                    ("grpc.min_reconnect_backoff_ms", 20),
                    ("grpc.max_reconnect_backoff_ms", 500),
                ]
                + (
                    # backoff state lives on the SUBCHANNEL, which the GLOBAL
                    # pool shares across channels to the same target:
                    # recreating a channel would still reuse the backed-off
                    # subchannel (measured: ~90 ms of refused probes after the
                    # runner was live, benchmarks/cold_trace.py). Probe
                    # channels opt into a private pool so a fresh channel
                    # connects immediately; everything else shares subchannels
                    # (100-way create bursts exhausted accept queues with
                    # all-private pools).
                    [("grpc.use_local_subchannel_pool", 1)]
                    if self._private_pool
                    else []
                ),
            )

    async def close(self) -> None:
        if self._channel is not None:
            await self._channel.close()
            self._channel = None

    async def call(self, service: str, method: str, payload: Any = None, timeout: Optional[float] = None) -> Any:
        await self.connect()
        fn = self._channel.unary_unary(f"/{service}/{method}", request_serializer=pack, response_deserializer=unpack)
        return await fn(payload, metadata=self._md(), timeout=timeout)

    async def stream(self, service: str, method: str, payload: Any = None) -> AsyncIterator[Any]:
        await self.connect()
        fn = self._channel.unary_stream(f"/{service}/{method}", request_serializer=pack, response_deserializer=unpack)
        async for item in fn(payload, metadata=self._md()):
            yield item

    async def client_stream(self, service: str, method: str, payloads: AsyncIterator[Any], timeout: Optional[float] = None) -> Any:
        await self.connect()
        fn = self._channel.stream_unary(f"/{service}/{method}", request_serializer=pack, response_deserializer=unpack)
        return await fn(payloads, metadata=self._md(), timeout=timeout)

    async def __aenter__(self) -> "RpcClient":
        await self.connect()
        return self

    async def __aexit__(self, *exc: Any) -> None:
        await self.close()
