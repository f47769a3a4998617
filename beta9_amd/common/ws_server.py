"""First-party WebSocket protocol for uvicorn (server-side RFC 6455).

The target image ships neither `websockets` nor `wsproto`, so uvicorn
cannot serve WebSocket routes for @realtime/@asgi stubs out of the box.
This module implements uvicorn's ws-protocol contract directly (the class
is handed the transport + the raw upgrade request by uvicorn's HTTP
protocol — see handle_websocket_upgrade in uvicorn/protocols/http) and
speaks the ASGI websocket interface to the user's app: connect/accept,
text/bytes send+receive, close, ping/pong. No extensions, no compression —
the subset real apps use (reference parity target:
proxyWebsocketConnection, pkg/abstractions/endpoint/buffer.go:958).
"""

from __future__ import annotations

import asyncio
import base64
import hashlib
import logging
import struct
from typing import Any, Dict, List, Optional, Tuple
from urllib.parse import unquote

log = logging.getLogger("b9.ws")

_GUID = b"258EAFA5-E914-47DA-95CA-C5AB0DC85B11"


class _FrameParser:
    """Incremental client-frame parser (client frames are masked)."""

    def __init__(self) -> None:
        self.buf = bytearray()

    def feed(self, data: bytes) -> List[Tuple[int, bytes]]:
        self.buf.extend(data)
        out: List[Tuple[int, bytes]] = []
        while True:
            frame = self._try_parse()
            if frame is None:
                return out
            out.append(frame)

    def _try_parse(self) -> Optional[Tuple[int, bytes]]:
        buf = self.buf
        if len(buf) < 2:
            return None
        op = buf[0] & 0x0F
        masked = bool(buf[1] & 0x80)
        n = buf[1] & 0x7F
        off = 2
        if n == 126:
            if len(buf) < 4:
                return None
            n = struct.unpack_from(">H", buf, 2)[0]
            off = 4
        elif n == 127:
            if len(buf) < 10:
                return None
            n = struct.unpack_from(">Q", buf, 2)[0]
            off = 10
        mask = b""
        if masked:
            if len(buf) < off + 4:
                return None
            mask = bytes(buf[off: off + 4])
            off += 4
        if len(buf) < off + n:
            return None
        payload = bytes(buf[off: off + n])
        if masked:
            payload = bytes(b ^ mask[i % 4] for i, b in enumerate(payload))
        del buf[: off + n]
        return op, payload


def _encode_frame(op: int, payload: bytes) -> bytes:
    header = bytes([0x80 | op])
    n = len(payload)
    if n < 126:
        header += bytes([n])
    elif n < (1 << 16):
        header += bytes([126]) + struct.pack(">H", n)
    else:
        header += bytes([127]) + struct.pack(">Q", n)
    return header + payload


class B9WebSocketProtocol(asyncio.Protocol):
    """uvicorn ws_protocol_class contract: constructed on upgrade, fed the
    raw request bytes, owns the transport from then on."""

    def __init__(self, config, server_state, app_state=None, _loop=None):
        self.config = config
        self.app = config.loaded_app
        self.app_state = app_state or {}
        self.server_state = server_state
        self.transport: Optional[asyncio.Transport] = None
        self.parser = _FrameParser()
        self.handshaken = False
        self.accepted = False
        self.closed = False
        self.recv_q: "asyncio.Queue[Dict[str, Any]]" = asyncio.Queue()
        self._app_task: Optional[asyncio.Task] = None
        self._hs_buf = bytearray()

    # -- asyncio.Protocol ---------------------------------------------------
    def connection_made(self, transport) -> None:
        self.transport = transport

    def connection_lost(self, exc) -> None:
        self.closed = True
        self.recv_q.put_nowait({"type": "websocket.disconnect", "code": 1006})

    def data_received(self, data: bytes) -> None:
        if not self.handshaken:
            self._hs_buf.extend(data)
            end = self._hs_buf.find(b"\r\n\r\n")
            if end < 0:
                return
            head = bytes(self._hs_buf[:end])
            rest = bytes(self._hs_buf[end + 4:])
            self._handshake(head)
            if rest:
                self._feed(rest)
            return
        self._feed(data)

    # -- handshake ----------------------------------------------------------
    def _handshake(self, head: bytes) -> None:
        lines = head.split(b"\r\n")
        method, _, target = lines[0].partition(b" ")
        target = target.rsplit(b" ", 1)[0]
        headers: List[Tuple[bytes, bytes]] = []
        key = b""
        subprotos: List[str] = []
        for line in lines[1:]:
            name, _, value = line.partition(b":")
            name = name.strip().lower()
            value = value.strip()
            headers.append((name, value))
            if name == b"sec-websocket-key":
                key = value
            elif name == b"sec-websocket-protocol":
                subprotos += [p.strip() for p in value.decode().split(",")]
        if method.upper() != b"GET" or not key:
            self._reject(400, "bad upgrade request")
            return
        self.handshaken = True
        self._accept_value = base64.b64encode(
            hashlib.sha1(key + _GUID).digest()
        ).decode()
        path, _, query = target.partition(b"?")
        scope = {
            "type": "websocket",
            "asgi": {"version": "3.0", "spec_version": "2.3"},
            "http_version": "1.1",
            "scheme": "ws",
            "path": unquote(path.decode()),
            "raw_path": path,
            "query_string": query,
            "root_path": self.config.root_path,
            "headers": headers,
            "client": None,
            "server": None,
            "subprotocols": subprotos,
            "state": dict(self.app_state) if isinstance(self.app_state, dict) else {},
        }
        self.recv_q.put_nowait({"type": "websocket.connect"})
        self._app_task = asyncio.ensure_future(self._run_app(scope))

    def _reject(self, code: int, reason: str) -> None:
        if self.transport is not None:
            self.transport.write(
                f"HTTP/1.1 {code} {reason}\r\ncontent-length: 0\r\n\r\n".encode()
            )
            self.transport.close()
        self.closed = True

    # -- frames -------------------------------------------------------------
    def _feed(self, data: bytes) -> None:
        for op, payload in self.parser.feed(data):
            if op == 0x1:
                self.recv_q.put_nowait(
                    {"type": "websocket.receive", "text": payload.decode("utf-8", "replace")}
                )
            elif op == 0x2:
                self.recv_q.put_nowait({"type": "websocket.receive", "bytes": payload})
            elif op == 0x8:
                code = struct.unpack(">H", payload[:2])[0] if len(payload) >= 2 else 1000
                self.recv_q.put_nowait({"type": "websocket.disconnect", "code": code})
                if not self.closed and self.transport is not None:
                    self.transport.write(_encode_frame(0x8, payload[:2]))
                    self.transport.close()
                self.closed = True
            elif op == 0x9:  # ping
                if self.transport is not None:
                    self.transport.write(_encode_frame(0xA, payload))

    # -- ASGI ---------------------------------------------------------------
    async def _run_app(self, scope) -> None:
        try:
            await self.app(scope, self._asgi_receive, self._asgi_send)
        except Exception:
            log.exception("websocket app crashed")
        finally:
            if not self.closed and self.transport is not None:
                try:
                    self.transport.close()
                except Exception:
                    pass
            self.closed = True

    async def _asgi_receive(self) -> Dict[str, Any]:
        return await self.recv_q.get()

    async def _asgi_send(self, message: Dict[str, Any]) -> None:
        t = message["type"]
        if self.transport is None or self.closed:
            return
        if t == "websocket.accept":
            proto = message.get("subprotocol")
            resp = (
                "HTTP/1.1 101 Switching Protocols\r\n"
                "Upgrade: websocket\r\n"
                "Connection: Upgrade\r\n"
                f"Sec-WebSocket-Accept: {self._accept_value}\r\n"
            )
            if proto:
                resp += f"Sec-WebSocket-Protocol: {proto}\r\n"
            self.transport.write((resp + "\r\n").encode())
            self.accepted = True
        elif t == "websocket.send":
            if message.get("text") is not None:
                self.transport.write(_encode_frame(0x1, message["text"].encode()))
            elif message.get("bytes") is not None:
                self.transport.write(_encode_frame(0x2, bytes(message["bytes"])))
        elif t == "websocket.close":
            code = int(message.get("code", 1000))
            if self.accepted:
                self.transport.write(_encode_frame(0x8, struct.pack(">H", code)))
            else:
                self._reject(403, "connection rejected")
            self.transport.close()
            self.closed = True

    def shutdown(self) -> None:  # uvicorn drain hook
        if self.transport is not None and not self.closed:
            self.transport.close()
