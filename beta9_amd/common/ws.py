"""Minimal RFC 6455 WebSocket client (no extensions, text/binary frames).

The target image ships no websocket client library; the platform needs one
to health-check/exercise realtime (ASGI + WebSocket) stubs, whose traffic
reaches containers through the protocol-agnostic TCP ingress (reference:
proxyWebsocketConnection, pkg/abstractions/endpoint/buffer.go:958 — our
TcpForwarder/SNI ingress splices bytes, so WS works through it verbatim).
"""

from __future__ import annotations

import asyncio
import base64
import hashlib
import os
import struct
from typing import Optional, Tuple

_GUID = "258EAFA5-E914-47DA-95CA-C5AB0DC85B11"


class WebSocketError(ConnectionError):
    pass


class WSClient:
    def __init__(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        self.reader = reader
        self.writer = writer

    @classmethod
    async def connect(cls, host: str, port: int, path: str = "/", timeout: float = 10.0) -> "WSClient":
        reader, writer = await asyncio.wait_for(
            asyncio.open_connection(host, port), timeout
        )
        key = base64.b64encode(os.urandom(16)).decode()
        req = (
            f"GET {path} HTTP/1.1\r\n"
            f"Host: {host}:{port}\r\n"
            "Upgrade: websocket\r\n"
            "Connection: Upgrade\r\n"
            f"Sec-WebSocket-Key: {key}\r\n"
            "Sec-WebSocket-Version: 13\r\n\r\n"
        )
        writer.write(req.encode())
        await writer.drain()
        status = await asyncio.wait_for(reader.readline(), timeout)
        if b"101" not in status:
            raise WebSocketError(f"handshake rejected: {status.decode(errors='replace').strip()}")
        accept = ""
        while True:
            line = await asyncio.wait_for(reader.readline(), timeout)
            if line in (b"\r\n", b""):
                break
            k, _, v = line.decode(errors="replace").partition(":")
            if k.strip().lower() == "sec-websocket-accept":
                accept = v.strip()
        want = base64.b64encode(hashlib.sha1((key + _GUID).encode()).digest()).decode()
        if accept != want:
            raise WebSocketError("bad Sec-WebSocket-Accept")
        return cls(reader, writer)

    async def send(self, data, opcode: Optional[int] = None) -> None:
        if isinstance(data, str):
            payload = data.encode()
            op = 0x1 if opcode is None else opcode
        else:
            payload = bytes(data)
            op = 0x2 if opcode is None else opcode
        mask = os.urandom(4)
        header = bytes([0x80 | op])
        n = len(payload)
        if n < 126:
            header += bytes([0x80 | n])
        elif n < (1 << 16):
            header += bytes([0x80 | 126]) + struct.pack(">H", n)
        else:
            header += bytes([0x80 | 127]) + struct.pack(">Q", n)
        masked = bytes(b ^ mask[i % 4] for i, b in enumerate(payload))
        self.writer.write(header + mask + masked)
        await self.writer.drain()

    async def recv(self, timeout: float = 10.0) -> Tuple[int, bytes]:
        """Returns (opcode, payload); handles ping transparently."""
        while True:
            hdr = await asyncio.wait_for(self.reader.readexactly(2), timeout)
            op = hdr[0] & 0x0F
            masked = hdr[1] & 0x80
            n = hdr[1] & 0x7F
            if n == 126:
                n = struct.unpack(">H", await self.reader.readexactly(2))[0]
            elif n == 127:
                n = struct.unpack(">Q", await self.reader.readexactly(8))[0]
            mkey = await self.reader.readexactly(4) if masked else b""
            payload = await asyncio.wait_for(self.reader.readexactly(n), timeout)
            if masked:
                payload = bytes(b ^ mkey[i % 4] for i, b in enumerate(payload))
            if op == 0x9:  # ping -> pong
                await self.send(payload, opcode=0xA)
                continue
            return op, payload

    async def close(self) -> None:
        try:
            await self.send(b"", opcode=0x8)
        except (OSError, WebSocketError):
            pass
        self.writer.close()
