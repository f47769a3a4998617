"""Raw sendfile content transport.

Out-of-band TCP reads with zero-copy sendfile on the server side
(reference: pkg/cache/raw_transport.go — binary {hash, offset, length}
request frames, admission control on inflight bytes; sendfile_linux.go).
Workers on one node (and across nodes later) pull image archives /
checkpoint payloads from whichever peer has them without re-hashing.

Frame formats (little-endian):
  request:  u8 magic=0xB9 | 64-byte hex hash | u64 offset | u64 length
  response: u8 status (0=ok,1=missing,2=range) | u64 length | payload
"""

from __future__ import annotations

import asyncio
import logging
import os
import socket
import struct
from typing import Optional

from .store import ContentStore

log = logging.getLogger(__name__)

MAGIC = 0xB9
MAX_REQUEST = 64 << 20  # 64 MiB per request (reference cap)


class RawTransportServer:
    def __init__(
        self,
        store: ContentStore,
        host: str = "127.0.0.1",
        port: int = 0,
        inflight_bytes_cap: int = 512 << 20,
        max_concurrent: int = 64,
    ):
        self.store = store
        self.host = host
        self.port = port
        self._server: Optional[asyncio.AbstractServer] = None
        self._sema = asyncio.Semaphore(max_concurrent)
        self._inflight_bytes = 0
        self._inflight_cap = inflight_bytes_cap

    async def start(self) -> int:
        self._server = await asyncio.start_server(self._handle, self.host, self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    async def _handle(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                header = await reader.readexactly(1 + 64 + 8 + 8)
                if header[0] != MAGIC:
                    return
                digest = header[1:65].decode()
                offset, length = struct.unpack("<QQ", header[65:81])
                if length > MAX_REQUEST:
                    writer.write(b"\x02" + struct.pack("<Q", 0))
                    await writer.drain()
                    continue
                path = self.store.get_path(digest)
                if path is None:
                    writer.write(b"\x01" + struct.pack("<Q", 0))
                    await writer.drain()
                    continue
                fsize = os.path.getsize(path)
                if length == 0:
                    length = max(0, fsize - offset)
                length = min(length, max(0, fsize - offset))
                async with self._sema:
                    while self._inflight_bytes + length > self._inflight_cap:
                        await asyncio.sleep(0.005)
                    self._inflight_bytes += length
                    try:
                        writer.write(b"\x00" + struct.pack("<Q", length))
                        await writer.drain()
                        await self._sendfile(writer, path, offset, length)
                        await writer.drain()
                    finally:
                        self._inflight_bytes -= length
        except (asyncio.IncompleteReadError, ConnectionResetError, BrokenPipeError):
            pass
        finally:
            writer.close()

    async def _sendfile(self, writer: asyncio.StreamWriter, path: str, offset: int, length: int) -> None:
        """Zero-copy page->socket via os.sendfile on the raw fd (the asyncio
        loop's sock_sendfile requires a plain socket transport)."""
        loop = asyncio.get_event_loop()
        sock: socket.socket = writer.get_extra_info("socket")
        with open(path, "rb") as f:
            if sock is not None:
                # flush buffered writer bytes first, then sendfile raw
                sent = 0
                f.seek(offset)
                try:
                    sock.setblocking(False)
                    while sent < length:
                        try:
                            n = os.sendfile(sock.fileno(), f.fileno(), offset + sent, length - sent)
                            if n == 0:
                                break
                            sent += n
                        except BlockingIOError:
                            await asyncio.sleep(0)
                    return
                except OSError:
                    pass  # fall back to buffered copy
            f.seek(offset)
            remaining = length
            while remaining > 0:
                chunk = f.read(min(1 << 20, remaining))
                if not chunk:
                    break
                writer.write(chunk)
                await writer.drain()
                remaining -= len(chunk)


class RawTransportClient:
    def __init__(self, host: str, port: int):
        self.host = host
        self.port = port
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None

    async def connect(self) -> None:
        if self._writer is None:
            self._reader, self._writer = await asyncio.open_connection(self.host, self.port)

    async def close(self) -> None:
        if self._writer is not None:
            self._writer.close()
            self._writer = None
            self._reader = None

    async def fetch(self, digest: str, offset: int = 0, length: int = 0) -> Optional[bytes]:
        await self.connect()
        req = bytes([MAGIC]) + digest.encode() + struct.pack("<QQ", offset, length)
        self._writer.write(req)
        await self._writer.drain()
        header = await self._reader.readexactly(9)
        status = header[0]
        size = struct.unpack("<Q", header[1:9])[0]
        if status != 0:
            return None
        return await self._reader.readexactly(size)

    async def fetch_to_file(self, digest: str, dest: str, chunk: int = 32 << 20) -> bool:
        """Pull large content in chunked requests into a local file."""
        await self.connect()
        offset = 0
        with open(dest + ".tmp", "wb") as f:
            while True:
                data = await self.fetch(digest, offset, chunk)
                if data is None:
                    if offset == 0:
                        os.unlink(dest + ".tmp")
                        return False
                    break
                if not data:
                    break
                f.write(data)
                offset += len(data)
                if len(data) < chunk:
                    break
        os.rename(dest + ".tmp", dest)
        return True
