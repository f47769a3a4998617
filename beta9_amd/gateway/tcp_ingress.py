"""Raw TCP ingress for pod ports.

(reference: pkg/abstractions/pod/tcp.go — TLS listener :1995 with per-stub
SNI certs forwarding raw streams to container ports.) Single-node build:
per-exposed-port TCP forwarders — ExposePort allocates a gateway-side port
and splices bytes to the container's port; no TLS termination needed inside
one node (terminate at your edge).
"""

from __future__ import annotations

import asyncio
import logging
from typing import Dict, Optional, Tuple

log = logging.getLogger(__name__)


class TcpForwarder:
    def __init__(self, target_host: str, target_port: int):
        self.target_host = target_host
        self.target_port = target_port
        self.port = 0
        self._server: Optional[asyncio.AbstractServer] = None

    async def start(self, host: str = "0.0.0.0") -> int:
        self._server = await asyncio.start_server(self._handle, host, 0)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    async def _handle(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        try:
            up_r, up_w = await asyncio.open_connection(self.target_host, self.target_port)
        except OSError:
            writer.close()
            return

        async def pump(src: asyncio.StreamReader, dst: asyncio.StreamWriter) -> None:
            try:
                while True:
                    data = await src.read(65536)
                    if not data:
                        # half-close: propagate EOF, keep the other
                        # direction flowing (an echo reply may still come)
                        if dst.can_write_eof():
                            try:
                                dst.write_eof()
                            except (OSError, RuntimeError):
                                pass
                        break
                    dst.write(data)
                    await dst.drain()
            except (ConnectionResetError, BrokenPipeError):
                pass

        try:
            await asyncio.gather(pump(reader, up_w), pump(up_r, writer))
        finally:
            for w in (up_w, writer):
                try:
                    w.close()
                except Exception:
                    pass


class TcpIngress:
    """container_id:port -> forwarder registry."""

    def __init__(self) -> None:
        self._forwarders: Dict[Tuple[str, int], TcpForwarder] = {}

    async def expose(self, container_id: str, container_host: str, container_port: int) -> int:
        key = (container_id, container_port)
        fwd = self._forwarders.get(key)
        if fwd is not None:
            return fwd.port
        fwd = TcpForwarder(container_host, container_port)
        port = await fwd.start()
        self._forwarders[key] = fwd
        return port

    async def unexpose(self, container_id: str) -> None:
        for key in [k for k in self._forwarders if k[0] == container_id]:
            await self._forwarders.pop(key).stop()

    async def shutdown(self) -> None:
        for fwd in self._forwarders.values():
            await fwd.stop()
        self._forwarders.clear()
