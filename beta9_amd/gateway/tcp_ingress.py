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


class SniTlsIngress:
    """Single TLS listener routing by SNI hostname to container ports
    (reference: pkg/abstractions/pod/tcp.go:48-144 — TLS listener :1995
    with per-stub SNI certs, self-signed fallback).

    Hostname contract: `{name}.tcp.{base_domain}` where name is a stub id
    or routing key registered via add_route(). Certs are self-signed
    per-hostname (openssl, cached on disk) — exactly the reference's
    fallback when no cert store is configured.
    """

    def __init__(self, base_domain: str = "b9.local", cert_dir: str = ""):
        import tempfile

        self.base_domain = base_domain
        self.cert_dir = cert_dir or tempfile.mkdtemp(prefix="b9-sni-")
        self.port = 0
        self._server: Optional[asyncio.AbstractServer] = None
        self._routes: Dict[str, Tuple[str, int]] = {}  # name -> (host, port)
        self._ctxs: Dict[str, "object"] = {}

    def add_route(self, name: str, host: str, port: int) -> None:
        self._routes[name] = (host, port)

    def remove_route(self, name: str) -> None:
        self._routes.pop(name, None)

    def _ctx_for(self, hostname: str):
        import ssl
        import subprocess

        ctx = self._ctxs.get(hostname)
        if ctx is not None:
            return ctx
        safe = "".join(c if c.isalnum() or c in ".-" else "_" for c in hostname)
        crt = f"{self.cert_dir}/{safe}.crt"
        key = f"{self.cert_dir}/{safe}.key"
        import os

        if not (os.path.exists(crt) and os.path.exists(key)):
            subprocess.run(
                ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
                 "-keyout", key, "-out", crt, "-days", "30",
                 "-subj", f"/CN={hostname}",
                 "-addext", f"subjectAltName=DNS:{hostname}"],
                check=True, capture_output=True,
            )
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(crt, key)
        self._ctxs[hostname] = ctx
        return ctx

    async def start(self, host: str = "0.0.0.0", port: int = 0) -> int:
        import ssl

        base = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        fallback = self._ctx_for(f"ingress.tcp.{self.base_domain}")

        def sni_cb(sslobj, server_name, _ctx):
            try:
                sslobj.context = self._ctx_for(server_name) if server_name else fallback
                sslobj._b9_sni = server_name  # stashed for routing
            except Exception:
                sslobj.context = fallback

        base.sni_callback = sni_cb
        # the base context needs SOME cert for no-SNI clients
        base.load_cert_chain(
            f"{self.cert_dir}/ingress.tcp.{self.base_domain}.crt",
            f"{self.cert_dir}/ingress.tcp.{self.base_domain}.key",
        )
        self._server = await asyncio.start_server(self._handle, host, port, ssl=base)
        self.port = self._server.sockets[0].getsockname()[1]
        return self.port

    async def _handle(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        sslobj = writer.get_extra_info("ssl_object")
        sni = getattr(sslobj, "_b9_sni", "") or ""
        suffix = f".tcp.{self.base_domain}"
        name = sni[: -len(suffix)] if sni.endswith(suffix) else sni
        target = self._routes.get(name)
        if target is None:
            writer.close()
            return
        try:
            up_r, up_w = await asyncio.open_connection(*target)
        except OSError:
            writer.close()
            return

        async def pump(src, dst):
            try:
                while True:
                    data = await src.read(65536)
                    if not data:
                        break
                    dst.write(data)
                    await dst.drain()
            except (ConnectionResetError, BrokenPipeError, OSError):
                pass
            finally:
                try:
                    dst.close()
                except Exception:
                    pass

        await asyncio.gather(pump(reader, up_w), pump(up_r, writer))

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None


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
