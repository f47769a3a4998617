"""CacheService gRPC surface + coordinator registry.

(reference: pkg/cache/cache.proto:7-17 GetContent/GetContentStream/
StoreContent/StoreContentFromSource, coordinator.go host registry with TTL
leases). The gateway hosts the coordinator; every worker's embedded store
registers as a logical host; bulk bytes move over the raw sendfile
transport (transport.py), gRPC carries lookups and small content.
"""

from __future__ import annotations

import time
from typing import Any, AsyncIterator, Dict, List

from ..common.state import StateStore
from ..rpc.channel import not_found, rpc
from .store import ContentStore


class CacheCoordinator:
    """Logical host registry with TTL leases."""

    def __init__(self, store: StateStore, lease_ttl_s: float = 30.0):
        self.store = store
        self.lease_ttl_s = lease_ttl_s

    async def register(self, host_id: str, addr: str, transport_port: int) -> None:
        await self.store.set(
            f"cache:host:{host_id}",
            {"addr": addr, "transport_port": transport_port, "ts": time.time()},
            ttl=self.lease_ttl_s,
        )

    async def hosts(self) -> Dict[str, Dict]:
        out = {}
        for key in await self.store.keys("cache:host:*"):
            v = await self.store.get(key)
            if v:
                out[key.split(":")[-1]] = v
        return out


class CacheService:
    service_name = "CacheService"

    def __init__(self, content: ContentStore, coordinator: CacheCoordinator, transport_port: int = 0):
        self.content = content
        self.coordinator = coordinator
        self.transport_port = transport_port

    @rpc()
    async def HasContent(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        digest = req["hash"]
        return {
            "exists": self.content.has(digest),
            "size": self.content.size(digest),
            "transport_port": self.transport_port,
        }

    @rpc()
    async def GetContent(self, req: Dict[str, Any], context, auth) -> bytes:
        data = self.content.read(req["hash"], int(req.get("offset", 0)), int(req.get("length", 0)))
        if data is None:
            raise not_found(req["hash"])
        return data

    @rpc(kind="unary_stream")
    async def GetContentStream(self, req: Dict[str, Any], context, auth) -> AsyncIterator[bytes]:
        path = self.content.get_path(req["hash"])
        if path is None:
            raise not_found(req["hash"])
        with open(path, "rb") as f:
            while True:
                chunk = f.read(1 << 20)
                if not chunk:
                    return
                yield chunk

    @rpc(kind="stream_unary")
    async def StoreContent(self, request_iter, context, auth) -> Dict[str, Any]:
        chunks: List[bytes] = []
        expected = ""
        async for msg in request_iter:
            if isinstance(msg, dict):
                expected = msg.get("hash", expected)
                data = msg.get("data", b"")
            else:
                data = msg
            if data:
                chunks.append(data)
        digest = self.content.add_stream(iter(chunks), expected_hash=expected)
        return {"hash": digest, "size": self.content.size(digest)}

    @rpc()
    async def StoreContentFromSource(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Ingest a file already on this node's filesystem (image archives,
        checkpoint snapshots)."""
        digest = self.content.add_file(req["path"], move=bool(req.get("move", False)))
        return {"hash": digest, "size": self.content.size(digest)}

    @rpc()
    async def RegisterCacheHost(self, req: Dict[str, Any], context, auth) -> bool:
        await self.coordinator.register(req["host_id"], req["addr"], int(req.get("transport_port", 0)))
        return True

    @rpc()
    async def ListCacheHosts(self, req: Dict[str, Any], context, auth) -> Dict[str, Dict]:
        return await self.coordinator.hosts()
