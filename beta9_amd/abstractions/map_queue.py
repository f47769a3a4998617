"""Map / SimpleQueue / Signal abstractions — store-backed distributed
primitives for user code (reference: pkg/abstractions/map/ 5 RPCs,
pkg/abstractions/queue/ 5 RPCs, pkg/abstractions/experimental/signal/ 3
RPCs).
"""

from __future__ import annotations

from typing import Any, Dict

from ..common import keys
from ..common.state import StateStore
from ..rpc.channel import rpc


class MapService:
    service_name = "MapService"

    def __init__(self, store: StateStore):
        self.store = store

    def _key(self, auth: Dict, name: str) -> str:
        return keys.map_key(auth["workspace_id"], name)

    @rpc()
    async def MapSet(self, req: Dict[str, Any], context, auth) -> bool:
        await self.store.hset(self._key(auth, req["name"]), {req["key"]: req["value"]})
        return True

    @rpc()
    async def MapGet(self, req: Dict[str, Any], context, auth) -> Any:
        return await self.store.hget(self._key(auth, req["name"]), req["key"])

    @rpc()
    async def MapDelete(self, req: Dict[str, Any], context, auth) -> bool:
        await self.store.hdel(self._key(auth, req["name"]), req["key"])
        return True

    @rpc()
    async def MapCount(self, req: Dict[str, Any], context, auth) -> int:
        return len(await self.store.hgetall(self._key(auth, req["name"])))

    @rpc()
    async def MapKeys(self, req: Dict[str, Any], context, auth) -> list:
        return list((await self.store.hgetall(self._key(auth, req["name"]))).keys())


class SimpleQueueService:
    service_name = "SimpleQueueService"

    def __init__(self, store: StateStore):
        self.store = store

    def _key(self, auth: Dict, name: str) -> str:
        return keys.queue_key(auth["workspace_id"], name)

    @rpc()
    async def SimpleQueuePut(self, req: Dict[str, Any], context, auth) -> bool:
        await self.store.rpush(self._key(auth, req["name"]), req["value"])
        return True

    @rpc()
    async def SimpleQueuePop(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        v = await self.store.blpop(self._key(auth, req["name"]), float(req.get("timeout", 0.0)))
        return {"ok": v is not None, "value": v}

    @rpc()
    async def SimpleQueuePeek(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        items = await self.store.lrange(self._key(auth, req["name"]), 0, 0)
        return {"ok": bool(items), "value": items[0] if items else None}

    @rpc()
    async def SimpleQueueEmpty(self, req: Dict[str, Any], context, auth) -> bool:
        return (await self.store.llen(self._key(auth, req["name"]))) == 0

    @rpc()
    async def SimpleQueueSize(self, req: Dict[str, Any], context, auth) -> int:
        return await self.store.llen(self._key(auth, req["name"]))


class SignalService:
    service_name = "SignalService"

    def __init__(self, store: StateStore):
        self.store = store

    def _key(self, auth: Dict, name: str) -> str:
        return keys.signal_key(auth["workspace_id"], name)

    @rpc()
    async def SignalSet(self, req: Dict[str, Any], context, auth) -> bool:
        ttl = float(req.get("ttl", 0)) or None
        await self.store.set(self._key(auth, req["name"]), 1, ttl=ttl)
        return True

    @rpc()
    async def SignalClear(self, req: Dict[str, Any], context, auth) -> bool:
        await self.store.delete(self._key(auth, req["name"]))
        return True

    @rpc()
    async def SignalMonitor(self, req: Dict[str, Any], context, auth) -> bool:
        return (await self.store.get(self._key(auth, req["name"]))) is not None
