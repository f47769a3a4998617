"""RequestBuffer: per-stub request ring with container discovery, health
probing, per-container concurrency tokens and HTTP dispatch.

Reference behavior (reference: pkg/abstractions/endpoint/buffer.go —
discoverContainers :451-556 via the container-address map + /health probe,
requestTokens :558-636, handleHttpRequest :778, heartBeat :902, afterRequest
keep-warm refresh :941-952).
"""

from __future__ import annotations

import asyncio
import sys
import logging
import time
from typing import Any, Dict, Optional

import httpx

from ...common import keys
from ...common.state import StateStore

log = logging.getLogger(__name__)


class _ContainerSlot:
    def __init__(self, addr: str, tokens: int):
        self.addr = addr
        self.sema = asyncio.Semaphore(tokens)
        self.healthy = False
        self.last_probe = 0.0
        self.rpc: "RpcClient" = None  # grpc protocol connection


class RequestBuffer:
    """protocol: "grpc" = the fast internal dispatch to our endpoint runner
    (msgpack-gRPC Invoke/Health — ~10x cheaper per request than the pure-
    python HTTP stack in this image); "http" = reverse proxy to user ASGI
    apps (reference behavior: buffer.go:778 handleHttpRequest)."""

    def __init__(
        self,
        store: StateStore,
        container_repo,
        stub_id: str,
        concurrent_requests: int = 1,
        keep_warm_seconds: float = 10.0,
        request_timeout: float = 300.0,
        on_inflight_change=None,
        protocol: str = "grpc",
    ):
        self.store = store
        self.containers = container_repo
        self.stub_id = stub_id
        self.tokens_per_container = max(1, concurrent_requests)
        self.keep_warm_seconds = keep_warm_seconds
        self.request_timeout = request_timeout
        self.on_inflight_change = on_inflight_change
        self.protocol = protocol
        self.inflight = 0
        self._slots: Dict[str, _ContainerSlot] = {}
        self._httpx: Optional[httpx.AsyncClient] = None
        # unhealthy->healthy detection granularity; probes are a local gRPC
        # call, so keep this tight — it bounds cold-start detection latency
        self._probe_interval = 0.005

    @property
    def _client(self) -> httpx.AsyncClient:
        # lazy: constructing an AsyncClient builds an SSL context (~60 ms);
        # the grpc dispatch protocol never needs it
        if self._httpx is None:
            self._httpx = httpx.AsyncClient(
                timeout=httpx.Timeout(self.request_timeout, connect=5.0),
                limits=httpx.Limits(max_connections=512, max_keepalive_connections=256),
            )
        return self._httpx

    async def close(self) -> None:
        if self._httpx is not None:
            await self._httpx.aclose()
        for slot in self._slots.values():
            if slot.rpc is not None:
                await slot.rpc.close()

    async def _discover(self) -> Dict[str, _ContainerSlot]:
        addrs = await self.containers.addresses_for_stub(self.stub_id)
        for cid in list(self._slots.keys()):
            if cid not in addrs:
                old = self._slots.pop(cid)
                if old.rpc is not None:
                    asyncio.ensure_future(old.rpc.close())
        now = time.monotonic()
        for cid, addr in addrs.items():
            slot = self._slots.get(cid)
            if slot is None or slot.addr != addr:
                slot = _ContainerSlot(addr, self.tokens_per_container)
                self._slots[cid] = slot
            if not slot.healthy and now - slot.last_probe > self._probe_interval:
                slot.last_probe = now
                slot.healthy = await self._probe(slot)
        return self._slots

    async def _probe(self, slot: "_ContainerSlot") -> bool:
        if self.protocol == "grpc":
            try:
                if slot.rpc is None:
                    from ...rpc.channel import RpcClient

                    slot.rpc = RpcClient(slot.addr, private_pool=True)
                r = await slot.rpc.call("Runner", "Health", {}, timeout=2.0)
                return bool(r.get("ok"))
            except Exception:
                # a probe that raced the runner's bind leaves the channel in
                # reconnect backoff; every retry through it then fails until
                # the backoff expires (~100 ms — measured, benchmarks/
                # cold_trace.py). A fresh channel per failed probe makes the
                # next 5 ms probe do an immediate TCP connect instead.
                if slot.rpc is not None:
                    rpc, slot.rpc = slot.rpc, None
                    asyncio.ensure_future(rpc.close())
                return False
        try:
            r = await self._client.get(f"http://{slot.addr}/health", timeout=2.0)
            return r.status_code == 200
        except Exception:
            return False

    async def _acquire_slot(self, deadline: float, affinity_key: Optional[str] = None):
        """Pick a healthy container with a free token; wait for one.

        affinity_key: consistent-hash routing so repeat requests with the
        same key land on the same container while it has free tokens —
        KV-cache/prefix reuse for LLM serving (reference:
        pkg/abstractions/pod/llm.go prefix-affinity hashing). Falls back to
        most-free when the preferred container is saturated.
        """
        while time.monotonic() < deadline:
            slots = await self._discover()
            if affinity_key is not None:
                healthy = sorted(
                    cid for cid, s in slots.items() if s.healthy and not s.sema.locked()
                )
                if healthy:
                    import zlib

                    cid = healthy[zlib.crc32(affinity_key.encode()) % len(healthy)]
                    slot = slots[cid]
                    try:
                        await asyncio.wait_for(slot.sema.acquire(), timeout=0.05)
                        return cid, slot
                    except asyncio.TimeoutError:
                        pass  # saturated: fall through to most-free
            best: Optional[tuple] = None
            for cid, slot in slots.items():
                if not slot.healthy:
                    continue
                free = slot.sema._value  # pick the most-free container
                if not slot.sema.locked() and (best is None or free > best[2]):
                    best = (cid, slot, free)
            if best is not None:
                cid, slot, _ = best
                try:
                    await asyncio.wait_for(slot.sema.acquire(), timeout=0.05)
                    return cid, slot
                except asyncio.TimeoutError:
                    continue
            await asyncio.sleep(0.005)
        raise TimeoutError("no healthy container available")

    async def dispatch(
        self,
        method: str = "POST",
        path: str = "/",
        headers: Optional[Dict[str, str]] = None,
        content: bytes = b"",
        json_body: Any = None,
        wait_timeout: float = 120.0,
        affinity_key: Optional[str] = None,
    ) -> Dict[str, Any]:
        """Dispatch one request; returns {"status_code", "body"}."""
        self.inflight += 1
        if self.on_inflight_change:
            self.on_inflight_change(self.inflight)
        try:
            cid, slot = await self._acquire_slot(time.monotonic() + wait_timeout, affinity_key)
            try:
                if self.protocol == "grpc":
                    args = json_body
                    if args is None and content:
                        import json as _json

                        try:
                            args = _json.loads(content)
                        except ValueError:
                            args = {}
                    r = await slot.rpc.call(
                        "Runner", "Invoke", {"args": args, "path": path},
                        timeout=self.request_timeout,
                    )
                    await self._after_request(cid)
                    return r  # {"status_code", "body"}
                url = f"http://{slot.addr}{path}"
                kwargs: Dict[str, Any] = {"headers": headers or {}}
                if json_body is not None:
                    kwargs["json"] = json_body
                elif content:
                    kwargs["content"] = content
                resp = await self._client.request(method, url, **kwargs)
                await self._after_request(cid)
                try:
                    body = resp.json()
                except ValueError:
                    body = resp.content
                return {"status_code": resp.status_code, "body": body}
            except (httpx.ConnectError, httpx.ReadError):
                slot.healthy = False
                raise
            except Exception:
                if self.protocol == "grpc":
                    import grpc as _grpc

                    exc = sys.exc_info()[1]
                    if isinstance(exc, _grpc.aio.AioRpcError) and exc.code() in (
                        _grpc.StatusCode.UNAVAILABLE,
                        _grpc.StatusCode.DEADLINE_EXCEEDED,
                    ):
                        slot.healthy = False
                raise
            finally:
                slot.sema.release()
        finally:
            self.inflight -= 1
            if self.on_inflight_change:
                self.on_inflight_change(self.inflight)

    async def _after_request(self, cid: str) -> None:
        if self.keep_warm_seconds > 0:
            await self.store.set(
                keys.container_keep_warm(self.stub_id, cid), 1, ttl=self.keep_warm_seconds
            )

    async def healthy_count(self) -> int:
        return sum(1 for s in (await self._discover()).values() if s.healthy)

    async def wait_healthy(self, n: int = 1, timeout: float = 120.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if await self.healthy_count() >= n:
                return True
            await asyncio.sleep(0.01)
        return False
