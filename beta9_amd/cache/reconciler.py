"""Cache reconciler: keeps a node's content store stocked from its peers.

Reference behavior (reference: pkg/cache/ — worker cache nodes register in
a host map, clients measure RTT and fetch pages from the nTopHosts=3
closest peers over the raw transport; hostmap.go:174-230,
raw_transport.go): each node runs one reconciler that

- renews its host-map lease (CacheCoordinator.register, TTL-scoped),
- resolves missing digests by ranking peers by measured connect RTT and
  pulling over the raw sendfile transport into the local store (hash
  verified by ContentStore.add_file).
"""

from __future__ import annotations

import asyncio
import logging
import os
import time
from typing import Dict, List, Optional

from .store import ContentStore
from .transport import RawTransportClient

log = logging.getLogger(__name__)


class CacheReconciler:
    N_TOP_HOSTS = 3

    def __init__(
        self,
        content: ContentStore,
        coordinator,
        host_id: str,
        addr: str,
        transport_port: int,
        lease_ttl_s: float = 30.0,
        rtt_refresh_s: float = 60.0,
    ):
        self.content = content
        self.coordinator = coordinator
        self.host_id = host_id
        self.addr = addr
        self.transport_port = transport_port
        self.lease_ttl_s = lease_ttl_s
        self.rtt_refresh_s = rtt_refresh_s
        self._rtt: Dict[str, tuple] = {}  # host_id -> (rtt_s, measured_at)
        self._task: Optional[asyncio.Task] = None

    async def start(self) -> None:
        await self.coordinator.register(self.host_id, self.addr, self.transport_port)
        if self._task is None:
            self._task = asyncio.create_task(self._lease_loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None

    async def _lease_loop(self) -> None:
        while True:
            await asyncio.sleep(self.lease_ttl_s / 3)
            try:
                await self.coordinator.register(self.host_id, self.addr, self.transport_port)
            except Exception:
                log.warning("cache host lease renewal failed", exc_info=True)

    # -- peer selection ------------------------------------------------------
    async def _measure_rtt(self, host_id: str, addr: str, port: int) -> float:
        t0 = time.perf_counter()
        try:
            reader, writer = await asyncio.wait_for(
                asyncio.open_connection(addr.split(":")[0], port), timeout=2.0
            )
            writer.close()
            try:
                await writer.wait_closed()
            except Exception:
                pass
            rtt = time.perf_counter() - t0
        except Exception:
            rtt = float("inf")
        self._rtt[host_id] = (rtt, time.monotonic())
        return rtt

    async def ranked_peers(self) -> List[Dict]:
        hosts = await self.coordinator.hosts()
        peers = []
        now = time.monotonic()
        for hid, h in hosts.items():
            if hid == self.host_id:
                continue
            cached = self._rtt.get(hid)
            if cached is None or now - cached[1] > self.rtt_refresh_s:
                rtt = await self._measure_rtt(hid, h["addr"], int(h["transport_port"]))
            else:
                rtt = cached[0]
            if rtt != float("inf"):
                peers.append({**h, "host_id": hid, "rtt": rtt})
        peers.sort(key=lambda p: p["rtt"])
        return peers[: self.N_TOP_HOSTS]

    # -- content resolution --------------------------------------------------
    async def ensure(self, digest: str) -> Optional[str]:
        """Local path for `digest`, pulling from the closest peer that has
        it when missing. Returns None if no peer can provide it."""
        p = self.content.get_path(digest)
        if p is not None:
            return p
        for peer in await self.ranked_peers():
            host = peer["addr"].split(":")[0]
            client = RawTransportClient(host, int(peer["transport_port"]))
            tmp = os.path.join(self.content.root, f".pull-{digest[:16]}-{os.getpid()}")
            try:
                if await client.fetch_to_file(digest, tmp):
                    got = self.content.add_file(tmp, move=True)
                    if got == digest:
                        return self.content.get_path(digest)
                    self.content.remove(got)
                    log.warning("peer %s served wrong content for %s", peer["host_id"], digest)
            except Exception:
                log.debug("pull of %s from %s failed", digest, peer["host_id"], exc_info=True)
            finally:
                if os.path.exists(tmp):
                    try:
                        os.unlink(tmp)
                    except OSError:
                        pass
        return None
