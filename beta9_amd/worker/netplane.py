"""Per-container network plane: netns slot pool + veth/bridge + IP
allocator + userspace port proxies.

Reference behavior being reproduced (reference: pkg/worker/network.go —
Setup :1414-1520 builds netns+veth onto a shared bridge with a
cluster-prefix IP allocator; maintainNetworkSlotPool :542-624 preallocates
slots so cold start skips netns/veth creation; container_port_proxy.go
:95-174 is the userspace per-port TCP proxy fallback). This image ships no
iptables, so exposed ports ALWAYS use the userspace proxy path, and
`--block-network` / allow-CIDR egress policy is route-based: a blocked
container's netns simply has no default route (loopback only), an
allow-list gets host routes for just those CIDRs.

Slots are held open by _netns_holder subprocesses; containers join a slot
via b9launch --netns-fd (setns before pivot), so the namespace survives
container restarts and the pool absorbs the setup cost off the cold path.
"""

from __future__ import annotations

import asyncio
import ipaddress
import logging
import os
import sys
import time
from typing import Dict, List, Optional, Set, Tuple

from .netlink import NetlinkError, Rtnl

log = logging.getLogger(__name__)


class NetworkSlot:
    def __init__(self, idx: int, ip: str, veth_host: str, veth_ns: str, holder):
        self.idx = idx
        self.ip = ip
        self.veth_host = veth_host
        self.veth_ns = veth_ns
        self.holder = holder  # asyncio.subprocess.Process
        self.ns_path = f"/proc/{holder.pid}/ns/net"
        self._ns_fd: Optional[int] = None

    def ns_fd(self) -> int:
        """Open (and cache) an fd for the namespace — keeps it alive even
        if the holder dies, and is what b9launch setns() consumes."""
        if self._ns_fd is None:
            self._ns_fd = os.open(self.ns_path, os.O_RDONLY)
        return self._ns_fd

    def close(self) -> None:
        if self._ns_fd is not None:
            try:
                os.close(self._ns_fd)
            except OSError:
                pass
            self._ns_fd = None


async def _holder_cmd(holder, cmd: str) -> None:
    holder.stdin.write((cmd + "\n").encode())
    await holder.stdin.drain()
    line = await asyncio.wait_for(holder.stdout.readline(), timeout=10.0)
    if not line.startswith(b"OK"):
        raise RuntimeError(f"netns holder: {line.decode().strip()} (cmd: {cmd})")


class NetworkPlane:
    def __init__(
        self,
        bridge: str = "b9br0",
        subnet: str = "10.201.0.0/24",
        prealloc_slots: int = 4,
    ):
        self.bridge = bridge
        self.net = ipaddress.ip_network(subnet)
        hosts = list(self.net.hosts())
        self.gateway_ip = str(hosts[0])
        self._free_ips: List[str] = [str(h) for h in hosts[1:]]
        self.prealloc = prealloc_slots
        self._pool: List[NetworkSlot] = []
        self._active: Dict[str, NetworkSlot] = {}  # container_id -> slot
        self._proxies: Dict[Tuple[str, int], asyncio.AbstractServer] = {}
        import uuid

        self._uid = uuid.uuid4().hex[:4]  # unique link names per plane
        self._n = 0
        self._started = False
        self._replenishing = False

    # -- lifecycle -----------------------------------------------------------
    async def start(self) -> None:
        nl = Rtnl()
        try:
            try:
                nl.link_add_bridge(self.bridge)
            except NetlinkError as e:
                if e.errno != 17:  # EEXIST: adopt the existing bridge
                    raise
            try:
                nl.addr_add(self.bridge, self.gateway_ip, self.net.prefixlen)
            except NetlinkError as e:
                if e.errno != 17:
                    raise
            nl.link_set_up(self.bridge)
        finally:
            nl.close()
        self._started = True
        # prealloc in the background: worker boot must not wait ~1s/slot
        # (holder python spawn); early pool misses build slots inline
        self._prealloc_task = asyncio.ensure_future(self._replenish())

    async def _add_slot(self, pooled: bool = True) -> Optional[NetworkSlot]:
        if not self._started:
            return None
        self._n += 1
        idx = self._n
        holder = await asyncio.create_subprocess_exec(
            sys.executable, "-u", "-m", "beta9_amd.worker._netns_holder",
            stdin=asyncio.subprocess.PIPE,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.DEVNULL,
            start_new_session=True,
        )
        line = await asyncio.wait_for(holder.stdout.readline(), timeout=30.0)
        if not line.startswith(b"READY"):
            raise RuntimeError(f"netns holder failed: {line.decode().strip()}")
        ip = self._free_ips.pop(0)
        veth_host = f"b9{self._uid}h{idx}"
        veth_ns = f"b9{self._uid}c{idx}"
        nl = Rtnl()
        try:
            nl.link_add_veth(veth_host, veth_ns)
            nl.link_set_master(veth_host, self.bridge)
            nl.link_set_up(veth_host)
            nl.link_set_netns_pid(veth_ns, holder.pid)
        finally:
            nl.close()
        await _holder_cmd(holder, f"addr {veth_ns} {ip} {self.net.prefixlen}")
        await _holder_cmd(holder, f"up {veth_ns}")
        await _holder_cmd(holder, f"route {self.gateway_ip}")
        # config done: the holder execs into `sleep` (python freed, ns kept)
        await _holder_cmd(holder, "hold")
        slot = NetworkSlot(idx, ip, veth_host, veth_ns, holder)
        if pooled:
            self._pool.append(slot)
        return slot

    # -- allocation ----------------------------------------------------------
    async def acquire(self, container_id: str) -> Tuple[NetworkSlot, bool]:
        """Returns (slot, pool_hit). Pool miss builds a slot inline."""
        hit = bool(self._pool)
        if self._pool:
            slot = self._pool.pop(0)
            asyncio.ensure_future(self._replenish())
        else:
            slot = await self._add_slot(pooled=False)
        self._active[container_id] = slot
        return slot, hit

    async def _replenish(self) -> None:
        if self._replenishing or not self._started:
            return
        self._replenishing = True
        try:
            while self._started and len(self._pool) < self.prealloc:
                await self._add_slot()
        except Exception:
            log.exception("slot replenish failed")
        finally:
            self._replenishing = False

    async def release(self, container_id: str) -> None:
        slot = self._active.pop(container_id, None)
        if slot is None:
            return
        # namespaces are single-use (the container may have reconfigured
        # routes/ports): tear down and let the pool replenish
        await self._destroy_slot(slot)

    async def _destroy_slot(self, slot: NetworkSlot) -> None:
        slot.close()
        if slot.holder.returncode is None:
            try:
                slot.holder.terminate()
                await asyncio.wait_for(slot.holder.wait(), timeout=5.0)
            except (asyncio.TimeoutError, ProcessLookupError):
                try:
                    slot.holder.kill()
                except ProcessLookupError:
                    pass
        nl = Rtnl()
        try:
            nl.link_del(slot.veth_host)
        except (NetlinkError, OSError):
            pass  # vanished with the namespace
        finally:
            nl.close()
        self._free_ips.append(slot.ip)

    # -- exposed ports (userspace DNAT) ---------------------------------------
    async def expose_port(self, host_port: int, container_ip: str, container_port: int) -> None:
        """host:host_port -> container_ip:container_port TCP proxy
        (reference: container_port_proxy.go:95-174)."""

        async def handle(reader, writer):
            try:
                r2, w2 = await asyncio.open_connection(container_ip, container_port)
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
                except (OSError, asyncio.CancelledError):
                    pass
                finally:
                    try:
                        dst.close()
                    except OSError:
                        pass

            await asyncio.gather(pump(reader, w2), pump(r2, writer))

        srv = await asyncio.start_server(handle, "0.0.0.0", host_port)
        self._proxies[(host_port, container_port)] = srv

    async def unexpose_port(self, host_port: int, container_port: int) -> None:
        srv = self._proxies.pop((host_port, container_port), None)
        if srv is not None:
            srv.close()
            await srv.wait_closed()

    async def shutdown(self) -> None:
        self._started = False  # stops in-flight replenish loops
        t = getattr(self, "_prealloc_task", None)
        if t is not None and not t.done():
            t.cancel()
            try:
                await t
            except (asyncio.CancelledError, Exception):
                pass
        for srv in self._proxies.values():
            srv.close()
        self._proxies.clear()
        for slot in self._pool + list(self._active.values()):
            await self._destroy_slot(slot)
        self._pool.clear()
        self._active.clear()
        nl = Rtnl()
        try:
            nl.link_del(self.bridge)
        except (NetlinkError, OSError):
            pass
        finally:
            nl.close()
        self._started = False


def netplane_available() -> bool:
    """Netlink + unshare(NEWNET) probe."""
    import ctypes

    try:
        nl = Rtnl()
        nl.close()
    except OSError:
        return False
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    pid = os.fork()
    if pid == 0:
        r = libc.unshare(0x40000000)
        os._exit(0 if r == 0 else 1)
    _, st = os.waitpid(pid, 0)
    return os.WEXITSTATUS(st) == 0
