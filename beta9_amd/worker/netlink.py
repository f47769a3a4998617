"""Zero-dependency rtnetlink: bridge/veth/addr/route/netns programming.

The reference shells out to iptables and uses netlink libraries for its
per-container network plane (reference: pkg/worker/network.go:1414-1520
Setup — netns + veth onto a shared bridge + IP allocator). This target
image ships NO `ip`, `iptables` or netlink python libs, so the plane is
built directly on AF_NETLINK sockets — first-party, ~300 lines, exactly
the subset the worker needs:

  - create bridge / veth pair            (RTM_NEWLINK + IFLA_LINKINFO)
  - enslave link to bridge               (IFLA_MASTER)
  - move link into a network namespace   (IFLA_NET_NS_PID / _FD)
  - set link up                          (IFF_UP)
  - add IPv4 address                     (RTM_NEWADDR)
  - add default route via gateway        (RTM_NEWROUTE)
  - delete link                          (RTM_DELLINK)
"""

from __future__ import annotations

import os
import socket
import struct
from typing import List, Optional, Tuple

# netlink
NLM_F_REQUEST = 0x1
NLM_F_ACK = 0x4
NLM_F_EXCL = 0x200
NLM_F_CREATE = 0x400
NLMSG_ERROR = 0x2

RTM_NEWLINK = 16
RTM_DELLINK = 17
RTM_NEWADDR = 20
RTM_NEWROUTE = 24

# link attrs
IFLA_IFNAME = 3
IFLA_MASTER = 10
IFLA_LINKINFO = 18
IFLA_NET_NS_PID = 19
IFLA_NET_NS_FD = 28
IFLA_INFO_KIND = 1
IFLA_INFO_DATA = 2
VETH_INFO_PEER = 1

# addr attrs
IFA_ADDRESS = 1
IFA_LOCAL = 2

# route attrs
RTA_DST = 1
RTA_GATEWAY = 5
RT_TABLE_MAIN = 254
RTPROT_BOOT = 3
RT_SCOPE_UNIVERSE = 0
RTN_UNICAST = 1

IFF_UP = 1


def _align(n: int) -> int:
    return (n + 3) & ~3


def _attr(attr_type: int, payload: bytes) -> bytes:
    length = 4 + len(payload)
    return struct.pack("=HH", length, attr_type) + payload + b"\0" * (_align(length) - length)


def _nested(attr_type: int, payload: bytes) -> bytes:
    return _attr(attr_type, payload)


class NetlinkError(OSError):
    pass


class Rtnl:
    def __init__(self) -> None:
        self.sock = socket.socket(socket.AF_NETLINK, socket.SOCK_RAW, socket.NETLINK_ROUTE)
        self.sock.bind((0, 0))
        self.seq = 0

    def close(self) -> None:
        self.sock.close()

    def _send(self, msg_type: int, flags: int, body: bytes) -> None:
        self.seq += 1
        hdr = struct.pack("=IHHII", 16 + len(body), msg_type, flags | NLM_F_ACK, self.seq, 0)
        self.sock.send(hdr + body)
        self._wait_ack(self.seq)

    def _wait_ack(self, seq: int) -> None:
        while True:
            data = self.sock.recv(65536)
            off = 0
            while off < len(data):
                ln, typ, _flags, mseq, _pid = struct.unpack_from("=IHHII", data, off)
                if mseq == seq and typ == NLMSG_ERROR:
                    err = struct.unpack_from("=i", data, off + 16)[0]
                    if err != 0:
                        raise NetlinkError(-err, os.strerror(-err))
                    return
                off += _align(ln)

    # -- links ---------------------------------------------------------------
    def _ifinfomsg(self, index: int = 0, flags: int = 0, change: int = 0) -> bytes:
        return struct.pack("=BBHiII", socket.AF_UNSPEC, 0, 0, index, flags, change)

    def link_add_bridge(self, name: str) -> None:
        body = self._ifinfomsg()
        body += _attr(IFLA_IFNAME, name.encode() + b"\0")
        body += _nested(IFLA_LINKINFO, _attr(IFLA_INFO_KIND, b"bridge"))
        self._send(RTM_NEWLINK, NLM_F_REQUEST | NLM_F_CREATE | NLM_F_EXCL, body)

    def link_add_veth(self, name: str, peer: str) -> None:
        peer_body = self._ifinfomsg() + _attr(IFLA_IFNAME, peer.encode() + b"\0")
        info_data = _nested(VETH_INFO_PEER, peer_body)
        body = self._ifinfomsg()
        body += _attr(IFLA_IFNAME, name.encode() + b"\0")
        body += _nested(
            IFLA_LINKINFO, _attr(IFLA_INFO_KIND, b"veth") + _nested(IFLA_INFO_DATA, info_data)
        )
        self._send(RTM_NEWLINK, NLM_F_REQUEST | NLM_F_CREATE | NLM_F_EXCL, body)

    def link_del(self, name: str) -> None:
        idx = socket.if_nametoindex(name)
        self._send(RTM_DELLINK, NLM_F_REQUEST, self._ifinfomsg(index=idx))

    def link_set_master(self, name: str, master: str) -> None:
        idx = socket.if_nametoindex(name)
        midx = socket.if_nametoindex(master)
        body = self._ifinfomsg(index=idx) + _attr(IFLA_MASTER, struct.pack("=I", midx))
        self._send(RTM_NEWLINK, NLM_F_REQUEST, body)

    def link_set_up(self, name: str) -> None:
        idx = socket.if_nametoindex(name)
        self._send(RTM_NEWLINK, NLM_F_REQUEST, self._ifinfomsg(index=idx, flags=IFF_UP, change=IFF_UP))

    def link_set_netns_pid(self, name: str, pid: int) -> None:
        idx = socket.if_nametoindex(name)
        body = self._ifinfomsg(index=idx) + _attr(IFLA_NET_NS_PID, struct.pack("=I", pid))
        self._send(RTM_NEWLINK, NLM_F_REQUEST, body)

    def link_set_netns_fd(self, name: str, fd: int) -> None:
        idx = socket.if_nametoindex(name)
        body = self._ifinfomsg(index=idx) + _attr(IFLA_NET_NS_FD, struct.pack("=I", fd))
        self._send(RTM_NEWLINK, NLM_F_REQUEST, body)

    # -- addresses ------------------------------------------------------------
    def addr_add(self, ifname: str, ip: str, prefixlen: int) -> None:
        idx = socket.if_nametoindex(ifname)
        body = struct.pack("=BBBBi", socket.AF_INET, prefixlen, 0, 0, idx)
        packed = socket.inet_aton(ip)
        body += _attr(IFA_LOCAL, packed) + _attr(IFA_ADDRESS, packed)
        self._send(RTM_NEWADDR, NLM_F_REQUEST | NLM_F_CREATE | NLM_F_EXCL, body)

    # -- routes ---------------------------------------------------------------
    def route_add_default(self, gateway: str) -> None:
        body = struct.pack(
            "=BBBBBBBBI",
            socket.AF_INET, 0, 0, 0,
            RT_TABLE_MAIN, RTPROT_BOOT, RT_SCOPE_UNIVERSE, RTN_UNICAST, 0,
        )
        body += _attr(RTA_GATEWAY, socket.inet_aton(gateway))
        self._send(RTM_NEWROUTE, NLM_F_REQUEST | NLM_F_CREATE, body)

    def route_add_net(self, dst: str, prefixlen: int, gateway: Optional[str] = None) -> None:
        body = struct.pack(
            "=BBBBBBBBI",
            socket.AF_INET, prefixlen, 0, 0,
            RT_TABLE_MAIN, RTPROT_BOOT, RT_SCOPE_UNIVERSE, RTN_UNICAST, 0,
        )
        body += _attr(RTA_DST, socket.inet_aton(dst))
        if gateway:
            body += _attr(RTA_GATEWAY, socket.inet_aton(gateway))
        self._send(RTM_NEWROUTE, NLM_F_REQUEST | NLM_F_CREATE, body)


def lo_up() -> None:
    """Bring up loopback in the CURRENT namespace."""
    nl = Rtnl()
    try:
        nl.link_set_up("lo")
    finally:
        nl.close()
