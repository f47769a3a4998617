"""Network-namespace slot holder.

A tiny subprocess that unshares a fresh network namespace and then holds it
open (the namespace lives as long as this process). The parent worker
programs the host side (veth/bridge) over rtnetlink and drives the
namespace side through this process's stdin — the holder executes netlink
ops INSIDE the namespace (reference equivalent: network slot pool children,
pkg/worker/network.go:542-624).

Protocol (line-based):
  parent <- "READY <pid>"
  parent -> "up <ifname>"          bring link up
  parent -> "addr <ifname> <ip> <plen>"
  parent -> "route <gw>"           default route
  parent -> "routenet <dst> <plen> [gw]"
  parent <- "OK" | "ERR <msg>"     after each command
"""

from __future__ import annotations

import ctypes
import os
import sys

CLONE_NEWNET = 0x40000000


def main() -> int:
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    # die with the worker (survives the exec into `sleep`): crashed workers
    # must not leak namespace holders
    PR_SET_PDEATHSIG = 1
    libc.prctl(PR_SET_PDEATHSIG, 9, 0, 0, 0)
    if os.getppid() == 1:
        return 1  # worker already gone
    if libc.unshare(CLONE_NEWNET) != 0:
        err = ctypes.get_errno()
        print(f"ERR unshare: {os.strerror(err)}", flush=True)
        return 1
    from beta9_amd.worker.netlink import Rtnl, lo_up

    lo_up()
    print(f"READY {os.getpid()}", flush=True)
    nl = Rtnl()
    for line in sys.stdin:
        parts = line.split()
        if not parts:
            continue
        try:
            if parts[0] == "up":
                nl.link_set_up(parts[1])
            elif parts[0] == "addr":
                nl.addr_add(parts[1], parts[2], int(parts[3]))
            elif parts[0] == "route":
                nl.route_add_default(parts[1])
            elif parts[0] == "routenet":
                nl.route_add_net(parts[1], int(parts[2]), parts[3] if len(parts) > 3 else None)
            elif parts[0] == "quit":
                print("OK", flush=True)
                return 0
            elif parts[0] == "hold":
                # config done: exec into a tiny sleeper (frees the python
                # interpreter; the process — and thus the namespace — lives on)
                print("OK", flush=True)
                sys.stdout.flush()
                for sleeper in ("/bin/sleep", "/usr/bin/sleep"):
                    if os.path.exists(sleeper):
                        os.execv(sleeper, [sleeper, "infinity"])
                import time

                while True:  # no sleep binary: park in python
                    time.sleep(3600)
            else:
                print(f"ERR unknown {parts[0]}", flush=True)
                continue
            print("OK", flush=True)
        except Exception as e:  # noqa: BLE001
            print(f"ERR {e}", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
