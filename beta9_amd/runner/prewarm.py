"""Prewarmed runner shell.

The MI355X cold-start answer to the reference's preallocated network-slot
pool (reference: pkg/worker/network.go:542-624 maintainNetworkSlotPool —
cold start skips netns/veth creation): on a dedicated GPU node the expensive
part of a cold start is not the namespace, it is the Python+torch import and
the HIP context creation. The worker keeps N of these shells per GPU —
spawned with the GPU pinned (HIP_VISIBLE_DEVICES fixed at spawn), torch
imported, HIP context initialized — parked on a unix socket. Starting a
container then costs only: send the spec, import the handler, stream
weights.

Protocol: one JSON line {env: {...}, module: "beta9_amd.runner.endpoint"}
over the socket; the shell replies "ok\n", merges env, and execs the runner
module in-process (it IS the container from then on).
"""

from __future__ import annotations

import json
import os
import runpy
import socket
import sys


def main() -> int:
    sock_path = os.environ["B9_PREWARM_SOCK"]
    # ---- warm the expensive imports -------------------------------------
    import importlib

    import torch  # noqa: F401  (the big one)

    # zygote mode: ALL imports warmed but NO HIP context — the shell stays
    # GPU-agnostic (HIP reads *_VISIBLE_DEVICES at first cuda.init, which
    # happens after activation pins the env). Serves prewarm MISSES for any
    # GPU at import-warm cost instead of full cold python+torch.
    zygote = os.environ.get("B9_PREWARM_ZYGOTE") == "1"

    if not zygote and os.environ.get("WORKER_GPU_DEVICES", "") != "":
        try:
            torch.cuda.init()
            torch.zeros(1, device="cuda:0")  # force full HIP context + pools
            # warm hipBLASLt (library init ~1 s on first GEMM) and our own
            # extension modules
            x = torch.randn(256, 256, dtype=torch.bfloat16, device="cuda:0")
            (x @ x.t()).sum().item()
            try:
                from beta9_amd.ops import gemm as _g

                if _g.has_native():
                    _g.warm_all()  # every kernel's first-launch cost
                from beta9_amd.ops import streams as _streams

                # per-request serving streams: pay HSA hw-queue creation
                # here, not in the first concurrent requests
                _streams.prewarm(8)
            except Exception:
                pass
            # pre-create the HBM arena + pinned staging (hipMalloc slab and
            # hipHostMalloc cost ~0.5 s — off the cold-start path)
            from beta9_amd.ops import hipsnap

            hipsnap.prewarm_arena(device=0)
            torch.cuda.synchronize()
        except Exception:
            pass
    for mod in ("fastapi", "uvicorn", "grpc", "msgpack", "httpx"):
        try:
            importlib.import_module(mod)
        except ImportError:
            pass
    import beta9_amd.runner.common  # noqa: F401
    import beta9_amd.runner.endpoint  # noqa: F401
    # the runner's serving stacks import lazily at start; pay for them here,
    # not inside the measured cold start (grpc ~20 ms, uvicorn/fastapi ~40 ms)
    import beta9_amd.rpc.channel  # noqa: F401  (grpc.aio)
    try:
        import fastapi  # noqa: F401
        import uvicorn  # noqa: F401
        import httpx  # noqa: F401
    except ImportError:
        pass
    import beta9_amd.runner.function  # noqa: F401
    import beta9_amd.runner.sandbox  # noqa: F401
    import beta9_amd.runner.taskqueue  # noqa: F401

    # ---- park on the socket ----------------------------------------------
    try:
        os.unlink(sock_path)
    except FileNotFoundError:
        pass
    srv = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    srv.bind(sock_path)
    srv.listen(1)
    print(f"[prewarm] ready on {sock_path}", flush=True)
    conn, _ = srv.accept()
    data = b""
    while not data.endswith(b"\n"):
        chunk = conn.recv(65536)
        if not chunk:
            return 0
        data += chunk
    spec = json.loads(data.decode())
    if os.environ.get("B9_BOOT_TRACE"):
        import time as _t
        print(f"[boottrace] shell spec received {_t.time():.6f}", flush=True)
    if spec.get("want_netns"):
        # the worker passes a network-namespace fd over the socket
        # (SCM_RIGHTS); join it before any server binds happen
        import ctypes

        _msg, fds, _flags, _addr = socket.recv_fds(conn, 16, 1)
        if fds:
            libc = ctypes.CDLL("libc.so.6", use_errno=True)
            CLONE_NEWNET = 0x40000000
            if libc.setns(fds[0], CLONE_NEWNET) != 0:
                err = ctypes.get_errno()
                conn.sendall(f"err setns: {os.strerror(err)}\n".encode())
                return 1
            os.close(fds[0])
    os.environ.update({str(k): str(v) for k, v in spec.get("env", {}).items()})
    conn.sendall(b"ok\n")
    conn.close()
    srv.close()
    try:
        os.unlink(sock_path)
    except FileNotFoundError:
        pass
    module = spec["module"]
    sys.argv = [module]
    runpy.run_module(module, run_name="__main__")
    return 0


if __name__ == "__main__":
    sys.exit(main())
