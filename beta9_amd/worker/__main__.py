"""Standalone worker process entrypoint (env contract: WORKER_ID,
WORKER_POOL_NAME, WORKER_TOKEN, GATEWAY_ADDR, CPU_LIMIT, MEMORY_LIMIT,
GPU_TYPE, GPU_COUNT — reference: cmd/worker/main.go + worker.go:361-394)."""

from __future__ import annotations

import asyncio
import logging
import os
import signal

from ..types import new_id
from .gpu import GPUManager
from .repo_client import RemoteRepoClient
from .runtime.proc import ProcRuntime
from .worker import Worker


def _select_runtime(name: str):
    if name == "native":
        from .runtime.native import NativeRuntime, namespaces_available

        if namespaces_available():
            return NativeRuntime()
    return ProcRuntime()


async def amain() -> int:
    logging.basicConfig(level=logging.INFO)
    e = os.environ
    worker_id = e.get("WORKER_ID") or new_id("w-")
    gateway_addr = e.get("GATEWAY_ADDR", "127.0.0.1:1993")
    host, _, port = gateway_addr.partition(":")
    repo = RemoteRepoClient(gateway_addr, token=e.get("WORKER_TOKEN", ""))
    gpu_type = e.get("GPU_TYPE", "")
    gpu_manager = GPUManager() if gpu_type else None
    if gpu_manager is not None and gpu_manager.total == 0:
        gpu_manager = None
        gpu_type = ""
    worker = Worker(
        worker_id=worker_id,
        repo=repo,
        cpu_limit=int(e.get("CPU_LIMIT", "64000")),
        memory_limit_mb=int(e.get("MEMORY_LIMIT", "131072")),
        gpu_type=gpu_type,
        gpu_count=int(e.get("GPU_COUNT", "0") or 0),
        pool_name=e.get("WORKER_POOL_NAME", "default"),
        runtime=_select_runtime(e.get("WORKER_RUNTIME", "proc")),
        gpu_manager=gpu_manager,
        gateway_host=host,
        gateway_grpc_port=int(port or 1993),
        idle_shutdown_s=float(e.get("IDLE_SHUTDOWN_S", "0") or 0),
        machine_id=e.get("MACHINE_ID", ""),
        # multi-node: the address other nodes reach this worker's containers
        # at (container addresses are published as WORKER_HOST:port)
        worker_host=e.get("WORKER_HOST", "127.0.0.1"),
    )
    stop = asyncio.Event()
    loop = asyncio.get_event_loop()
    for sig in (signal.SIGTERM, signal.SIGINT):
        loop.add_signal_handler(sig, stop.set)
    await worker.start()
    await stop.wait()
    await worker.shutdown()
    await repo.close()
    return 0


if __name__ == "__main__":
    raise SystemExit(asyncio.run(amain()))
