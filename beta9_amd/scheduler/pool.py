"""Worker pool controllers.

The reference provisions workers as k8s Jobs / cloud machines / BYO agents
(reference: pkg/scheduler/pool_local.go, pool_provider.go, pool_agent.go).
This framework's primary deployment is a single MI355X node, so the local
controller spawns worker processes directly with the same env contract the
reference's pod spec guarantees (WORKER_ID, WORKER_POOL_NAME, CPU_LIMIT,
MEMORY_LIMIT, GPU_TYPE, GPU_COUNT — reference: pool_local.go:168-310), and a
process-pool sizer keeps a warm buffer (reference: pool_sizing.go:44-103).
"""

from __future__ import annotations

import asyncio
import logging
import os
import subprocess
import sys
from typing import Dict, List, Optional

from ..common.config import AppConfig, PoolConfig
from ..types import new_id

log = logging.getLogger(__name__)


class FakePoolController:
    """Test seam (reference tests inject fake controllers —
    pkg/scheduler/scheduler_test.go)."""

    def __init__(self, name: str = "default", gpu_type: str = ""):
        self.name = name
        self.gpu_type = gpu_type
        self.added: List[Dict] = []

    async def add_worker(self, cpu: int, memory_mb: int, gpu_count: int) -> Optional[str]:
        wid = new_id("w-")
        self.added.append({"id": wid, "cpu": cpu, "memory_mb": memory_mb, "gpu_count": gpu_count})
        return wid

    async def shutdown(self) -> None:
        pass


class LocalProcessPoolController:
    """Spawns `python -m beta9_amd.worker` processes on this node."""

    def __init__(
        self,
        cfg: AppConfig,
        pool: PoolConfig,
        gateway_addr: str,
        worker_token: str = "",
        max_workers: int = 8,
        extra_env: Optional[Dict[str, str]] = None,
    ):
        self.cfg = cfg
        self.pool = pool
        self.name = pool.name
        self.gpu_type = pool.gpu_type
        self.gateway_addr = gateway_addr
        self.worker_token = worker_token
        self.max_workers = max_workers
        self.extra_env = extra_env or {}
        self._procs: Dict[str, subprocess.Popen] = {}
        self._lock = asyncio.Lock()

    async def add_worker(self, cpu: int, memory_mb: int, gpu_count: int) -> Optional[str]:
        async with self._lock:
            self._reap()
            if len(self._procs) >= self.max_workers:
                return None
            worker_id = new_id("w-")
            env = dict(os.environ)
            env.update(self.extra_env)
            env.update(
                {
                    "WORKER_ID": worker_id,
                    "WORKER_POOL_NAME": self.pool.name,
                    "WORKER_TOKEN": self.worker_token,
                    "GATEWAY_ADDR": self.gateway_addr,
                    "CPU_LIMIT": str(max(cpu, 1000)),
                    "MEMORY_LIMIT": str(max(memory_mb, 1024)),
                    "GPU_TYPE": self.pool.gpu_type,
                    "GPU_COUNT": str(gpu_count if self.pool.gpu_type else 0),
                    "WORKER_RUNTIME": self.pool.runtime,
                    "B9_IMAGE_STORE": self.cfg.image_service.store_path,
                }
            )
            proc = subprocess.Popen(
                [sys.executable, "-m", "beta9_amd.worker"],
                env=env,
                start_new_session=True,
            )
            self._procs[worker_id] = proc
            log.info("provisioned local worker %s (pid %d)", worker_id, proc.pid)
            return worker_id

    def _reap(self) -> None:
        dead = [wid for wid, p in self._procs.items() if p.poll() is not None]
        for wid in dead:
            del self._procs[wid]

    async def shutdown(self) -> None:
        for wid, p in self._procs.items():
            if p.poll() is None:
                p.terminate()
        for wid, p in self._procs.items():
            try:
                p.wait(timeout=5)
            except subprocess.TimeoutExpired:
                p.kill()
                p.wait(timeout=5)
        self._procs.clear()
