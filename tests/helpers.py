"""Test harness: in-process gateway + embedded worker ("node in a process").

Mirrors the reference's test strategy: fakes for state (our
store IS embeddable), interface seams for pools/GPU info, no live cluster.
"""

from __future__ import annotations

import asyncio
import contextlib
import os
import tempfile
from typing import Optional

from beta9_amd.common.config import AppConfig
from beta9_amd.gateway.gateway import Gateway
from beta9_amd.scheduler.pool import FakePoolController
from beta9_amd.worker.gpu import GPUManager, make_fake_topology
from beta9_amd.worker.repo_client import DirectRepoClient
from beta9_amd.worker.runtime.proc import ProcRuntime
from beta9_amd.worker.worker import Worker


class Cluster:
    """Gateway + N embedded workers in one event loop."""

    def __init__(self, tmpdir: str, gpu_workers: int = 0, fake_gpus: int = 0,
                 db_path: str = ":memory:"):
        cfg = AppConfig().with_root(tmpdir)
        cfg.gateway.grpc_port = 0
        cfg.gateway.http_port = 0
        cfg.database.path = db_path
        self.cfg = cfg
        self.gateway = Gateway(cfg)
        self.workers: list[Worker] = []
        self.tmpdir = tmpdir
        self.fake_gpus = fake_gpus
        self.creds: dict = {}

    async def start(self, n_workers: int = 1, cpu_per_worker: int = 64000) -> "Cluster":
        await self.gateway.start()
        self.gateway.scheduler.register_pool(FakePoolController())
        self.creds = self.gateway.bootstrap_workspace("test")
        repo = DirectRepoClient(self.gateway.worker_repo, self.gateway.container_repo, events=self.gateway.events)
        for i in range(n_workers):
            gpu_manager: Optional[GPUManager] = None
            gpu_type = ""
            if self.fake_gpus:
                topo = make_fake_topology(
                    os.path.join(self.tmpdir, f"topo{i}"), n_gpus=self.fake_gpus
                )
                gpu_manager = GPUManager(topology_path=topo)
                gpu_type = "MI355X"
            w = Worker(
                worker_id=f"w{i}",
                repo=repo,
                cpu_limit=cpu_per_worker,
                memory_limit_mb=131072,
                gpu_type=gpu_type,
                gpu_manager=gpu_manager,
                gateway_host=self.cfg.gateway.host,
                gateway_grpc_port=self.gateway.grpc_port,
                image_store_path=self.cfg.image_service.store_path,
                data_root=self.tmpdir,
            )
            await w.start()
            self.workers.append(w)
        import beta9_amd

        beta9_amd.configure(
            f"{self.cfg.gateway.host}:{self.gateway.grpc_port}", self.creds["token"]
        )
        return self

    async def stop(self) -> None:
        for w in self.workers:
            await w.shutdown()
        await self.gateway.shutdown()

    @property
    def grpc_addr(self) -> str:
        return f"{self.cfg.gateway.host}:{self.gateway.grpc_port}"

    @property
    def token(self) -> str:
        return self.creds["token"]


@contextlib.asynccontextmanager
async def cluster(n_workers: int = 1, fake_gpus: int = 0, db_path: str = ":memory:", **kw):
    with tempfile.TemporaryDirectory() as tmp:
        c = Cluster(tmp, fake_gpus=fake_gpus, db_path=db_path)
        await c.start(n_workers=n_workers, **kw)
        try:
            yield c
        finally:
            await c.stop()
