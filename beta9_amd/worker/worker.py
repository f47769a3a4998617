"""Worker: the node data plane.

Mirrors the reference's worker main loop (reference: pkg/worker/worker.go —
NewWorker :351, Run :663: stream GetNextContainerRequest, ack with delivery
token :749-777, semaphore-bounded concurrent starts :777-816, capacity
keepalive, idle shutdown :1006). One worker process (or embedded instance)
per node; on an MI355X node it owns all 8 GPUs through the GPUManager.
"""

from __future__ import annotations

import asyncio
import logging
import os
from typing import Optional

from ..types import ContainerRequest, Worker as WorkerState, WorkerStatus
from .gpu import GPUManager
from .lifecycle import Lifecycle
from .runtime.base import Runtime
from .runtime.proc import ProcRuntime

log = logging.getLogger(__name__)


class Worker:
    def __init__(
        self,
        worker_id: str,
        repo,  # DirectRepoClient | RemoteRepoClient
        cpu_limit: int = 64000,
        memory_limit_mb: int = 131072,
        gpu_type: str = "",
        gpu_count: int = 0,
        pool_name: str = "default",
        runtime: Optional[Runtime] = None,
        gpu_manager: Optional[GPUManager] = None,
        gateway_host: str = "127.0.0.1",
        gateway_grpc_port: int = 0,
        container_start_limit: int = 32,
        keepalive_interval_s: float = 2.0,
        idle_shutdown_s: float = 0.0,
        machine_id: str = "",
        prewarm: bool = True,
        prewarm_cpu_shells: int = 2,
        prewarm_shells_per_gpu: int = 2,
        image_store_path: str = "",
        worker_host: str = "127.0.0.1",
        isolation: str = "native",
        data_root: str = "",
        network: str = "netns",  # netns | host
        network_slots: int = 4,
    ):
        self.id = worker_id
        self.repo = repo
        self.pool_name = pool_name
        self.runtime = runtime or ProcRuntime()
        if gpu_manager is None and gpu_type:
            gpu_manager = GPUManager()
        self.gpu_manager = gpu_manager
        actual_gpus = gpu_manager.total if gpu_manager else 0
        if gpu_count and gpu_manager:
            actual_gpus = min(gpu_count, gpu_manager.total)
        self.state = WorkerState(
            id=worker_id,
            status=WorkerStatus.available,
            total_cpu=cpu_limit,
            total_memory_mb=memory_limit_mb,
            total_gpu_count=actual_gpus,
            free_cpu=cpu_limit,
            free_memory_mb=memory_limit_mb,
            free_gpu_count=actual_gpus,
            gpu=gpu_type or (gpu_manager.gpu_type if gpu_manager else ""),
            pool_name=pool_name,
            machine_id=machine_id,
        )
        if not image_store_path:
            image_store_path = os.environ.get("B9_IMAGE_STORE", "")
        if not data_root:
            data_root = os.environ.get("B9_DATA_ROOT", "")
        # precise masks: the backend DB (tokens/secrets), tenant workspace
        # storage, and the cache are hidden from containers; volume binds
        # re-expose a container's own volumes AFTER the mask
        mask_paths = (
            [os.path.join(data_root, sub) for sub in ("db", "workspaces", "cache")]
            if data_root
            else []
        )
        from .runtime.native import namespaces_available as _ns_ok

        if isolation == "native" and not _ns_ok():
            import logging as _logging

            _logging.getLogger(__name__).warning(
                "host denies CLONE_NEWNS: b9launch isolation unavailable, "
                "degrading to ProcRuntime with device-env isolation"
            )
        self.prewarm_pool = None
        if prewarm:
            from .prewarm import PrewarmPool
            from .runtime.native import namespaces_available

            ordinals = [d.ordinal for d in gpu_manager.devices] if gpu_manager else []
            if data_root:  # bind sources must exist at shell spawn
                os.makedirs(os.path.join(data_root, "workspaces/objects"), exist_ok=True)
            self.prewarm_pool = PrewarmPool(
                gpu_ordinals=ordinals,
                cpu_shells=prewarm_cpu_shells,
                shells_per_gpu=prewarm_shells_per_gpu,
                isolate=(isolation == "native" and namespaces_available()),
                device_paths=gpu_manager.device_nodes() if gpu_manager else {},
                mask_paths=mask_paths,
                bind_paths=[p for p in (
                    image_store_path,
                    # synced workspace code extracts live here
                    os.path.join(data_root, "workspaces/objects") if data_root else "",
                ) if p],
            )
        self.netplane = None
        if network == "netns" and isolation == "native":
            try:
                from .netplane import NetworkPlane, netplane_available
                from .runtime.native import namespaces_available

                if namespaces_available() and netplane_available():
                    # unique-ish bridge per worker so parallel test
                    # clusters on one host don't collide
                    subnet_octet = (os.getpid() + hash(worker_id)) % 200 + 16
                    self.netplane = NetworkPlane(
                        bridge=f"b9br{subnet_octet}",
                        subnet=f"10.{subnet_octet}.77.0/24",
                        prealloc_slots=network_slots,
                    )
            except Exception:
                self.netplane = None
        self.lifecycle = Lifecycle(
            worker_id=worker_id,
            repo=repo,
            runtime=self.runtime,
            gpu_manager=gpu_manager,
            gateway_host=gateway_host,
            gateway_grpc_port=gateway_grpc_port,
            prewarm=self.prewarm_pool,
            image_store_path=image_store_path,
            worker_host=worker_host,
            isolation=isolation,
            mask_paths=mask_paths,
            netplane=self.netplane,
            data_root=data_root,
        )
        self._sema = asyncio.Semaphore(container_start_limit)
        self._keepalive_interval_s = keepalive_interval_s
        self._idle_shutdown_s = idle_shutdown_s
        self._tasks: set = set()
        self._stopping = asyncio.Event()
        self._run_task: Optional[asyncio.Task] = None
        self._idle_since = None

    async def start(self) -> None:
        await self.repo.register_worker(self.state)
        if self.netplane is not None:
            try:
                await self.netplane.start()
            except Exception:
                import logging

                logging.getLogger(__name__).exception("network plane failed; host networking")
                self.netplane = None
                self.lifecycle.netplane = None
        if self.prewarm_pool is not None:
            await self.prewarm_pool.start()
        self._run_task = asyncio.create_task(self._run())
        ka = asyncio.create_task(self._keepalive_loop())
        self._tasks.add(ka)
        self._tasks.add(asyncio.create_task(self._stop_event_loop()))

    async def _stop_event_loop(self) -> None:
        """ONE stop-event stream per worker (keyspace events) instead of
        per-container polling; reconnects on stream failure."""
        while not self._stopping.is_set():
            try:
                async for cid in self.repo.stream_stop_events():
                    if cid in self.lifecycle.containers:
                        asyncio.create_task(self.lifecycle.stop_container(cid))
            except asyncio.CancelledError:
                return
            except Exception:
                await asyncio.sleep(1.0)  # gateway restart etc: retry

    async def wait_prewarmed(self, timeout: float = 120.0) -> bool:
        """Block until every prewarm shell is parked (bench setup)."""
        if self.prewarm_pool is None:
            return True
        import time as _time

        deadline = _time.monotonic() + timeout
        while _time.monotonic() < deadline:
            shells = [s for lst in self.prewarm_pool._shells.values() for s in lst]
            if shells and all(
                s.proc.returncode is None and os.path.exists(s.sock_path) for s in shells
            ):
                return True
            await asyncio.sleep(0.05)
        return False

    async def _run(self) -> None:
        import time as _time

        self._idle_since = _time.monotonic()
        while not self._stopping.is_set():
            try:
                d = await self.repo.next_container_request(self.id, timeout=1.0)
            except asyncio.CancelledError:
                raise
            except Exception:
                if self._stopping.is_set():
                    return
                log.exception("request stream error; retrying")
                await asyncio.sleep(0.5)
                continue
            if d is None:
                if (
                    self._idle_shutdown_s > 0
                    and not self.lifecycle.containers
                    and _time.monotonic() - self._idle_since > self._idle_shutdown_s
                ):
                    log.info("worker %s idle shutdown", self.id)
                    await self.shutdown()
                    return
                continue
            request = ContainerRequest.model_validate(d)
            ok = await self.repo.ack_container(self.id, request.container_id, request.delivery_token)
            if not ok:
                log.warning("stale delivery token for %s; skipping", request.container_id)
                continue
            self._idle_since = _time.monotonic()
            t = asyncio.create_task(self._handle(request))
            self._tasks.add(t)
            t.add_done_callback(self._tasks.discard)

    async def _handle(self, request: ContainerRequest) -> None:
        async with self._sema:
            try:
                await self.lifecycle.run_container(request)
            finally:
                await self.repo.release_capacity(self.id, request)
                import time as _time

                self._idle_since = _time.monotonic()

    async def _keepalive_loop(self) -> None:
        try:
            while not self._stopping.is_set():
                await self.repo.keepalive(self.id)
                await asyncio.sleep(self._keepalive_interval_s)
        except asyncio.CancelledError:
            pass
        except Exception:
            log.exception("keepalive loop died")

    async def drain(self, timeout: float = 30.0) -> None:
        """Wait for running containers to finish (graceful stop)."""
        deadline = asyncio.get_event_loop().time() + timeout
        while self.lifecycle.containers and asyncio.get_event_loop().time() < deadline:
            await asyncio.sleep(0.1)

    async def shutdown(self) -> None:
        self._stopping.set()
        if self.prewarm_pool is not None:
            await self.prewarm_pool.shutdown()
        for cid in list(self.lifecycle.containers):
            await self.lifecycle.runtime_of(cid).kill(cid, 15)
        await self.drain(timeout=10.0)
        for cid in list(self.lifecycle.containers):
            await self.lifecycle.runtime_of(cid).kill(cid, 9)
        for t in list(self._tasks):
            t.cancel()
        if self._run_task is not None and self._run_task is not asyncio.current_task():
            self._run_task.cancel()
        if self.netplane is not None:
            try:
                await self.netplane.shutdown()
            except Exception:
                pass
        try:
            await self.repo.remove_worker(self.id)
        except Exception:
            pass
