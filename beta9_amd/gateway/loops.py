"""Gateway background loops: cron schedules, pool cleaning, pool sizing,
deployment rehydration.

(reference: schedule stubs via FunctionSchedule; pkg/scheduler/
pool_cleaner.go — reap keepalive-expired workers; pool_sizing.go:44-103 —
keep a warm buffer of min free capacity; pkg/abstractions/common/
instance.go:444-640 InstanceController rehydrating deployments at boot.)
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Optional

from ..common import keys
from ..types import WorkerStatus

log = logging.getLogger(__name__)


class CronLoop:
    """Fires schedule-stub tasks on their interval (reference: schedule
    stub type + FunctionSchedule)."""

    def __init__(self, gateway, tick_s: float = 1.0):
        self.gw = gateway
        self.tick_s = tick_s
        self._task: Optional[asyncio.Task] = None

    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass

    async def _loop(self) -> None:
        while True:
            try:
                await asyncio.sleep(self.tick_s)
                now = time.time()
                for key in await self.gw.store.keys("schedule:*"):
                    sched = await self.gw.store.get(key)
                    if not sched:
                        continue
                    if now - sched.get("last", 0) >= sched["interval_s"]:
                        sched["last"] = now
                        await self.gw.store.set(key, sched)
                        _, ws_id, stub_id = key.split(":", 2)
                        try:
                            await self.gw.dispatcher.send_and_execute(
                                "function", ws_id, stub_id, None
                            )
                        except Exception:
                            log.exception("cron dispatch failed for %s", stub_id)
            except asyncio.CancelledError:
                raise
            except Exception:
                log.exception("cron loop error")


class PoolCleaner:
    """Reaps workers whose keepalive expired and re-queues their unacked
    container requests (reference: pool_cleaner.go + delivery-token
    recovery worker_redis.go)."""

    def __init__(self, gateway, interval_s: float = 5.0, ack_timeout_s: float = 30.0):
        self.gw = gateway
        self.interval_s = interval_s
        self.ack_timeout_s = ack_timeout_s
        self._task: Optional[asyncio.Task] = None

    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass

    async def _loop(self) -> None:
        while True:
            try:
                await asyncio.sleep(self.interval_s)
                for w in await self.gw.worker_repo.list():
                    alive = await self.gw.worker_repo.is_alive(w.id)
                    if alive:
                        # recover requests the worker never acked
                        stale = await self.gw.worker_repo.unacked_requests(w.id, self.ack_timeout_s)
                        for req in stale:
                            log.warning("re-queueing unacked request %s", req.container_id)
                            await self.gw.store.hdel(keys.worker_pending_acks(w.id), req.container_id)
                            await self.gw.worker_repo.release_capacity(w.id, req)
                            await self.gw.scheduler.run(req)
                        continue
                    log.warning("reaping dead worker %s", w.id)
                    # re-queue anything still on its queue or unacked
                    while True:
                        d = await self.gw.store.lpop(keys.worker_request_queue(w.id))
                        if d is None:
                            break
                        from ..types import ContainerRequest

                        await self.gw.scheduler.run(ContainerRequest.model_validate(d))
                    for req in await self.gw.worker_repo.unacked_requests(w.id, 0):
                        await self.gw.scheduler.run(req)
                    await self.gw.worker_repo.remove(w.id)
                    if getattr(self.gw, "events", None):
                        await self.gw.events.emit("worker.removed", worker_id=w.id, reason="keepalive_expired")
            except asyncio.CancelledError:
                raise
            except Exception:
                log.exception("pool cleaner error")


class PoolSizer:
    """Keeps a warm worker buffer per pool (reference: pool_sizing.go —
    minFreeCpu/minFreeGpu/minFreeMemory under a sizer lock)."""

    def __init__(self, gateway, interval_s: float = 5.0):
        self.gw = gateway
        self.interval_s = interval_s
        self._task: Optional[asyncio.Task] = None

    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass

    async def _loop(self) -> None:
        while True:
            try:
                await asyncio.sleep(self.interval_s)
                pools = {p.name: p for p in self.gw.cfg.scheduler.pools}
                workers = await self.gw.worker_repo.list()
                for name, pool_cfg in pools.items():
                    if not (pool_cfg.min_free_cpu or pool_cfg.min_free_gpu or pool_cfg.min_free_memory_mb):
                        continue
                    free_cpu = sum(w.free_cpu for w in workers if w.pool_name == name and w.status == WorkerStatus.available)
                    free_gpu = sum(w.free_gpu_count for w in workers if w.pool_name == name and w.status == WorkerStatus.available)
                    free_mem = sum(w.free_memory_mb for w in workers if w.pool_name == name and w.status == WorkerStatus.available)
                    if (
                        free_cpu < pool_cfg.min_free_cpu
                        or free_gpu < pool_cfg.min_free_gpu
                        or free_mem < pool_cfg.min_free_memory_mb
                    ):
                        controller = self.gw.scheduler.pools.get(name)
                        if controller is not None:
                            await controller.add_worker(
                                cpu=self.gw.cfg.worker.cpu_limit,
                                memory_mb=self.gw.cfg.worker.memory_limit_mb,
                                gpu_count=pool_cfg.min_free_gpu,
                            )
            except asyncio.CancelledError:
                raise
            except Exception:
                log.exception("pool sizer error")


class InstanceController:
    """Rehydrates active deployments at gateway boot and warms
    min_containers (reference: instance.go:444-640)."""

    def __init__(self, gateway):
        self.gw = gateway

    async def load(self) -> None:
        from ..abstractions.endpoint import get_instance
        from ..abstractions.taskqueue import get_tq_instance

        for dep in self.gw.db.list_active_deployments():
            kind = dep["stub_type"].split("/")[0]
            try:
                if kind in ("endpoint", "asgi"):
                    inst = await get_instance(self.gw, dep["stub_id"])
                    inst.instance.autoscaler.trigger()
                elif kind == "taskqueue":
                    inst = await get_tq_instance(self.gw, dep["stub_id"])
                    inst.instance.autoscaler.trigger()
            except Exception:
                log.exception("rehydrate failed for deployment %s", dep["id"])
