"""Scheduler: backlog, worker selection (filter chain -> score -> best-fit),
retries, and pool provisioning.

Reference behavior reproduced (reference: pkg/scheduler/scheduler.go —
Run :391, Stop :506, StartProcessingRequests :629-677,
selectWorkerFromWorkersByStatus :1295-1352, scoreWorkerForRequest :1608,
workerFreeCapacityScore :1634; backlog.go:30-75; reserve.go): requests go to a
priority backlog; a processing loop pops batches, lists workers once per
batch, filters (machine pin -> pool selector -> GPU type -> resources ->
status), scores (pool priority + GPU-preference-order modifier), then picks
the FULLEST worker that fits (best-fit, so idle workers drain to zero), and
commits atomically with capacity deduction. No fit -> provision through the
pool controller, with retry/backoff capped by count and age.

MI355X-specific: GPU normalization maps "any" to the AMD Instinct preference
order, and multi-GPU requests (gpu_count>1) are only placed on workers that
can provide an xGMI-connected device set (all 8 GPUs of one MI355X node are
xGMI-linked, so single-node placement is sufficient; the worker's GPU manager
does the topology-aware ordinal selection, beta9_amd/worker/gpu.py).
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Dict, List, Optional, Protocol

from ..backend.repos import ContainerRepo, WorkerRepo
from ..common import keys
from ..common.config import AppConfig, PoolConfig
from ..common.state import StateStore
from ..types import (
    GPU_PRIORITY_ORDER,
    GPU_TYPE_ANY,
    ContainerRequest,
    ContainerState,
    ContainerStatus,
    Worker,
    WorkerStatus,
)

log = logging.getLogger(__name__)


class PoolController(Protocol):
    name: str
    gpu_type: str

    async def add_worker(self, cpu: int, memory_mb: int, gpu_count: int) -> Optional[str]:
        """Provision a new worker sized for the request; returns worker id."""
        ...

    async def shutdown(self) -> None: ...


class ConcurrencyLimitError(Exception):
    pass


class SchedulerMetrics:
    def __init__(self) -> None:
        self.scheduled = 0
        self.retries = 0
        self.failed = 0
        self.provisioned = 0

    def snapshot(self) -> Dict[str, int]:
        return dict(self.__dict__)


class Scheduler:
    def __init__(
        self,
        cfg: AppConfig,
        store: StateStore,
        worker_repo: WorkerRepo,
        container_repo: ContainerRepo,
    ):
        self.cfg = cfg
        self.store = store
        self.workers = worker_repo
        self.containers = container_repo
        self.pools: Dict[str, PoolController] = {}
        self.metrics = SchedulerMetrics()
        # set by the gateway: (workspace_id, stub_id) -> checkpoint row dict
        self.checkpoint_lookup = None
        # set by the gateway: async (request) -> bool (workspace concurrency
        # limits, reference: CheckConcurrencyLimit scheduler.go:391-505)
        self.concurrency_check = None
        self.events = None  # EventSink
        self.metrics_registry = None  # MetricsRegistry
        self._loop_task: Optional[asyncio.Task] = None
        self._wake = asyncio.Event()
        # provisioning tracker (reference: reserve.go:343-501
        # provisioningTracker): request -> (pending worker id, ts) so retries
        # of an already-provisioned request don't spawn duplicate workers
        self._reservations: Dict[str, tuple] = {}
        self.reservation_ttl_s = 60.0

    def register_pool(self, pool: PoolController) -> None:
        self.pools[pool.name] = pool

    # -- public API ---------------------------------------------------------
    async def run(self, request: ContainerRequest) -> None:
        """Accept a container request: normalize, record state, enqueue."""
        self._normalize_gpu(request)
        if self.concurrency_check is not None and not await self.concurrency_check(request):
            raise ConcurrencyLimitError(
                f"workspace {request.workspace_id} at its concurrency limit"
            )
        if request.checkpoint_enabled and self.checkpoint_lookup is not None:
            cp = self.checkpoint_lookup(request.workspace_id, request.stub_id)
            if cp is not None and (not cp.get("accelerator") or not request.gpu or cp["accelerator"] == request.gpu):
                from ..types import CheckpointState, CheckpointStatus

                request.checkpoint = CheckpointState(
                    checkpoint_id=cp["id"],
                    stub_id=cp["stub_id"],
                    workspace_id=cp["workspace_id"],
                    status=CheckpointStatus.available,
                    remote_key=cp["remote_key"],
                    accelerator=cp.get("accelerator", ""),
                    gpu_count=cp.get("gpu_count", 0),
                    size_bytes=cp.get("size_bytes", 0),
                    cache_digest=cp.get("cache_digest", ""),
                    obj_digest=cp.get("obj_digest", ""),
                )
        st = ContainerState(
            container_id=request.container_id,
            stub_id=request.stub_id,
            workspace_id=request.workspace_id,
            status=ContainerStatus.pending,
            scheduled_at=time.time(),
            gpu=request.gpu,
            gpu_count=request.gpu_count,
        )
        await self.containers.set_state(st)
        await self._enqueue(request, priority=request.timestamp)

    async def stop(self, container_id: str) -> None:
        """Request a container stop; the owning worker reacts via its stop
        event subscription (reference: scheduler.go:506 Stop)."""
        await self.containers.request_stop(container_id)
        await self.containers.update_status(container_id, ContainerStatus.stopping)

    async def start(self) -> None:
        if self._loop_task is None:
            self._loop_task = asyncio.create_task(self._process_loop())

    async def shutdown(self) -> None:
        if self._loop_task is not None:
            self._loop_task.cancel()
            try:
                await self._loop_task
            except asyncio.CancelledError:
                pass
            self._loop_task = None
        for pool in self.pools.values():
            await pool.shutdown()

    # -- internals ----------------------------------------------------------
    def _normalize_gpu(self, request: ContainerRequest) -> None:
        if request.gpu and not request.gpu_request:
            request.gpu_request = [request.gpu]
        if GPU_TYPE_ANY in request.gpu_request:
            request.gpu_request = [g for g in GPU_PRIORITY_ORDER if g != GPU_TYPE_ANY]
        # config-driven failover chains: append ranked alternatives
        for gpu in list(request.gpu_request):
            for alt in self.cfg.scheduler.gpu_failover.get(gpu, []):
                if alt not in request.gpu_request:
                    request.gpu_request.append(alt)
        if request.gpu_request and request.gpu_count == 0:
            request.gpu_count = 1
        if request.gpu_request:
            request.gpu = request.gpu_request[0]

    async def _enqueue(self, request: ContainerRequest, priority: float) -> None:
        await self.store.set(keys.scheduler_request(request.container_id), request.model_dump())
        await self.store.zadd(keys.scheduler_backlog(), request.container_id, priority)
        self._wake.set()

    async def backlog_len(self) -> int:
        return await self.store.zcard(keys.scheduler_backlog())

    async def _process_loop(self) -> None:
        poll = self.cfg.scheduler.backlog_poll_interval_s
        while True:
            try:
                popped = await self.store.zpop_min(
                    keys.scheduler_backlog(), self.cfg.scheduler.batch_size
                )
                if not popped:
                    self._wake.clear()
                    try:
                        await asyncio.wait_for(self._wake.wait(), timeout=max(poll, 0.05))
                    except asyncio.TimeoutError:
                        pass
                    continue
                # entries scored in the future are durable retry-backoffs
                # (survive a gateway crash, unlike an in-memory sleep):
                # push them back and only process what is ready
                now = time.time()
                ready = [(m, s) for m, s in popped if s <= now]
                for m, s in popped:
                    if s > now:
                        await self.store.zadd(keys.scheduler_backlog(), m, s)
                if not ready:
                    await asyncio.sleep(min(poll if poll > 0 else 0.05, 0.05))
                    continue
                popped = ready
                workers = await self._live_workers()
                for container_id, _score in popped:
                    d = await self.store.get(keys.scheduler_request(container_id))
                    if d is None:
                        continue
                    request = ContainerRequest.model_validate(d)
                    if await self.containers.stop_requested(container_id):
                        await self.store.delete(keys.scheduler_request(container_id))
                        continue
                    await self._schedule_one(request, workers)
            except asyncio.CancelledError:
                raise
            except Exception:
                log.exception("scheduler loop error")
                await asyncio.sleep(0.1)

    async def _live_workers(self) -> List[Worker]:
        out = []
        for w in await self.workers.list():
            if w.status == WorkerStatus.disabled or w.cordoned:
                continue
            if w.status == WorkerStatus.available and not await self.workers.is_alive(w.id):
                continue
            out.append(w)
        return out

    async def _schedule_one(self, request: ContainerRequest, workers: List[Worker]) -> None:
        candidates = self._filter(request, [w for w in workers if w.status == WorkerStatus.available])
        candidates.sort(key=lambda w: (-self._score(request, w), -self._fullness(w)))
        for w in candidates:
            if await self.workers.schedule_container_request(w, request):
                await self._committed(request, w)
                return
            # capacity raced; refresh this worker snapshot and continue
            fresh = await self.workers.get(w.id)
            if fresh and await self.workers.schedule_container_request(fresh, request):
                await self._committed(request, fresh)
                return
        await self._no_fit(request, workers)

    async def _committed(self, request: ContainerRequest, worker: Worker) -> None:
        self._reservations.pop(request.container_id, None)
        await self.store.delete(keys.scheduler_request(request.container_id))
        self.metrics.scheduled += 1
        wait_ms = (time.time() - request.timestamp) * 1000.0
        if self.metrics_registry is not None:
            self.metrics_registry.inc("scheduler_containers_scheduled")
            self.metrics_registry.observe("scheduler_schedule_wait_ms", wait_ms)
        if self.events is not None:
            await self.events.emit(
                "container.scheduled",
                container_id=request.container_id,
                worker_id=worker.id,
                stub_id=request.stub_id,
                gpu=request.gpu,
                gpu_count=request.gpu_count,
                retries=request.retry_count,
                wait_ms=round(wait_ms, 2),
            )

    def _filter(self, request: ContainerRequest, workers: List[Worker]) -> List[Worker]:
        """Filter chain (reference order: scheduler.go:1306-1316)."""
        out = []
        for w in workers:
            if request.machine_id and w.machine_id != request.machine_id:
                continue
            if request.pool_selector and w.pool_name != request.pool_selector:
                continue
            if not request.pool_selector and w.requires_pool_selector:
                continue
            if request.requires_gpu():
                if w.gpu not in request.gpu_request and GPU_TYPE_ANY not in request.gpu_request:
                    continue
                if w.free_gpu_count < request.gpu_units():
                    continue
            elif w.gpu and w.total_gpu_count > 0 and request.preemptable is False and w.pool_name != request.pool_selector:
                # keep CPU-only work off GPU workers unless explicitly pooled
                if any(p.gpu_type == "" for p in self.cfg.scheduler.pools):
                    continue
            if w.free_cpu < request.cpu or w.free_memory_mb < request.memory_mb:
                continue
            out.append(w)
        return out

    def _score(self, request: ContainerRequest, w: Worker) -> float:
        """Pool priority + GPU-preference-order modifier (reference:
        scoreWorkerForRequest scheduler.go:1608)."""
        score = float(w.priority) * 100.0
        if request.gpu_request and w.gpu in request.gpu_request:
            score -= request.gpu_request.index(w.gpu) * 10.0
        return score

    def _fullness(self, w: Worker) -> float:
        """Best-fit: prefer the FULLEST worker that fits so idle workers
        drain to zero and spin down (reference: workerFreeCapacityScore
        scheduler.go:1634 — free GPU weighted 1e6 above cpu+mem)."""
        free = w.free_gpu_count * 1e6 + w.free_cpu + w.free_memory_mb
        return -free  # larger = fuller

    async def _no_fit(self, request: ContainerRequest, workers: List[Worker]) -> None:
        # pending worker already provisioning with room? just retry.
        pending_ok = any(
            w.status == WorkerStatus.pending
            and (not request.requires_gpu() or (w.gpu in request.gpu_request and w.free_gpu_count >= request.gpu_units()))
            for w in workers
        )
        res = self._reservations.get(request.container_id)
        if res is not None and time.time() - res[1] > self.reservation_ttl_s:
            self._reservations.pop(request.container_id, None)
            res = None
        if not pending_ok and res is None:
            pool = self._pool_for(request)
            if pool is not None:
                wid = await pool.add_worker(
                    cpu=max(request.cpu, self.cfg.worker.cpu_limit // 4),
                    memory_mb=max(request.memory_mb, self.cfg.worker.memory_limit_mb // 4),
                    gpu_count=request.gpu_count,
                )
                if wid is not None:
                    self.metrics.provisioned += 1
                    self._reservations[request.container_id] = (wid, time.time())
        await self._retry(request)

    def _pool_for(self, request: ContainerRequest) -> Optional[PoolController]:
        if request.pool_selector:
            return self.pools.get(request.pool_selector)
        for pool in self.pools.values():
            if request.requires_gpu():
                if pool.gpu_type and pool.gpu_type in request.gpu_request:
                    return pool
            else:
                if not pool.gpu_type:
                    return pool
        return next(iter(self.pools.values()), None)

    async def _retry(self, request: ContainerRequest) -> None:
        request.retry_count += 1
        age = time.time() - request.timestamp
        if (
            request.retry_count > self.cfg.scheduler.max_retries
            or age > self.cfg.scheduler.max_schedule_age_s
        ):
            log.warning("dropping unschedulable request %s", request.container_id)
            self.metrics.failed += 1
            await self.containers.set_exit_code(request.container_id, 137)
            await self.containers.delete_state(request.container_id)
            await self.store.delete(keys.scheduler_request(request.container_id))
            return
        self.metrics.retries += 1
        backoff = min(
            self.cfg.scheduler.retry_backoff_base_s * (1.5 ** min(request.retry_count, 10)),
            self.cfg.scheduler.retry_backoff_max_s,
        )

        # durable backoff: the entry goes straight back into the persisted
        # backlog with a future ready-time score; the loop skips it until
        # then. A crash mid-backoff no longer loses the request.
        await self._enqueue(request, priority=time.time() + backoff)
