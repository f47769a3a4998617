"""Task dispatcher: executor registry, task lifecycle, claims, retry monitor.

Reference behavior (reference: pkg/task/dispatch.go:22-227): abstractions
register task factories by stub kind; SendAndExecute creates the task row and
invokes the executor; claims and heartbeats live in the state store; a
monitor loop re-dispatches timed-out or heartbeat-dead tasks per TaskPolicy.
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, Awaitable, Callable, Dict, Optional

from ..backend.db import BackendDB
from ..common import keys
from ..common.state import StateStore
from ..types import Task, TaskPolicy, TaskStatus, new_id

log = logging.getLogger(__name__)

# Executor: async fn(task, payload) -> None. It is responsible for moving the
# task to a terminal state (or leaving it for retry).
Executor = Callable[[Task, Any], Awaitable[None]]


class Dispatcher:
    def __init__(self, store: StateStore, db: BackendDB, monitor_interval_s: float = 1.0, heartbeat_ttl_s: float = 60.0):
        self.store = store
        self.db = db
        self.heartbeat_ttl_s = heartbeat_ttl_s
        self._monitor_interval_s = monitor_interval_s
        self._executors: Dict[str, Executor] = {}
        self._monitor_task: Optional[asyncio.Task] = None
        # event-driven completion (hot path: no polling between a runner's
        # EndTask and the caller's wait())
        self._done_events: Dict[str, asyncio.Event] = {}
        self.metrics = None  # MetricsRegistry, wired by the gateway
        # non-terminal task ids the monitor watches: scanning + validating
        # EVERY task:state:* key per tick was O(all tasks ever) and measured
        # quadratic under sustained load (2M model_validate calls in a
        # 1000-step bench). Rebuilt from the store once at start() so
        # restart recovery still re-adopts in-flight tasks.
        self._active: set = set()

    def register(self, executor_name: str, fn: Executor) -> None:
        self._executors[executor_name] = fn

    async def start(self) -> None:
        if self._monitor_task is None:
            # restart recovery: re-adopt non-terminal tasks persisted by a
            # previous gateway (one full scan, at boot only)
            for key in await self.store.keys("task:state:*"):
                d = await self.store.get(key)
                if d and not TaskStatus(d.get("status", "PENDING")).is_terminal:
                    self._active.add(d["task_id"])
            self._monitor_task = asyncio.create_task(self._monitor_loop())

    async def shutdown(self) -> None:
        if self._monitor_task is not None:
            self._monitor_task.cancel()
            try:
                await self._monitor_task
            except asyncio.CancelledError:
                pass
            self._monitor_task = None

    # -- task lifecycle -----------------------------------------------------
    async def send_and_execute(
        self,
        executor_name: str,
        workspace_id: str,
        stub_id: str,
        payload: Any,
        policy: Optional[TaskPolicy] = None,
        task_id: Optional[str] = None,
    ) -> Task:
        task = Task(
            task_id=task_id or new_id("task-"),
            stub_id=stub_id,
            workspace_id=workspace_id,
            status=TaskStatus.pending,
            policy=policy or TaskPolicy(),
            executor=executor_name,
        )
        self.db.create_task_async(task.task_id, workspace_id, stub_id, task.status.value)
        # kv mirrors are the HOT copies; the sqlite row is the durable
        # record. TTL them so sustained traffic doesn't grow the store
        # (and its WAL snapshots) without bound.
        ttl = max(3600.0, (policy.timeout if policy else 3600.0) * 2)
        await self.store.set(keys.task_state(task.task_id), task.model_dump(), ttl=ttl)
        await self.store.set(keys.task_args(task.task_id), payload, ttl=ttl)
        self._active.add(task.task_id)
        await self._execute(task, payload)
        return task

    async def _execute(self, task: Task, payload: Any) -> None:
        fn = self._executors.get(task.executor)
        if fn is None:
            raise KeyError(f"no executor registered for {task.executor!r}")
        await fn(task, payload)

    async def get_task(self, task_id: str) -> Optional[Task]:
        d = await self.store.get(keys.task_state(task_id))
        return Task.model_validate(d) if d else None

    async def update_status(self, task_id: str, status: TaskStatus, container_id: str = "") -> None:
        task = await self.get_task(task_id)
        if task is None:
            return
        task.status = status
        if container_id:
            task.container_id = container_id
        if status == TaskStatus.running and task.started_at == 0:
            task.started_at = time.time()
        if status.is_terminal:
            task.ended_at = time.time()
        # terminal states keep a short kv tail for result polling; running
        # tasks keep the long lease (see send_and_execute)
        ttl = 3600.0 if status.is_terminal else max(3600.0, task.policy.timeout * 2)
        await self.store.set(keys.task_state(task_id), task.model_dump(), ttl=ttl)
        self.db.update_task_async(
            task_id,
            status=status.value,
            container_id=task.container_id,
            started_at=task.started_at,
            ended_at=task.ended_at,
            retries=task.retries,
        )
        if status.is_terminal:
            self._active.discard(task_id)
            ev = self._done_events.pop(task_id, None)
            if ev is not None:
                ev.set()

    async def claim(self, task_id: str, container_id: str) -> bool:
        """A runner claims a task before executing it; the claim plus its
        heartbeat is what the monitor watches (reference: dispatch.go Claim)."""
        if not await self.store.set_nx(keys.task_claim(task_id), container_id, ttl=self.heartbeat_ttl_s):
            return False
        await self.heartbeat(task_id)
        await self.update_status(task_id, TaskStatus.running, container_id=container_id)
        return True

    async def heartbeat(self, task_id: str) -> None:
        await self.store.set(keys.task_heartbeat(task_id), time.time(), ttl=self.heartbeat_ttl_s)
        await self.store.expire(keys.task_claim(task_id), self.heartbeat_ttl_s)

    async def complete(self, task_id: str, result: Any, status: TaskStatus = TaskStatus.complete, result_ttl_s: float = 600.0) -> None:
        await self.store.set(keys.task_result(task_id), result, ttl=result_ttl_s)
        await self.store.delete(keys.task_claim(task_id))
        task = await self.get_task(task_id)
        await self.update_status(task_id, status)
        if task is not None:
            await self._track_usage(task, status)
            await self._fire_callback(task, status, result)

    async def _track_usage(self, task: Task, status: TaskStatus) -> None:
        """Meter the finished task into workspace usage counters; cost is
        derived from the stub's PricingPolicy (reference:
        pkg/abstractions/common/usage.go TrackTaskCost/TrackTaskCount)."""
        from ..common import keys as _keys

        ukey = _keys.workspace_usage(task.workspace_id or "default")
        await self.store.hincr(ukey, "task_count", 1)
        dur_ms = max(0.0, (task.ended_at or time.time()) - (task.started_at or task.created_at)) * 1000.0
        await self.store.hincrbyfloat(ukey, "task_duration_ms", dur_ms)
        cfg = self._stub_config_cached(task.stub_id)
        pricing = getattr(cfg, "pricing", None) if cfg else None
        if pricing is None or status != TaskStatus.complete:
            return
        if pricing.cost_model == "duration":
            cost_cents = pricing.cost_per_task_duration_ms * dur_ms * 100.0
        else:
            cost_cents = pricing.cost_per_task * 100.0
        if cost_cents:
            await self.store.hincrbyfloat(ukey, "cost_cents", cost_cents)
            if self.metrics is not None:
                self.metrics.inc(
                    "task_cost_cents", cost_cents,
                    workspace_id=task.workspace_id, stub_id=task.stub_id,
                )

    async def _fire_callback(self, task: Task, status: TaskStatus, result: Any) -> None:
        """POST the terminal result to the stub's callback_url (reference:
        end_task_and_send_callback, sdk runner/common.py)."""
        cfg = self._stub_config_cached(task.stub_id)
        url = getattr(cfg, "callback_url", "") if cfg else ""
        if not url:
            return

        async def post() -> None:
            try:
                import httpx

                async with httpx.AsyncClient(timeout=10.0) as client:
                    await client.post(
                        url,
                        json={
                            "task_id": task.task_id,
                            "stub_id": task.stub_id,
                            "status": status.value,
                            "result": result,
                        },
                    )
            except Exception:
                log.warning("callback to %s failed", url, exc_info=True)

        asyncio.create_task(post())

    def _stub_config_cached(self, stub_id: str):
        cache = getattr(self, "_cfg_cache", None)
        if cache is None:
            cache = self._cfg_cache = {}
        if stub_id not in cache:
            try:
                cache[stub_id] = self.db.get_stub_config(stub_id)
            except Exception:
                cache[stub_id] = None
        return cache[stub_id]

    async def get_result(self, task_id: str) -> Any:
        return await self.store.get(keys.task_result(task_id))

    async def cancel(self, task_id: str) -> None:
        await self.store.set(keys.task_cancel(task_id), 1, ttl=600.0)
        await self.update_status(task_id, TaskStatus.cancelled)

    async def is_cancelled(self, task_id: str) -> bool:
        return await self.store.get(keys.task_cancel(task_id)) is not None

    async def wait(self, task_id: str, timeout: float = 300.0, poll: float = 0.005) -> Task:
        task = await self.get_task(task_id)
        if task is not None and task.status.is_terminal:
            return task
        ev = self._done_events.get(task_id)
        if ev is None:
            ev = self._done_events[task_id] = asyncio.Event()
            # re-check after publishing the event (complete() may have run
            # between the status read and the event registration)
            task = await self.get_task(task_id)
            if task is not None and task.status.is_terminal:
                self._done_events.pop(task_id, None)
                return task
        try:
            await asyncio.wait_for(ev.wait(), timeout=timeout)
        except asyncio.TimeoutError:
            self._done_events.pop(task_id, None)
            raise TimeoutError(f"task {task_id} did not finish in {timeout}s")
        task = await self.get_task(task_id)
        if task is None:
            raise TimeoutError(f"task {task_id} vanished")
        return task

    # -- retry monitor ------------------------------------------------------
    async def _monitor_loop(self) -> None:
        """Re-dispatch tasks whose runner died (claimed but heartbeat
        expired) or that out-ran their timeout (reference:
        dispatch.go:177-227 monitorTasks)."""
        while True:
            try:
                await asyncio.sleep(self._monitor_interval_s)
                for task_id in list(self._active):
                    d = await self.store.get(keys.task_state(task_id))
                    if not d:
                        self._active.discard(task_id)
                        continue
                    task = Task.model_validate(d)
                    if task.status.is_terminal:
                        self._active.discard(task_id)
                        continue
                    now = time.time()
                    timed_out = (
                        task.policy.timeout > 0
                        and task.started_at > 0
                        and now - task.started_at > task.policy.timeout
                    )
                    hb_dead = (
                        task.status == TaskStatus.running
                        and await self.store.get(keys.task_heartbeat(task.task_id)) is None
                    )
                    if timed_out:
                        await self.complete(task.task_id, None, status=TaskStatus.timeout)
                    elif hb_dead:
                        await self._retry(task)
            except asyncio.CancelledError:
                raise
            except Exception:
                log.exception("task monitor error")

    async def _retry(self, task: Task) -> None:
        if task.retries >= task.policy.max_retries:
            await self.complete(task.task_id, None, status=TaskStatus.error)
            return
        task.retries += 1
        task.status = TaskStatus.retry
        await self.store.delete(keys.task_claim(task.task_id))
        await self.store.set(
            keys.task_state(task.task_id), task.model_dump(),
            ttl=max(3600.0, task.policy.timeout * 2),
        )
        payload = await self.store.get(keys.task_args(task.task_id))
        log.info("retrying task %s (attempt %d)", task.task_id, task.retries)
        await self._execute(task, payload)
