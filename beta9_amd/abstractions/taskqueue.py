"""@task_queue abstraction: async queue with worker containers popping tasks.

Reference behavior (reference: pkg/abstractions/taskqueue/ — taskqueue.proto
Put/Pop/Monitor/Complete/Length/StartServe; runner pops via TaskQueuePop,
autoscaler scales by queue length / tasks_per_container).
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, AsyncIterator, Dict

from ..common import keys
from ..rpc.channel import not_found, rpc
from ..types import ContainerRequest, StubConfig, Task, TaskStatus, new_id
from .common.autoscaler import AutoscaledInstance, deployment_scale_fn

log = logging.getLogger(__name__)


class TaskQueueInstance:
    def __init__(self, gateway, stub_id: str):
        self.gw = gateway
        self.stub_id = stub_id
        self.stub_row = gateway.db.get_stub(stub_id)
        if self.stub_row is None:
            raise KeyError(f"stub {stub_id}")
        self.cfg: StubConfig = gateway.db.get_stub_config(stub_id)
        self.queue_key = keys.taskqueue_list(self.stub_row["workspace_id"], stub_id)
        serve_mode = self.stub_row["type"].endswith("/serve")

        async def sample() -> float:
            return float(await gateway.store.llen(self.queue_key))

        self.instance = AutoscaledInstance(
            gateway,
            stub_id,
            self.cfg,
            self.stub_row,
            self._make_request,
            sample,
            scale_fn=deployment_scale_fn(self.cfg, gateway.cfg.gateway.stub_limits_max_replicas),
            serve_mode=serve_mode,
        )
        self.instance.start()

    def _make_request(self) -> ContainerRequest:
        cfg = self.cfg
        env = self.gw.runner_env(self.stub_row["workspace_id"], self.stub_row, cfg)
        env.update({"HANDLER": cfg.handler, "ON_START": cfg.on_start, "TIMEOUT": str(cfg.timeout)})
        env.update(cfg.env)
        return ContainerRequest(
            container_id=f"taskqueue-{self.stub_id[-8:]}-{new_id()}",
            env=env,
            cpu=cfg.runtime.cpu,
            memory_mb=cfg.runtime.memory_mb,
            gpu=cfg.runtime.gpu,
            gpu_request=list(cfg.runtime.gpus),
            gpu_count=cfg.runtime.gpu_count,
            gpu_fraction=cfg.runtime.gpu_fraction,
            image_id=cfg.runtime.image_id,
            stub_id=self.stub_id,
            workspace_id=self.stub_row["workspace_id"],
            stub_type=self.stub_row["type"],
            pool_selector=cfg.pool,
            mounts=self.gw.resolve_mounts(self.stub_row["workspace_id"], cfg.volumes),
            timestamp=time.time(),
        )

    async def shutdown(self) -> None:
        await self.instance.shutdown()


async def get_tq_instance(gateway, stub_id: str) -> TaskQueueInstance:
    key = f"tq:{stub_id}"
    inst = gateway._instances.get(key)
    if inst is None:
        inst = TaskQueueInstance(gateway, stub_id)
        gateway._instances[key] = inst
    return inst


class TaskQueueExecutor:
    """Executor for dispatched taskqueue tasks: push the task id onto the
    queue list; in-container runners pop and run it."""

    def __init__(self, gateway):
        self.gw = gateway

    async def execute(self, task: Task, payload: Any) -> None:
        inst = await get_tq_instance(self.gw, task.stub_id)
        await self.gw.store.rpush(inst.queue_key, task.task_id)
        inst.instance.autoscaler.trigger()


class TaskQueueService:
    service_name = "TaskQueueService"

    def __init__(self, gateway):
        self.gw = gateway

    @rpc()
    async def TaskQueuePut(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        stub_id = req["stub_id"]
        cfg = self.gw.db.get_stub_config(stub_id)
        if cfg is None:
            raise not_found(f"stub {stub_id}")
        pending = await self.gw.store.llen(
            keys.taskqueue_list(auth["workspace_id"], stub_id)
        )
        if pending >= cfg.max_pending_tasks:
            return {"ok": False, "error": "queue full"}
        task = await self.gw.dispatcher.send_and_execute(
            "taskqueue", auth["workspace_id"], stub_id, req.get("args"), policy=cfg.task_policy
        )
        return {"ok": True, "task_id": task.task_id}

    @rpc()
    async def TaskQueuePop(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Runner-side pop: claim the next task (blocking up to timeout)."""
        stub_id = req["stub_id"]
        container_id = req.get("container_id", "")
        queue_key = keys.taskqueue_list(auth["workspace_id"], stub_id)
        task_id = await self.gw.store.blpop(queue_key, float(req.get("timeout", 1.0)))
        if task_id is None:
            return {"task_id": None}
        if not await self.gw.dispatcher.claim(task_id, container_id):
            return {"task_id": None}
        args = await self.gw.store.get(keys.task_args(task_id))
        return {"task_id": task_id, "args": args}

    @rpc()
    async def TaskQueueComplete(self, req: Dict[str, Any], context, auth) -> bool:
        status = TaskStatus(req.get("status", TaskStatus.complete.value))
        await self.gw.dispatcher.complete(req["task_id"], req.get("result"), status=status)
        return True

    @rpc()
    async def TaskQueueLength(self, req: Dict[str, Any], context, auth) -> int:
        return await self.gw.store.llen(
            keys.taskqueue_list(auth["workspace_id"], req["stub_id"])
        )

    @rpc(kind="unary_stream")
    async def TaskQueueMonitor(self, req: Dict[str, Any], context, auth) -> AsyncIterator[Dict]:
        """Stream cancellation/timeout state for a running task (reference:
        taskqueue monitor stream for cancellation)."""
        task_id = req["task_id"]
        while True:
            cancelled = await self.gw.dispatcher.is_cancelled(task_id)
            t = await self.gw.dispatcher.get_task(task_id)
            yield {
                "cancelled": cancelled,
                "status": t.status.value if t else "unknown",
            }
            if cancelled or (t is not None and t.status.is_terminal):
                return
            await asyncio.sleep(0.5)

    @rpc()
    async def StartTaskQueueServe(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        stub_id = req["stub_id"]
        key = keys.scheduler_serve_lock(auth["workspace_id"], stub_id)
        await self.gw.store.set(key, 1, ttl=float(req.get("timeout", 600.0)))
        inst = await get_tq_instance(self.gw, stub_id)
        inst.instance.autoscaler.trigger()
        return {"ok": True}
