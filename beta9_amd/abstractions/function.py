"""@function abstraction: one container per invocation.

Reference behavior (reference: pkg/abstractions/function/function.go:112-124
FunctionInvoke, task.go:24-112 Execute — create task row, build
ContainerRequest with container id `function-<task>`, scheduler.Run; result
600 s TTL function.go:40-41; runner fetches args / stores result / ends
task). Cron `schedule` stubs reuse this executor.
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, AsyncIterator, Dict

from ..backend.db import BackendDB
from ..common import keys
from ..common.state import StateStore
from ..rpc.channel import not_found, rpc
from ..scheduler.scheduler import Scheduler
from ..task.dispatch import Dispatcher
from ..types import ContainerRequest, StubType, Task, TaskStatus

log = logging.getLogger(__name__)

EXECUTOR = "function"


class FunctionExecutor:
    def __init__(self, db: BackendDB, scheduler: Scheduler, store: StateStore, runner_env_fn):
        self.db = db
        self.scheduler = scheduler
        self.store = store
        self.runner_env_fn = runner_env_fn  # (workspace_id, stub, cfg) -> env dict

    async def execute(self, task: Task, payload: Any) -> None:
        stub = self.db.get_stub(task.stub_id)
        if stub is None:
            raise KeyError(f"stub {task.stub_id} missing")
        cfg = self.db.get_stub_config(task.stub_id)
        env = self.runner_env_fn(task.workspace_id, stub, cfg)
        env.update(
            {
                "TASK_ID": task.task_id,
                "HANDLER": cfg.handler,
                "ON_START": cfg.on_start,
                "TIMEOUT": str(cfg.timeout),
            }
        )
        env.update(cfg.env)
        request = ContainerRequest(
            container_id=f"function-{task.task_id}",
            env=env,
            cpu=cfg.runtime.cpu,
            memory_mb=cfg.runtime.memory_mb,
            gpu=cfg.runtime.gpu,
            gpu_request=list(cfg.runtime.gpus),
            gpu_count=cfg.runtime.gpu_count,
            gpu_fraction=cfg.runtime.gpu_fraction,
            image_id=cfg.runtime.image_id,
            stub_id=task.stub_id,
            workspace_id=task.workspace_id,
            stub_type=stub["type"],
            task_id=task.task_id,
            pool_selector=cfg.pool,
            mounts=self.scheduler._gateway.resolve_mounts(task.workspace_id, cfg.volumes),
            timestamp=time.time(),
        )
        await self.scheduler.run(request)


class FunctionService:
    service_name = "FunctionService"

    def __init__(self, dispatcher: Dispatcher, db: BackendDB, store: StateStore):
        self.dispatcher = dispatcher
        self.db = db
        self.store = store

    @rpc(kind="unary_stream")
    async def FunctionInvoke(self, req: Dict[str, Any], context, auth) -> AsyncIterator[Dict[str, Any]]:
        stub_id = req["stub_id"]
        cfg = self.db.get_stub_config(stub_id)
        if cfg is None:
            raise not_found(f"stub {stub_id}")
        task = await self.dispatcher.send_and_execute(
            EXECUTOR, auth["workspace_id"], stub_id, req.get("args"), policy=cfg.task_policy
        )
        yield {"type": "task", "task_id": task.task_id}
        container_id = f"function-{task.task_id}"
        stream_logs = bool(req.get("stream_logs", True))
        sent_logs = 0
        last_status = None
        deadline = time.monotonic() + cfg.timeout + 60.0
        from ..backend.repos import ContainerRepo

        while time.monotonic() < deadline:
            t = await self.dispatcher.get_task(task.task_id)
            if t is None:
                break
            if stream_logs:
                lines = await self.store.lrange(keys.container_logs(container_id), 0, -1)
                for line in lines[sent_logs:]:
                    yield {"type": "log", "line": line}
                sent_logs = len(lines)
            if t.status != last_status:
                last_status = t.status
                yield {"type": "status", "status": t.status.value}
            if t.status.is_terminal:
                result = await self.dispatcher.get_result(task.task_id)
                yield {"type": "result", "status": t.status.value, "result": result}
                return
            await asyncio.sleep(0.005)
        yield {"type": "result", "status": TaskStatus.timeout.value, "result": None}

    @rpc()
    async def FunctionSchedule(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Register a cron schedule for a stub (minimal: interval seconds;
        reference: function.proto:16 FunctionSchedule with cron strings)."""
        stub_id = req["stub_id"]
        interval = float(req.get("interval_s", 60.0))
        key = f"schedule:{auth['workspace_id']}:{stub_id}"
        await self.store.set(key, {"interval_s": interval, "last": 0.0})
        return {"ok": True}
