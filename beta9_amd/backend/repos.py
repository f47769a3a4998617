"""State repositories over the embedded store.

Worker/container/task state accessors (reference: pkg/repository/
worker_redis.go, container_redis.go, task_redis.go). The atomic scheduling
commit — deduct capacity, bump resource version, write container state, push
the request onto the worker queue — is a single event-loop step here, giving
the same atomicity the reference gets from its Lua script under a per-worker
Redis lock (reference: pkg/repository/worker_redis.go:1473-1570).
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from ..common import keys
from ..common.state import StateStore
from ..types import (
    ContainerRequest,
    ContainerState,
    ContainerStatus,
    Worker,
    WorkerStatus,
    new_id,
)


class WorkerRepo:
    def __init__(self, store: StateStore, keepalive_ttl_s: float = 15.0):
        self.store = store
        self.keepalive_ttl_s = keepalive_ttl_s

    async def register(self, worker: Worker) -> None:
        await self.store.hset(keys.worker_index(), {worker.id: 1})
        await self.store.set(keys.worker_state(worker.id), worker.model_dump())
        await self.keepalive(worker.id)

    async def remove(self, worker_id: str) -> None:
        await self.store.hdel(keys.worker_index(), worker_id)
        await self.store.delete(keys.worker_state(worker_id))
        await self.store.delete(keys.worker_keepalive(worker_id))
        await self.store.delete(keys.worker_request_queue(worker_id))

    async def keepalive(self, worker_id: str) -> None:
        await self.store.set(keys.worker_keepalive(worker_id), time.time(), ttl=self.keepalive_ttl_s)

    async def is_alive(self, worker_id: str) -> bool:
        return await self.store.get(keys.worker_keepalive(worker_id)) is not None

    async def get(self, worker_id: str) -> Optional[Worker]:
        d = await self.store.get(keys.worker_state(worker_id))
        return Worker.model_validate(d) if d else None

    async def update(self, worker: Worker) -> None:
        await self.store.set(keys.worker_state(worker.id), worker.model_dump())

    async def list(self) -> List[Worker]:
        out: List[Worker] = []
        for wid in (await self.store.hgetall(keys.worker_index())).keys():
            w = await self.get(wid)
            if w is not None:
                out.append(w)
        return out

    async def set_status(self, worker_id: str, status: WorkerStatus) -> None:
        w = await self.get(worker_id)
        if w:
            w.status = status
            await self.update(w)

    async def schedule_container_request(self, worker: Worker, request: ContainerRequest) -> bool:
        """Atomic capacity deduction + enqueue (single event-loop step: no
        await between the capacity check and the queue push touches shared
        state that another scheduler step could race on)."""
        cur = await self.get(worker.id)
        if cur is None or cur.resource_version != worker.resource_version:
            return False
        gpu_needed = request.gpu_units()
        if (
            cur.free_cpu < request.cpu
            or cur.free_memory_mb < request.memory_mb
            or cur.free_gpu_count < gpu_needed
        ):
            return False
        cur.free_cpu -= request.cpu
        cur.free_memory_mb -= request.memory_mb
        cur.free_gpu_count -= gpu_needed
        cur.resource_version += 1
        cur.active_containers += 1
        request.delivery_token = new_id("tok-")
        await self.store.set(keys.worker_state(cur.id), cur.model_dump())
        await self.store.hset(
            keys.worker_pending_acks(cur.id),
            {request.container_id: {"token": request.delivery_token, "ts": time.time(), "request": request.model_dump()}},
        )
        await self.store.rpush(keys.worker_request_queue(cur.id), request.model_dump())
        return True

    async def next_container_request(self, worker_id: str, timeout: float) -> Optional[Dict[str, Any]]:
        return await self.store.blpop(keys.worker_request_queue(worker_id), timeout)

    async def ack_container(self, worker_id: str, container_id: str, token: str) -> bool:
        pending = await self.store.hget(keys.worker_pending_acks(worker_id), container_id)
        if pending is None or pending.get("token") != token:
            return False
        await self.store.hdel(keys.worker_pending_acks(worker_id), container_id)
        return True

    async def unacked_requests(self, worker_id: str, older_than_s: float) -> List[ContainerRequest]:
        now = time.time()
        out = []
        for cid, entry in (await self.store.hgetall(keys.worker_pending_acks(worker_id))).items():
            if now - entry.get("ts", 0) > older_than_s:
                out.append(ContainerRequest.model_validate(entry["request"]))
        return out

    async def release_capacity(self, worker_id: str, request: ContainerRequest) -> None:
        cur = await self.get(worker_id)
        if cur is None:
            return
        gpu = request.gpu_units()
        cur.free_cpu = min(cur.total_cpu, cur.free_cpu + request.cpu)
        cur.free_memory_mb = min(cur.total_memory_mb, cur.free_memory_mb + request.memory_mb)
        cur.free_gpu_count = min(cur.total_gpu_count, cur.free_gpu_count + gpu)
        cur.resource_version += 1
        cur.active_containers = max(0, cur.active_containers - 1)
        await self.update(cur)


class ContainerRepo:
    def __init__(self, store: StateStore):
        self.store = store

    async def set_state(self, st: ContainerState) -> None:
        is_new = await self.store.get(keys.container_state(st.container_id)) is None
        await self.store.set(keys.container_state(st.container_id), st.model_dump())
        await self.store.hset(keys.containers_by_stub(st.stub_id), {st.container_id: st.status.value})
        if is_new and st.workspace_id:
            await self.store.hincr("containers:count", st.workspace_id, 1)
            if st.gpu_count > 0:
                await self.store.hincr("containers:gpu_count", st.workspace_id, st.gpu_count)

    async def get_state(self, container_id: str) -> Optional[ContainerState]:
        d = await self.store.get(keys.container_state(container_id))
        return ContainerState.model_validate(d) if d else None

    async def update_status(self, container_id: str, status: ContainerStatus) -> None:
        st = await self.get_state(container_id)
        if st:
            st.status = status
            await self.set_state(st)

    async def delete_state(self, container_id: str) -> None:
        st = await self.get_state(container_id)
        if st:
            await self.store.hdel(keys.containers_by_stub(st.stub_id), container_id)
            if st.workspace_id:
                await self.store.hincr("containers:count", st.workspace_id, -1)
                if st.gpu_count > 0:
                    await self.store.hincr("containers:gpu_count", st.workspace_id, -st.gpu_count)
        await self.store.delete(keys.container_state(container_id))
        await self.store.delete(keys.container_address(container_id))

    async def workspace_counts(self, workspace_id: str) -> Dict[str, int]:
        return {
            "containers": int(await self.store.hget("containers:count", workspace_id) or 0),
            "gpus": int(await self.store.hget("containers:gpu_count", workspace_id) or 0),
        }

    async def containers_for_stub(self, stub_id: str) -> Dict[str, str]:
        return await self.store.hgetall(keys.containers_by_stub(stub_id))

    async def set_address(self, container_id: str, stub_id: str, addr: str) -> None:
        await self.store.set(keys.container_address(container_id), addr)
        await self.store.hset(keys.container_addresses_for_stub(stub_id), {container_id: addr})

    async def get_address(self, container_id: str) -> Optional[str]:
        return await self.store.get(keys.container_address(container_id))

    async def addresses_for_stub(self, stub_id: str) -> Dict[str, str]:
        return await self.store.hgetall(keys.container_addresses_for_stub(stub_id))

    async def remove_address(self, container_id: str, stub_id: str) -> None:
        await self.store.delete(keys.container_address(container_id))
        await self.store.hdel(keys.container_addresses_for_stub(stub_id), container_id)

    async def set_exit_code(self, container_id: str, code: int) -> None:
        await self.store.set(keys.container_exit_code(container_id), code, ttl=600.0)

    async def get_exit_code(self, container_id: str) -> Optional[int]:
        return await self.store.get(keys.container_exit_code(container_id))

    async def request_stop(self, container_id: str) -> None:
        await self.store.set(keys.container_stop_requested(container_id), 1, ttl=120.0)

    async def stop_requested(self, container_id: str) -> bool:
        return await self.store.get(keys.container_stop_requested(container_id)) is not None

    async def stop_events(self):
        """Async iterator of container ids whose stop key was just set —
        the keyspace-event-driven stop path (reference: worker
        processStopContainerEvents via Redis keyspace events; replaces
        per-container 5 Hz polling)."""
        sub = await self.store.subscribe("container:stop:*")
        try:
            while True:
                ev = await sub.get()
                if ev is None:
                    continue
                event, key = ev
                if event == "set":
                    yield key.rsplit(":", 1)[-1]
        finally:
            sub.close()

    async def append_log(self, container_id: str, line: str, cap: int = 2000) -> None:
        key = keys.container_logs(container_id)
        await self.store.rpush(key, line)
        if await self.store.llen(key) > cap:
            await self.store.lpop(key)

    async def get_logs(self, container_id: str, tail: int = 200) -> List[str]:
        lines = await self.store.lrange(keys.container_logs(container_id), 0, -1)
        return lines[-tail:]

    async def push_span(self, container_id: str, span: Dict[str, Any]) -> None:
        await self.store.rpush(keys.container_lifecycle_spans(container_id), span)
        # OTLP-JSON export when configured (reference: OTel traces to Tempo,
        # pkg/common/trace.go)
        from ..common.otlp import get_exporter

        exp = get_exporter()
        if exp is not None:
            try:
                exp.export({"container_id": container_id, **span})
            except Exception:
                pass

    async def get_spans(self, container_id: str) -> List[Dict[str, Any]]:
        return await self.store.lrange(keys.container_lifecycle_spans(container_id), 0, -1)
