"""Worker-side repository interfaces.

Workers never touch the state store directly — they go through the gateway's
repository services (reference: pkg/gateway/services/repository/
worker_repo.proto — the reference re-exports its Redis repos over gRPC so
workers need no Redis; same shape here over msgpack-gRPC). Two impls:

- DirectRepoClient: in-process (gateway and worker share an event loop —
  tests, single-process bench).
- RemoteRepoClient: gRPC to the gateway (spawned worker processes).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..backend.repos import ContainerRepo, WorkerRepo
from ..rpc.channel import RpcClient
from ..types import ContainerRequest, ContainerState, ContainerStatus, Worker


class DirectRepoClient:
    def __init__(self, worker_repo: WorkerRepo, container_repo: ContainerRepo, events=None):
        self._w = worker_repo
        self._c = container_repo
        self._events = events

    # worker state
    async def register_worker(self, worker: Worker) -> None:
        await self._w.register(worker)

    async def remove_worker(self, worker_id: str) -> None:
        await self._w.remove(worker_id)

    async def keepalive(self, worker_id: str) -> None:
        await self._w.keepalive(worker_id)

    async def get_worker(self, worker_id: str) -> Optional[Worker]:
        return await self._w.get(worker_id)

    async def update_worker(self, worker: Worker) -> None:
        await self._w.update(worker)

    async def next_container_request(self, worker_id: str, timeout: float) -> Optional[Dict[str, Any]]:
        return await self._w.next_container_request(worker_id, timeout)

    async def ack_container(self, worker_id: str, container_id: str, token: str) -> bool:
        return await self._w.ack_container(worker_id, container_id, token)

    async def release_capacity(self, worker_id: str, request: ContainerRequest) -> None:
        await self._w.release_capacity(worker_id, request)

    # container state
    async def set_container_state(self, st: ContainerState) -> None:
        await self._c.set_state(st)

    async def update_container_status(self, container_id: str, status: str) -> None:
        await self._c.update_status(container_id, ContainerStatus(status))

    async def delete_container_state(self, container_id: str) -> None:
        await self._c.delete_state(container_id)

    async def set_container_address(self, container_id: str, stub_id: str, addr: str) -> None:
        await self._c.set_address(container_id, stub_id, addr)

    async def remove_container_address(self, container_id: str, stub_id: str) -> None:
        await self._c.remove_address(container_id, stub_id)

    async def set_exit_code(self, container_id: str, code: int) -> None:
        await self._c.set_exit_code(container_id, code)

    async def stop_requested(self, container_id: str) -> bool:
        return await self._c.stop_requested(container_id)

    def stream_stop_events(self):
        return self._c.stop_events()

    async def append_log(self, container_id: str, line: str) -> None:
        await self._c.append_log(container_id, line)

    async def push_span(self, container_id: str, span: Dict[str, Any]) -> None:
        await self._c.push_span(container_id, span)

    async def push_event(self, name: str, **payload: Any) -> None:
        if self._events is not None:
            await self._events.emit(name, **payload)


SVC = "WorkerRepositoryService"


class RemoteRepoClient:
    """gRPC-backed version of the same interface for worker processes."""

    def __init__(self, gateway_addr: str, token: str = ""):
        self._rpc = RpcClient(gateway_addr, token=token)

    async def close(self) -> None:
        await self._rpc.close()

    async def register_worker(self, worker: Worker) -> None:
        await self._rpc.call(SVC, "RegisterWorker", worker.model_dump())

    async def remove_worker(self, worker_id: str) -> None:
        await self._rpc.call(SVC, "RemoveWorker", {"worker_id": worker_id})

    async def keepalive(self, worker_id: str) -> None:
        await self._rpc.call(SVC, "Keepalive", {"worker_id": worker_id})

    async def get_worker(self, worker_id: str) -> Optional[Worker]:
        d = await self._rpc.call(SVC, "GetWorker", {"worker_id": worker_id})
        return Worker.model_validate(d) if d else None

    async def update_worker(self, worker: Worker) -> None:
        await self._rpc.call(SVC, "UpdateWorker", worker.model_dump())

    async def next_container_request(self, worker_id: str, timeout: float) -> Optional[Dict[str, Any]]:
        return await self._rpc.call(
            SVC, "GetNextContainerRequest", {"worker_id": worker_id, "timeout": timeout},
            timeout=timeout + 10.0,
        )

    async def ack_container(self, worker_id: str, container_id: str, token: str) -> bool:
        return bool(
            await self._rpc.call(
                SVC, "AddContainerToWorker",
                {"worker_id": worker_id, "container_id": container_id, "token": token},
            )
        )

    async def release_capacity(self, worker_id: str, request: ContainerRequest) -> None:
        await self._rpc.call(SVC, "ReleaseCapacity", {"worker_id": worker_id, "request": request.model_dump()})

    async def set_container_state(self, st: ContainerState) -> None:
        await self._rpc.call(SVC, "SetContainerState", st.model_dump())

    async def update_container_status(self, container_id: str, status: str) -> None:
        await self._rpc.call(SVC, "UpdateContainerStatus", {"container_id": container_id, "status": status})

    async def delete_container_state(self, container_id: str) -> None:
        await self._rpc.call(SVC, "DeleteContainerState", {"container_id": container_id})

    async def set_container_address(self, container_id: str, stub_id: str, addr: str) -> None:
        await self._rpc.call(
            SVC, "SetContainerAddress", {"container_id": container_id, "stub_id": stub_id, "addr": addr}
        )

    async def remove_container_address(self, container_id: str, stub_id: str) -> None:
        await self._rpc.call(SVC, "RemoveContainerAddress", {"container_id": container_id, "stub_id": stub_id})

    async def set_exit_code(self, container_id: str, code: int) -> None:
        await self._rpc.call(SVC, "SetExitCode", {"container_id": container_id, "code": code})

    async def stop_requested(self, container_id: str) -> bool:
        return bool(await self._rpc.call(SVC, "StopRequested", {"container_id": container_id}))

    async def stream_stop_events(self):
        async for msg in self._rpc.stream(SVC, "StreamStopEvents", {}):
            cid = msg.get("container_id") if isinstance(msg, dict) else None
            if cid:
                yield cid

    async def append_log(self, container_id: str, line: str) -> None:
        await self._rpc.call(SVC, "AppendLog", {"container_id": container_id, "line": line})

    async def push_span(self, container_id: str, span: Dict[str, Any]) -> None:
        await self._rpc.call(SVC, "PushSpan", {"container_id": container_id, "span": span})

    async def push_event(self, name: str, **payload: Any) -> None:
        await self._rpc.call(SVC, "PushEvent", {"name": name, "payload": payload})
