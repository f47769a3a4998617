"""WorkerRepositoryService — re-exports the embedded state repos to workers
over gRPC so worker processes need no direct store access (reference:
pkg/gateway/services/repository/worker_repo.proto:7-82, worker_repo.go:55-78).
Method names mirror beta9_amd/worker/repo_client.py RemoteRepoClient.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Optional

from ...backend.repos import ContainerRepo, WorkerRepo
from ...rpc.channel import rpc
from ...types import ContainerRequest, ContainerState, ContainerStatus, Worker


class WorkerRepositoryService:
    service_name = "WorkerRepositoryService"

    def __init__(self, worker_repo: WorkerRepo, container_repo: ContainerRepo, events=None,
                 content_store=None, storage_root: str = ""):
        self._w = worker_repo
        self._c = container_repo
        self._events = events
        self._content_store = content_store
        self._storage_root = storage_root
        self._volume_snap_cache: Dict[str, tuple] = {}

    @rpc()
    async def RegisterWorker(self, req: Dict[str, Any], context, auth) -> bool:
        await self._w.register(Worker.model_validate(req))
        return True

    @rpc()
    async def RemoveWorker(self, req: Dict[str, Any], context, auth) -> bool:
        await self._w.remove(req["worker_id"])
        return True

    @rpc()
    async def Keepalive(self, req: Dict[str, Any], context, auth) -> bool:
        await self._w.keepalive(req["worker_id"])
        return True

    @rpc()
    async def GetWorker(self, req: Dict[str, Any], context, auth) -> Optional[Dict]:
        w = await self._w.get(req["worker_id"])
        return w.model_dump() if w else None

    @rpc()
    async def UpdateWorker(self, req: Dict[str, Any], context, auth) -> bool:
        await self._w.update(Worker.model_validate(req))
        return True

    @rpc()
    async def GetNextContainerRequest(self, req: Dict[str, Any], context, auth) -> Optional[Dict]:
        return await self._w.next_container_request(req["worker_id"], float(req.get("timeout", 1.0)))

    @rpc()
    async def AddContainerToWorker(self, req: Dict[str, Any], context, auth) -> bool:
        return await self._w.ack_container(req["worker_id"], req["container_id"], req["token"])

    @rpc()
    async def ReleaseCapacity(self, req: Dict[str, Any], context, auth) -> bool:
        await self._w.release_capacity(
            req["worker_id"], ContainerRequest.model_validate(req["request"])
        )
        return True

    @rpc()
    async def SetContainerState(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.set_state(ContainerState.model_validate(req))
        return True

    @rpc()
    async def UpdateContainerStatus(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.update_status(req["container_id"], ContainerStatus(req["status"]))
        return True

    @rpc()
    async def DeleteContainerState(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.delete_state(req["container_id"])
        return True

    @rpc()
    async def SetContainerAddress(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.set_address(req["container_id"], req["stub_id"], req["addr"])
        return True

    @rpc()
    async def RemoveContainerAddress(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.remove_address(req["container_id"], req["stub_id"])
        return True

    @rpc()
    async def SetExitCode(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.set_exit_code(req["container_id"], int(req["code"]))
        return True

    @rpc()
    async def StopRequested(self, req: Dict[str, Any], context, auth) -> bool:
        return await self._c.stop_requested(req["container_id"])

    @rpc()
    async def SnapshotVolumeContent(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Tar a workspace-storage subtree into the content store so a
        REMOTE worker can materialize read-only volume mounts over the
        cache transport (the offline stand-in for GeeseFS/S3 workspace
        storage — reference: pkg/storage/geese.go; L8). Results cached by
        tree mtime."""
        import asyncio as _a
        import hashlib as _h
        import io as _io
        import tarfile as _tar

        path = os.path.realpath(req["path"])
        root = os.path.realpath(self._storage_root)
        if os.path.commonpath([root, path]) != root or not os.path.isdir(path):
            from ...rpc.channel import invalid

            raise invalid(f"not a workspace-storage dir: {req['path']}")

        def latest_mtime(p: str) -> float:
            mt = os.path.getmtime(p)
            for dp, _dn, fns in os.walk(p):
                for fn in fns:
                    try:
                        mt = max(mt, os.path.getmtime(os.path.join(dp, fn)))
                    except OSError:
                        pass
            return mt

        mt = await _a.to_thread(latest_mtime, path)
        cache_key = _h.sha256(f"{path}:{mt}".encode()).hexdigest()
        cached = self._volume_snap_cache.get(path)
        if cached and cached[0] == cache_key:
            return {"digest": cached[1]}

        def build() -> str:
            buf = _io.BytesIO()
            with _tar.open(fileobj=buf, mode="w") as tf:
                tf.add(path, arcname=".")
            return self._content_store.add_bytes(buf.getvalue())

        if self._content_store is None:
            from ...rpc.channel import invalid

            raise invalid("no content store on this gateway")
        digest = await _a.to_thread(build)
        self._volume_snap_cache[path] = (cache_key, digest)
        return {"digest": digest}

    @rpc(kind="unary_stream")
    async def StreamStopEvents(self, req: Dict[str, Any], context, auth):
        """Keyspace-event-driven container stops (reference: worker
        processStopContainerEvents via Redis keyspace events) — one stream
        per worker instead of 5 Hz polling per container."""
        async for cid in self._c.stop_events():
            yield {"container_id": cid}

    @rpc()
    async def AppendLog(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.append_log(req["container_id"], req["line"])
        return True

    @rpc()
    async def PushSpan(self, req: Dict[str, Any], context, auth) -> bool:
        await self._c.push_span(req["container_id"], req["span"])
        return True

    @rpc()
    async def PushEvent(self, req: Dict[str, Any], context, auth) -> bool:
        """Worker-originated lifecycle events relayed through the gateway
        (reference: pkg/repository/events_worker.go)."""
        if self._events is not None:
            await self._events.emit(req["name"], **(req.get("payload") or {}))
        return True
