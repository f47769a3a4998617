"""GatewayService + RunnerService.

GatewayService is the SDK/CLI-facing catch-all (reference:
pkg/gateway/gateway.proto:12-444 — auth, object upload, stub get-or-create,
deploy, container list/stop/logs, task list/status). RunnerService is the
in-container loop the runners use (reference: the runner dials gateway gRPC
with BETA9_TOKEN for StartTask/EndTask/args/results — SURVEY.md §A.10;
containers are gateway clients too).
"""

from __future__ import annotations

import hashlib
import os
import time
from typing import Any, AsyncIterator, Dict, Optional

from ...backend.db import BackendDB
from ...backend.repos import ContainerRepo, WorkerRepo
from ...common import keys
from ...common.state import StateStore
from ...rpc.channel import RpcError, invalid, not_found, permission_denied, rpc


def _is_admin(auth: Dict[str, Any]) -> bool:
    return auth.get("token_type") == "cluster_admin"
from ...task.dispatch import Dispatcher
from ...types import StubConfig, StubType, TaskStatus


class GatewayService:
    service_name = "GatewayService"

    def __init__(
        self,
        db: BackendDB,
        store: StateStore,
        dispatcher: Dispatcher,
        container_repo: ContainerRepo,
        worker_repo: WorkerRepo,
        scheduler,
        objects_path: str,
    ):
        self.db = db
        self.store = store
        self.dispatcher = dispatcher
        self.containers = container_repo
        self.workers = worker_repo
        self.scheduler = scheduler
        self.objects_path = objects_path
        os.makedirs(objects_path, exist_ok=True)

    # -- auth ---------------------------------------------------------------
    @rpc(open=True)
    async def Authorize(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        token = (req or {}).get("token", "")
        ctx = self.db.authorize_token(token)
        if ctx is None:
            return {"ok": False}
        return {"ok": True, "workspace_id": ctx["workspace_id"], "workspace_name": ctx["workspace_name"]}

    @rpc()
    async def SignPayload(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """HMAC-sign a payload with the workspace signing key so public
        (authorized=False) endpoints can verify callers (reference:
        auth SignPayload RPC). Signature format: v1=<unix_ts>:<hex hmac of
        ts.method.stub_id.payload> — method + stub bound into the MAC so a
        signature is not replayable against other stubs/methods in the
        workspace within the timestamp window."""
        import hashlib
        import hmac as _hmac

        payload = req.get("payload", b"")
        if isinstance(payload, str):
            payload = payload.encode()
        method = str(req.get("method", "POST")).upper()
        stub_id = str(req.get("stub_id", ""))
        key = self.db.get_or_create_signing_key(auth["workspace_id"]).encode()
        ts = str(int(time.time()))
        msg = ts.encode() + b"." + method.encode() + b"." + stub_id.encode() + b"." + payload
        sig = _hmac.new(key, msg, hashlib.sha256).hexdigest()
        return {"signature": f"v1={ts}:{sig}", "expires_in_s": 300}

    # -- workspaces / tokens (ops surface) -----------------------------------
    @rpc()
    async def CreateWorkspace(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Admin-only: creating tenants is a cluster operation (reference
        gates cross-workspace ops behind ClusterAdmin, pkg/auth/http.go:106)."""
        if not _is_admin(auth):
            raise permission_denied("cluster_admin token required to create workspaces")
        ws = self.db.get_workspace_by_name(req["name"]) or self.db.create_workspace(req["name"])
        return {"workspace_id": ws["id"], "name": req["name"]}

    @rpc()
    async def CreateToken(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        ws_id = req.get("workspace_id") or auth["workspace_id"]
        if ws_id != auth["workspace_id"] and not _is_admin(auth):
            raise permission_denied("cannot mint tokens for another workspace")
        return {"token": self.db.create_token(ws_id), "workspace_id": ws_id}

    @rpc()
    async def RevokeToken(self, req: Dict[str, Any], context, auth) -> bool:
        row = self.db.get_token(req["token"])
        if row is None:
            return True  # already gone / never existed — idempotent
        if row["workspace_id"] != auth["workspace_id"] and not _is_admin(auth):
            raise permission_denied("cannot revoke another workspace's token")
        self.db.revoke_token(req["token"])
        evict = getattr(self, "auth_evict", None)
        if evict is not None:
            evict(req["token"])
        return True

    @rpc()
    async def ListApps(self, req: Dict[str, Any], context, auth) -> list:
        return self.db.list_apps(auth["workspace_id"])

    @rpc()
    async def RolloutDeployment(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Point a deployment name at a specific existing version
        (reference: gateway rollout RPCs)."""
        ws = auth["workspace_id"]
        name, version = req["name"], int(req["version"])
        dep = self.db.get_deployment(ws, name, version=version)
        if dep is None:
            raise not_found(f"deployment {name} v{version}")
        self.db._exec(
            "UPDATE deployments SET active=0 WHERE workspace_id=? AND name=?", (ws, name)
        )
        self.db._exec("UPDATE deployments SET active=1 WHERE id=?", (dep["id"],))
        return {"deployment_id": dep["id"], "version": version, "active": True}

    # -- objects (workspace code sync) ---------------------------------------
    @rpc(kind="stream_unary")
    async def PutObjectStream(self, request_iter, context, auth) -> Dict[str, Any]:
        """Upload workspace code as chunked bytes; content-addressed dedupe
        (reference: gateway.proto PutObjectStream + head object)."""
        ws = auth["workspace_id"]
        h = hashlib.sha256()
        chunks = []
        async for msg in request_iter:
            data = msg.get("data", b"")
            h.update(data)
            chunks.append(data)
        digest = h.hexdigest()
        path = os.path.join(self.objects_path, ws, digest)
        if not os.path.exists(path):
            os.makedirs(os.path.dirname(path), exist_ok=True)
            with open(path + ".tmp", "wb") as f:
                for c in chunks:
                    f.write(c)
            os.rename(path + ".tmp", path)
        size = sum(len(c) for c in chunks)
        obj_id = self.db.create_object(ws, digest, size, path)
        return {"object_id": obj_id, "hash": digest, "size": size}

    @rpc()
    async def HeadObject(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        ws = auth["workspace_id"]
        digest = req.get("hash", "")
        path = os.path.join(self.objects_path, ws, digest)
        if os.path.exists(path):
            obj_id = self.db.create_object(ws, digest, os.path.getsize(path), path)
            return {"exists": True, "object_id": obj_id}
        return {"exists": False}

    # -- stubs / deployments -------------------------------------------------
    @rpc()
    async def GetOrCreateStub(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        ws = auth["workspace_id"]
        name = req.get("name", "")
        stub_type = req.get("stub_type", "")
        try:
            st = StubType(stub_type)
        except ValueError:
            raise invalid(f"unknown stub type {stub_type!r}")
        cfg = StubConfig.model_validate(req.get("config", {}))
        # normalize per type (reference: pkg/gateway/services/stub.go:32-75)
        if st.kind in ("function", "schedule"):
            cfg.keep_warm_seconds = 0
        if st.is_serve:
            cfg.autoscaler.min_containers = 0
        app = self.db.get_or_create_app(ws, req.get("app_name", name or "default"))
        row = self.db.get_or_create_stub(
            ws, name, stub_type, cfg, app_id=app["id"], force_create=bool(req.get("force_create"))
        )
        return {"stub_id": row["id"], "app_id": app["id"]}

    @rpc()
    async def DeployStub(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        ws = auth["workspace_id"]
        stub_id = req["stub_id"]
        name = req.get("name", "")
        stub = self.db.get_stub(stub_id)
        if stub is None:
            raise not_found(f"stub {stub_id}")
        dep = self.db.create_deployment(ws, name, stub_id, stub["type"], app_id=stub["app_id"])
        await self.store.set(f"deploy:announce:{stub_id}", dep)
        # warm the instance now (reference: InstanceController warms min
        # containers at deploy) so the first request pays no instance setup
        kind = stub["type"].split("/")[0]
        try:
            if kind in ("endpoint", "asgi"):
                from ...abstractions.endpoint import get_instance

                inst = await get_instance(self._gateway(), stub_id)
                inst.instance.autoscaler.trigger()
            elif kind == "taskqueue":
                from ...abstractions.taskqueue import get_tq_instance

                inst = await get_tq_instance(self._gateway(), stub_id)
                inst.instance.autoscaler.trigger()
        except Exception:
            pass
        base_kind = stub["type"].split("/")[0]
        if self.scheduler.cfg.gateway.invoke_url_type == "host":
            base = self.scheduler.cfg.gateway.base_domain
            invoke_url = f"http://{req['name']}-v{dep['version']}.app.{base}"
        else:
            invoke_url = f"/{base_kind}/id/{stub_id}"
        return {"deployment_id": dep["id"], "version": dep["version"], "invoke_url": invoke_url}

    def _gateway(self):
        # the service only holds sub-objects; recover the gateway through
        # the scheduler backref set at registration
        return self.scheduler._gateway

    @rpc()
    async def GetStubConfig(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        cfg = self.db.get_stub_config(req["stub_id"])
        if cfg is None:
            raise not_found(f"stub {req['stub_id']}")
        return cfg.model_dump()

    @rpc()
    async def ListDeployments(self, req: Dict[str, Any], context, auth) -> list:
        return self.db.list_deployments(auth["workspace_id"])

    @rpc()
    async def StopDeployment(self, req: Dict[str, Any], context, auth) -> bool:
        self.db.stop_deployment(req["deployment_id"])
        return True

    # -- containers ----------------------------------------------------------
    @rpc()
    async def ListContainers(self, req: Dict[str, Any], context, auth) -> list:
        out = []
        for key in await self.store.keys("container:state:*"):
            d = await self.store.get(key)
            if d and d.get("workspace_id") == auth["workspace_id"]:
                out.append(d)
        return out

    async def _check_container_access(self, container_id: str, auth) -> None:
        """404 unless the container belongs to the caller's workspace (or
        admin) — cross-tenant kill/log-disclosure guard (reference:
        workspace-scoped auth on container RPCs)."""
        if _is_admin(auth):
            return
        st = await self.containers.get_state(container_id)
        if st is None:
            # exited containers: fall back to the task audit rows
            row = self.db._query(
                "SELECT workspace_id FROM tasks WHERE container_id=? LIMIT 1", (container_id,)
            )
            if row and row[0]["workspace_id"] == auth["workspace_id"]:
                return
            raise not_found(f"container {container_id}")
        if st.get("workspace_id") != auth["workspace_id"]:
            raise not_found(f"container {container_id}")

    async def _check_task_access(self, task_id: str, auth) -> None:
        if _is_admin(auth):
            return
        t = await self.dispatcher.get_task(task_id)
        ws = t.workspace_id if t is not None else None
        if ws is None:
            row = self.db.get_task(task_id)
            ws = row.get("workspace_id") if row else None
        if ws != auth["workspace_id"]:
            raise not_found(f"task {task_id}")

    @rpc()
    async def StopContainer(self, req: Dict[str, Any], context, auth) -> bool:
        await self._check_container_access(req["container_id"], auth)
        await self.scheduler.stop(req["container_id"])
        return True

    @rpc()
    async def GetContainerLogs(self, req: Dict[str, Any], context, auth) -> list:
        await self._check_container_access(req["container_id"], auth)
        return await self.containers.get_logs(req["container_id"], tail=int(req.get("tail", 200)))

    @rpc()
    async def GetContainerSpans(self, req: Dict[str, Any], context, auth) -> list:
        await self._check_container_access(req["container_id"], auth)
        return await self.containers.get_spans(req["container_id"])

    @rpc(kind="unary_stream")
    async def StreamContainerLogs(self, req: Dict[str, Any], context, auth) -> AsyncIterator[Dict]:
        """Follow a container's log ring until it exits (reference:
        ContainerStreamLogs)."""
        import asyncio as _asyncio

        cid = req["container_id"]
        await self._check_container_access(cid, auth)
        sent = 0
        idle = 0.0
        while True:
            lines = await self.containers.get_logs(cid, tail=100000)
            for line in lines[sent:]:
                yield {"line": line}
            sent = len(lines)
            st = await self.containers.get_state(cid)
            if st is None:
                code = await self.containers.get_exit_code(cid)
                yield {"exit_code": code if code is not None else -1}
                return
            await _asyncio.sleep(0.2)
            idle += 0.2
            if idle > float(req.get("timeout", 3600.0)):
                return

    # -- tasks ----------------------------------------------------------------
    @rpc()
    async def GetTask(self, req: Dict[str, Any], context, auth) -> Optional[Dict]:
        await self._check_task_access(req["task_id"], auth)
        t = await self.dispatcher.get_task(req["task_id"])
        return t.model_dump() if t else None

    @rpc()
    async def ListTasks(self, req: Dict[str, Any], context, auth) -> list:
        return self.db.list_tasks(auth["workspace_id"], req.get("stub_id", ""), int(req.get("limit", 100)))

    @rpc()
    async def StopTask(self, req: Dict[str, Any], context, auth) -> bool:
        await self._check_task_access(req["task_id"], auth)
        await self.dispatcher.cancel(req["task_id"])
        return True

    @rpc()
    async def GetTaskResult(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        await self._check_task_access(req["task_id"], auth)
        t = await self.dispatcher.get_task(req["task_id"])
        if t is None:
            raise not_found(f"task {req['task_id']}")
        result = await self.dispatcher.get_result(req["task_id"])
        return {"status": t.status.value, "result": result}

    @rpc()
    async def SyncServeWorkspace(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Serve-mode hot reload: push changed files to every running
        container of a stub (reference: ContainerSyncWorkspace,
        worker.proto:19 + sdk sync.py)."""
        from ...rpc.channel import RpcClient

        stub_id = req["stub_id"]
        results = {}
        for cid, addr in (await self.containers.addresses_for_stub(stub_id)).items():
            client = RpcClient(addr)
            try:
                results[cid] = await client.call(
                    "Runner", "SyncFiles", {"files": req.get("files", [])}, timeout=30.0
                )
            except Exception as e:
                results[cid] = {"error": str(e)}
            finally:
                await client.close()
        return results

    # -- workers (ops surface) -------------------------------------------------
    @rpc()
    async def GetMetrics(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        gw = self._gateway()
        return {
            "scheduler": gw.scheduler.metrics.snapshot(),
            "backlog": await gw.scheduler.backlog_len(),
            "registry": gw.metrics.snapshot() if getattr(gw, "metrics", None) else {},
        }

    @rpc()
    async def ListWorkers(self, req: Dict[str, Any], context, auth) -> list:
        return [w.model_dump() for w in await self.workers.list()]

    @rpc()
    async def DrainWorker(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Cordon + wait for the worker's containers to finish, then remove
        (reference: workers cordon-drain surface)."""
        import asyncio as _asyncio

        w = await self.workers.get(req["worker_id"])
        if w is None:
            raise not_found(req["worker_id"])
        w.cordoned = True
        await self.workers.update(w)
        deadline = time.time() + float(req.get("timeout", 60.0))
        while time.time() < deadline:
            w = await self.workers.get(req["worker_id"])
            if w is None or w.active_containers == 0:
                break
            await _asyncio.sleep(0.25)
        remaining = w.active_containers if w else 0
        if req.get("remove", False) and remaining == 0:
            await self.workers.remove(req["worker_id"])
        return {"drained": remaining == 0, "active_containers": remaining}

    @rpc()
    async def CordonWorker(self, req: Dict[str, Any], context, auth) -> bool:
        w = await self.workers.get(req["worker_id"])
        if w is None:
            return False
        w.cordoned = bool(req.get("cordoned", True))
        await self.workers.update(w)
        return True


class RunnerService:
    """In-container runner loop: claim/heartbeat/args/result/end."""

    service_name = "RunnerService"

    def __init__(self, dispatcher: Dispatcher, store: StateStore, db: BackendDB, checkpoint_dir: str = ""):
        self.dispatcher = dispatcher
        self.store = store
        self.db = db
        self.checkpoint_dir = checkpoint_dir

    @rpc()
    async def StartTask(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        ok = await self.dispatcher.claim(req["task_id"], req.get("container_id", ""))
        return {"ok": ok}

    @rpc()
    async def GetTaskArgs(self, req: Dict[str, Any], context, auth) -> Any:
        return await self.store.get(keys.task_args(req["task_id"]))

    @rpc()
    async def Heartbeat(self, req: Dict[str, Any], context, auth) -> bool:
        await self.dispatcher.heartbeat(req["task_id"])
        return True

    @rpc()
    async def IsCancelled(self, req: Dict[str, Any], context, auth) -> bool:
        return await self.dispatcher.is_cancelled(req["task_id"])

    @rpc()
    async def EndTask(self, req: Dict[str, Any], context, auth) -> bool:
        status = TaskStatus(req.get("status", TaskStatus.complete.value))
        await self.dispatcher.complete(req["task_id"], req.get("result"), status=status)
        return True

    @rpc()
    async def SetKeepWarm(self, req: Dict[str, Any], context, auth) -> bool:
        """Refresh the keep-warm TTL key for a serving container."""
        ttl = float(req.get("ttl", 10.0))
        if ttl > 0:
            await self.store.set(
                keys.container_keep_warm(req["stub_id"], req["container_id"]), 1, ttl=ttl
            )
        return True

    @rpc()
    async def ReportPressure(self, req: Dict[str, Any], context, auth) -> bool:
        """LLM serving containers report token pressure (0..N, ~queue depth
        in units of tasks_per_container) for the llm_token_pressure
        autoscaler (reference: pod/llm.go + LLMTokenPressureAutoscaler)."""
        await self.store.set(
            f"llm:pressure:{req['stub_id']}:{req['container_id']}",
            float(req.get("pressure", 0.0)),
            ttl=15.0,
        )
        return True

    @rpc()
    async def CheckpointReady(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Runner signals on_start finished; if the stub has no checkpoint
        yet, THIS container is granted the (single-flight, stub-scoped)
        hipsnap snapshot (reference: READY sentinel + stub-level Redis lock —
        runner/common.py:618, pkg/worker/criu.go:475-510)."""
        stub_id = req["stub_id"]
        ws = auth["workspace_id"]
        await self.store.set(
            f"checkpoint:ready:{stub_id}:{req['container_id']}", time.time()
        )
        if not self.checkpoint_dir:
            return {"take": False}
        if self.db.latest_checkpoint(ws, stub_id) is not None:
            return {"take": False}
        lock_key = f"checkpoint:taking:{stub_id}"
        if not await self.store.set_nx(lock_key, req["container_id"], ttl=600.0):
            return {"take": False}
        key = f"{ws}-{stub_id}"
        return {"take": True, "key": key, "dir": self.checkpoint_dir}

    @rpc()
    async def CommitCheckpoint(self, req: Dict[str, Any], context, auth) -> bool:
        """Record a completed hipsnap snapshot (scheduler attaches it to
        future requests for scale-to-zero resume)."""
        stub_id = req["stub_id"]
        ws = auth["workspace_id"]
        stub = self.db.get_stub(stub_id)
        # durable tier: ingest the snapshot (+ objsnap) into the
        # content-addressed store so ANY node can materialize it over the
        # cache transport (reference: checkpoint upload to cache/S3,
        # pkg/worker/criu.go:320-430)
        cache_digest = obj_digest = ""
        cs = getattr(self, "content_store", None)
        if cs is not None and self.checkpoint_dir:
            import asyncio as _a

            snap = os.path.join(self.checkpoint_dir, f"{req['key']}.hipsnap")
            if os.path.exists(snap):
                cache_digest = await _a.to_thread(cs.add_file, snap)
            if os.path.exists(snap + ".obj"):
                obj_digest = await _a.to_thread(cs.add_file, snap + ".obj")
        self.db.upsert_checkpoint(
            {
                "id": req["key"],
                "workspace_id": ws,
                "stub_id": stub_id,
                "source_container_id": req.get("container_id", ""),
                "status": "available",
                "remote_key": req["key"],
                "stub_type": stub["type"] if stub else "",
                "accelerator": req.get("accelerator", "MI355X"),
                "gpu_count": int(req.get("gpu_count", 1)),
                "size_bytes": int(req.get("size_bytes", 0)),
                "created_at": time.time(),
                "cache_digest": cache_digest,
                "obj_digest": obj_digest,
            }
        )
        await self.store.delete(f"checkpoint:taking:{stub_id}")
        return True
