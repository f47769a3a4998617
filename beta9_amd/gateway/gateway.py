"""Gateway: control-plane boot.

Boots config -> embedded state store -> backend DB -> repos -> scheduler ->
task dispatcher -> abstraction services -> gRPC server (+ HTTP API, http.py)
(reference: pkg/gateway/gateway.go — NewGateway :105, registerServices :376,
Start :618). Single process per deployment; workers connect over gRPC (or
embed in-process for tests/bench).
"""

from __future__ import annotations

import asyncio
import logging
import os
import time
from typing import Dict, List, Optional, Tuple

from ..backend.db import BackendDB
from ..backend.repos import ContainerRepo, WorkerRepo
from ..common.config import AppConfig, load_config
from ..common.state import StateStore
from ..rpc.channel import RpcServer
from ..scheduler.scheduler import Scheduler
from ..task.dispatch import Dispatcher
from ..types import StubConfig
from .services.core import GatewayService, RunnerService
from .services.repository import WorkerRepositoryService

log = logging.getLogger(__name__)


class Gateway:
    def __init__(self, cfg: Optional[AppConfig] = None):
        self.cfg = cfg or load_config()
        if self.cfg.state.persist_path:
            from ..common.persist import PersistentStateStore

            self.store = PersistentStateStore(self.cfg.state.persist_path)
        else:
            self.store = StateStore()
        self.db = BackendDB(self.cfg.database.path)
        self.worker_repo = WorkerRepo(self.store, keepalive_ttl_s=self.cfg.worker.keepalive_ttl_s)
        self.container_repo = ContainerRepo(self.store)
        self.dispatcher = Dispatcher(
            self.store,
            self.db,
            monitor_interval_s=self.cfg.task.monitor_interval_s,
            heartbeat_ttl_s=self.cfg.task.heartbeat_ttl_s,
        )
        self.scheduler = Scheduler(self.cfg, self.store, self.worker_repo, self.container_repo)
        self.scheduler._gateway = self  # backref for services needing the gateway
        self.rpc_server = RpcServer(
            host=self.cfg.gateway.host, port=self.cfg.gateway.grpc_port, auth_fn=self._auth
        )
        self.grpc_port: int = 0
        self.http_port: int = 0
        self._http_server = None
        self._services: List = []
        self._instances = {}  # stub_id -> autoscaled instance (endpoint/taskqueue)
        self._admin_token: str = ""
        self._runner_tokens: Dict[str, str] = {}  # workspace_id -> token
        self._started = False
        self.draining = False  # reference: drainMiddleware + 10 s propagation
        self._auth_cache: Dict[str, Tuple[Dict, float]] = {}

    # -- auth ----------------------------------------------------------------
    async def _auth(self, token: str, full_method: str):
        # per-token TTL cache: authorize_token is a blocking sqlite query
        # and this runs on EVERY rpc (reference: workspace auth cache,
        # pkg/repository — token lookups cached in Redis). Revocation
        # invalidates via _auth_evict.
        now = time.monotonic()
        hit = self._auth_cache.get(token)
        if hit is not None and hit[1] > now:
            return hit[0]
        ctx = self.db.authorize_token(token)
        if ctx is None:
            return None
        out = {
            "workspace_id": ctx["workspace_id"],
            "workspace_name": ctx["workspace_name"],
            "token": token,
            "token_type": ctx.get("token_type", "workspace"),
        }
        if len(self._auth_cache) > 4096:
            self._auth_cache.clear()
        self._auth_cache[token] = (out, now + 60.0)
        return out

    def _auth_evict(self, token: str) -> None:
        self._auth_cache.pop(token, None)

    def bootstrap_workspace(self, name: str = "default") -> Dict[str, str]:
        """Create (or fetch) a workspace + cluster-admin token — first-boot
        convenience (reference: initial ClusterAdmin token,
        pkg/auth/http.go:106)."""
        ws = self.db.get_workspace_by_name(name)
        if ws is None:
            ws = self.db.create_workspace(name)
        token = self.db.create_token(ws["id"], token_type="cluster_admin")
        return {"workspace_id": ws["id"], "workspace_name": name, "token": token}

    def runner_token(self, workspace_id: str) -> str:
        tok = self._runner_tokens.get(workspace_id)
        if tok is None:
            tok = self.db.create_token(workspace_id)
            self._runner_tokens[workspace_id] = tok
        return tok

    def runner_env(self, workspace_id: str, stub: Dict, cfg: StubConfig) -> Dict[str, str]:
        """Base env contract for in-container runners (reference env list:
        SURVEY.md §A.10 / pkg/abstractions/endpoint/instance.go:51-151)."""
        env = {
            "B9_TOKEN": self.runner_token(workspace_id),
            "B9_GATEWAY_HOST": self.cfg.gateway.host,
            "B9_GATEWAY_PORT": str(self.grpc_port),
            "STUB_ID": stub["id"],
            "STUB_TYPE": stub["type"],
            "WORKSPACE_ID": workspace_id,
            "WORKERS": str(cfg.workers),
            "KEEP_WARM_SECONDS": str(cfg.keep_warm_seconds),
            "CONCURRENT_REQUESTS": str(cfg.concurrent_requests),
            "CHECKPOINT_ENABLED": "1" if cfg.checkpoint_enabled else "0",
            "B9_CHECKPOINT_DIR": self.cfg.checkpoint.store_path,
            "CALLBACK_URL": cfg.callback_url,
            "PYTHONUNBUFFERED": "1",
        }
        obj_id = (cfg.extra or {}).get("object_id")
        if obj_id:
            obj = self.db.get_object(obj_id)
            if obj:
                env["B9_WORKSPACE_DIR"] = self._extracted_dir(obj)
        for name in cfg.secrets:
            val = self.db.get_secret(workspace_id, name)
            if val is not None:
                env[name] = val
        return env

    def resolve_mounts(self, workspace_id: str, mounts) -> list:
        """Volume mounts name host-relative volumes in the SDK; resolve them
        to absolute host paths for the worker (reference: mounts from
        volumes, pkg/abstractions/common/mount.go)."""
        out = []
        for m in mounts:
            m = m.model_copy() if hasattr(m, "model_copy") else m
            local = m.local_path
            if local and not os.path.isabs(local):
                path = os.path.join(self.cfg.storage.root, workspace_id, "volumes", local)
                os.makedirs(path, exist_ok=True)
                m.local_path = path
            out.append(m)
        return out

    def _extracted_dir(self, obj: Dict) -> str:
        """Extract an uploaded workspace zip once, content-addressed."""
        import zipfile

        target = os.path.join(self.cfg.storage.root, "extracted", obj["hash"])
        if not os.path.isdir(target):
            os.makedirs(target + ".tmp", exist_ok=True)
            try:
                with zipfile.ZipFile(obj["path"]) as z:
                    z.extractall(target + ".tmp")
                os.rename(target + ".tmp", target)
            except Exception:
                log.exception("failed to extract object %s", obj["id"])
                return ""
        return target

    # -- boot ------------------------------------------------------------------
    async def start(self) -> None:
        if self._started:
            return
        os.makedirs(self.cfg.storage.root, exist_ok=True)
        await self.store.start()
        from ..common.events import EventSink
        from ..common.metrics import REGISTRY

        self.events = EventSink(self.store, jsonl_path=self.cfg.monitoring.span_sink_path)
        self.metrics = REGISTRY
        self.scheduler.events = self.events
        self.scheduler.metrics_registry = self.metrics
        self.dispatcher.metrics = self.metrics
        await self.dispatcher.start()
        await self.scheduler.start()
        self._register_services()
        self.grpc_port = await self.rpc_server.start()
        if self.cfg.cache.enabled:
            from ..cache.transport import RawTransportServer

            self.cache_transport = RawTransportServer(
                self.content_store,
                host=self.cfg.gateway.host,
                inflight_bytes_cap=self.cfg.cache.inflight_bytes_cap,
            )
            self.cache_service.transport_port = await self.cache_transport.start()
            # this node's reconciler: host-map lease + peer pulls for missing
            # content (multi-node worker caches run the same class against
            # the gateway coordinator)
            from ..cache.reconciler import CacheReconciler

            self.cache_reconciler = CacheReconciler(
                self.content_store,
                self.cache_coordinator,
                host_id=f"gateway-{os.getpid()}",
                addr=self.cfg.gateway.host,
                transport_port=self.cache_service.transport_port,
            )
            await self.cache_reconciler.start()
        await self._start_http()
        from .loops import CronLoop, InstanceController, PoolCleaner, PoolSizer

        self._loops = [CronLoop(self), PoolCleaner(self), PoolSizer(self)]
        for loop in self._loops:
            loop.start()
        await InstanceController(self).load()
        self._started = True
        log.info("gateway up: grpc=%d http=%d", self.grpc_port, self.http_port)

    def _register_services(self) -> None:
        from ..abstractions.endpoint import EndpointExecutor, EndpointService
        from ..abstractions.function import EXECUTOR as FN_EXEC
        from ..abstractions.function import FunctionExecutor, FunctionService
        from ..abstractions.map_queue import MapService, SignalService, SimpleQueueService
        from ..abstractions.secret import SecretService
        from ..abstractions.taskqueue import TaskQueueExecutor, TaskQueueService
        from ..abstractions.volume import VolumeService
        from ..abstractions.output import OutputService
        from ..abstractions.bot import BotService
        from ..abstractions.pod import PodService
        from ..abstractions.image import ImageService

        gw_svc = GatewayService(
            self.db,
            self.store,
            self.dispatcher,
            self.container_repo,
            self.worker_repo,
            self.scheduler,
            objects_path=os.path.join(self.cfg.storage.root, "objects"),
        )
        gw_svc.auth_evict = self._auth_evict  # revocation invalidates cache
        os.makedirs(self.cfg.checkpoint.store_path, exist_ok=True)
        runner_svc = RunnerService(
            self.dispatcher, self.store, self.db,
            checkpoint_dir=self.cfg.checkpoint.store_path if self.cfg.checkpoint.enabled else "",
        )
        # scheduler attaches the latest available checkpoint to
        # checkpoint-enabled requests (reference: prepareWorkerRequest +
        # scheduler/checkpoint.go attachLatestCheckpoint)
        self.scheduler.checkpoint_lookup = (
            lambda ws, stub_id: self.db.latest_checkpoint(ws, stub_id)
        )

        async def concurrency_check(request) -> bool:
            ws = self.db.get_workspace(request.workspace_id)
            if ws is None:
                return True
            counts = await self.container_repo.workspace_counts(request.workspace_id)
            cpu_limit = ws.get("concurrency_limit_cpu") or 0
            gpu_limit = ws.get("concurrency_limit_gpu") or 0
            if cpu_limit and counts["containers"] >= cpu_limit:
                return False
            if gpu_limit and request.gpu_count > 0 and counts["gpus"] + request.gpu_count > gpu_limit:
                return False
            if request.gpu_count > 1 and not ws.get("multi_gpu_enabled", 1):
                return False
            return True

        self.scheduler.concurrency_check = concurrency_check
        repo_svc = WorkerRepositoryService(
            self.worker_repo, self.container_repo, events=self.events,
            storage_root=self.cfg.storage.root,
        )

        fn_exec = FunctionExecutor(self.db, self.scheduler, self.store, self.runner_env)
        self.dispatcher.register(FN_EXEC, fn_exec.execute)
        fn_svc = FunctionService(self.dispatcher, self.db, self.store)

        self.endpoint_exec = EndpointExecutor(self)
        self.dispatcher.register("endpoint", self.endpoint_exec.execute)
        ep_svc = EndpointService(self)

        self.taskqueue_exec = TaskQueueExecutor(self)
        self.dispatcher.register("taskqueue", self.taskqueue_exec.execute)
        tq_svc = TaskQueueService(self)

        from ..cache.service import CacheCoordinator, CacheService
        from ..cache.store import ContentStore

        self.content_store = ContentStore(
            self.cfg.cache.root,
            page_size=self.cfg.cache.page_size,
            quota_bytes=self.cfg.cache.disk_quota_bytes,
            eviction_watermark=self.cfg.cache.eviction_watermark,
        )
        self.cache_coordinator = CacheCoordinator(self.store)
        self.cache_service = CacheService(self.content_store, self.cache_coordinator)
        # durable checkpoint tier + remote volume snapshots ride on the
        # content store
        runner_svc.content_store = self.content_store
        repo_svc._content_store = self.content_store

        from ..abstractions.disk import DiskService
        from ..worker.disk import DiskManager

        self.disk_manager = DiskManager(os.path.join(self.cfg.storage.root, "disks"))
        self.disk_manager.recover()

        self._services = [
            gw_svc,
            runner_svc,
            repo_svc,
            self.cache_service,
            DiskService(self, self.disk_manager),
            fn_svc,
            ep_svc,
            tq_svc,
            SecretService(self.db),
            MapService(self.store),
            SimpleQueueService(self.store),
            SignalService(self.store),
            VolumeService(self.db, self.cfg.storage.root),
            OutputService(self.db, self.cfg.storage.root),
            PodService(self),
            BotService(self),
        ]
        self.image_service = ImageService(self)
        self._services.append(self.image_service)
        for svc in self._services:
            self.rpc_server.register_service(svc)

    async def _start_http(self) -> None:
        try:
            from .http import build_http_app
            import uvicorn

            app = build_http_app(self)
            config = uvicorn.Config(
                app,
                host=self.cfg.gateway.host,
                port=self.cfg.gateway.http_port,
                log_level="warning",
                lifespan="off",
            )
            self._http_server = uvicorn.Server(config)
            self._http_task = asyncio.create_task(self._http_server.serve())
            # wait for bind
            for _ in range(200):
                if self._http_server.started:
                    break
                await asyncio.sleep(0.01)
            servers = getattr(self._http_server, "servers", [])
            if servers and servers[0].sockets:
                self.http_port = servers[0].sockets[0].getsockname()[1]
        except Exception:
            log.exception("http server failed to start; gRPC-only mode")

    async def drain(self, timeout: float = 10.0) -> None:
        """Stop accepting new invokes (503), let inflight requests finish
        (reference: gateway.go:99-101 draining flag + propagation window)."""
        self.draining = True
        deadline = asyncio.get_event_loop().time() + timeout
        while asyncio.get_event_loop().time() < deadline:
            inflight = sum(
                getattr(getattr(inst, "buffer", None), "inflight", 0)
                for inst in self._instances.values()
            )
            if inflight == 0:
                return
            await asyncio.sleep(0.1)

    async def shutdown(self) -> None:
        if getattr(self, "cache_reconciler", None) is not None:
            await self.cache_reconciler.stop()
        if not self._started:
            return
        await self.drain(timeout=self.cfg.gateway.shutdown_timeout_s)
        for loop in getattr(self, "_loops", []):
            await loop.stop()
        for inst in list(self._instances.values()):
            await inst.shutdown()
        await self.scheduler.shutdown()
        await self.dispatcher.shutdown()
        if self._http_server is not None:
            self._http_server.should_exit = True
            try:
                await asyncio.wait_for(self._http_task, timeout=5.0)
            except (asyncio.TimeoutError, asyncio.CancelledError):
                pass
        if getattr(self, "tcp_ingress", None) is not None:
            await self.tcp_ingress.shutdown()
        if getattr(self, "sni_ingress", None) is not None:
            await self.sni_ingress.stop()
        if getattr(self, "cache_transport", None) is not None:
            await self.cache_transport.stop()
        await self.rpc_server.stop()
        await self.store.stop()
        self.db.close()
        self._started = False
