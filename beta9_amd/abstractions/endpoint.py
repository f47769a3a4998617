"""@endpoint / @asgi abstraction: sync HTTP serving with autoscaling,
keep-warm, and request buffering.

Reference behavior (reference: pkg/abstractions/endpoint/ — endpoint.go
service + task factory, buffer.go request ring, instance.go:51-151
startContainers env contract, autoscaler.go scale funcs, serve.go hot-reload
serve locks). Requests arrive over the gateway HTTP router
(/endpoint/id/<stub>) or the EndpointService.EndpointInvoke RPC; each request
is a tracked task dispatched through the request buffer to a healthy
container with free concurrency tokens.
"""

from __future__ import annotations

import asyncio
import logging
import time
from typing import Any, AsyncIterator, Dict, Optional

from ..common import keys
from ..rpc.channel import not_found, rpc
from ..types import ContainerRequest, StubConfig, Task, TaskStatus, new_id
from .common.autoscaler import AutoscaledInstance, deployment_scale_fn
from .common.buffer import RequestBuffer

log = logging.getLogger(__name__)


class EndpointInstance:
    def __init__(self, gateway, stub_id: str):
        self.gw = gateway
        self.stub_id = stub_id
        self.stub_row = gateway.db.get_stub(stub_id)
        if self.stub_row is None:
            raise KeyError(f"stub {stub_id}")
        self.cfg: StubConfig = gateway.db.get_stub_config(stub_id)
        serve_mode = self.stub_row["type"].endswith("/serve")
        kind = self.stub_row["type"].split("/")[0]
        self.buffer = RequestBuffer(
            gateway.store,
            gateway.container_repo,
            stub_id,
            concurrent_requests=self.cfg.concurrent_requests,
            keep_warm_seconds=self.cfg.keep_warm_seconds,
            request_timeout=self.cfg.timeout,
            protocol="http" if kind == "asgi" else "grpc",
        )

        from ..types import AutoscalerType

        if self.cfg.autoscaler.type == AutoscalerType.llm_token_pressure:
            # LLM-aware scaling (reference: pod/llm.go LLMTokenPressure —
            # vLLM-style token pressure instead of request depth): containers
            # report pressure via RunnerService.ReportPressure; sample = sum.
            async def sample() -> float:
                total = 0.0
                for cid in (await gateway.container_repo.containers_for_stub(stub_id)).keys():
                    v = await gateway.store.get(f"llm:pressure:{stub_id}:{cid}")
                    total += float(v or 0.0)
                return max(total, float(self.buffer.inflight))

        else:

            async def sample() -> float:
                return float(self.buffer.inflight)

        scale_fn = deployment_scale_fn(self.cfg, gateway.cfg.gateway.stub_limits_max_replicas)
        self.instance = AutoscaledInstance(
            gateway,
            stub_id,
            self.cfg,
            self.stub_row,
            self._make_request,
            sample,
            scale_fn=scale_fn,
            serve_mode=serve_mode,
        )
        self.instance.start()

    def _make_request(self) -> ContainerRequest:
        cfg = self.cfg
        env = self.gw.runner_env(self.stub_row["workspace_id"], self.stub_row, cfg)
        env.update(
            {
                "HANDLER": cfg.handler,
                "ON_START": cfg.on_start,
                "TIMEOUT": str(cfg.timeout),
            }
        )
        if cfg.inputs:
            import json as _json

            env["B9_INPUTS"] = _json.dumps(cfg.inputs)
        env.update(cfg.env)
        return ContainerRequest(
            container_id=f"endpoint-{self.stub_id[-8:]}-{new_id()}",
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
            checkpoint_enabled=cfg.checkpoint_enabled,
            timestamp=time.time(),
        )

    async def forward(self, method: str, path: str, headers=None, content: bytes = b"", json_body=None, affinity_key=None):
        self.instance.autoscaler.trigger()
        return await self.buffer.dispatch(
            method, path, headers=headers, content=content, json_body=json_body,
            wait_timeout=min(self.cfg.timeout, 120.0), affinity_key=affinity_key,
        )

    async def shutdown(self) -> None:
        await self.instance.shutdown()
        await self.buffer.close()


class EndpointExecutor:
    """Task executor: tracks each endpoint request as a task and forwards it
    through the buffer (reference: endpoint/task.go EndpointTask.Execute)."""

    def __init__(self, gateway):
        self.gw = gateway

    async def execute(self, task: Task, payload: Any) -> None:
        instance = await get_instance(self.gw, task.stub_id)
        payload = payload or {}
        try:
            # affinity: explicit payload key or X-B9-Affinity header ->
            # consistent-hash container routing (KV-cache/prefix reuse)
            hdrs = payload.get("headers") or {}
            resp = await instance.forward(
                payload.get("method", "POST"),
                payload.get("path", "/"),
                headers=payload.get("headers"),
                json_body=payload.get("json"),
                content=payload.get("content", b""),
                affinity_key=payload.get("affinity") or hdrs.get("x-b9-affinity"),
            )
            await self.gw.dispatcher.complete(
                task.task_id,
                resp,
                status=TaskStatus.complete if resp["status_code"] < 500 else TaskStatus.error,
            )
        except Exception as e:
            log.warning("endpoint task %s failed: %s", task.task_id, e)
            await self.gw.dispatcher.complete(task.task_id, {"error": str(e)}, status=TaskStatus.error)


async def get_instance(gateway, stub_id: str) -> EndpointInstance:
    inst = gateway._instances.get(stub_id)
    if inst is None:
        inst = EndpointInstance(gateway, stub_id)
        gateway._instances[stub_id] = inst
    return inst


class EndpointService:
    service_name = "EndpointService"

    def __init__(self, gateway):
        self.gw = gateway

    @rpc()
    async def EndpointInvoke(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Synchronous invoke from the SDK (the HTTP route is the other
        entry; both run the same task path)."""
        stub_id = req["stub_id"]
        if self.gw.db.get_stub(stub_id) is None:
            raise not_found(f"stub {stub_id}")
        payload = {"method": "POST", "path": "/", "json": req.get("args")}
        if req.get("affinity"):
            payload["affinity"] = str(req["affinity"])
        task = await self.gw.dispatcher.send_and_execute(
            "endpoint", auth["workspace_id"], stub_id, payload,
        )
        done = await self.gw.dispatcher.wait(task.task_id, timeout=float(req.get("timeout", 300.0)))
        result = await self.gw.dispatcher.get_result(task.task_id)
        return {"task_id": task.task_id, "status": done.status.value, "result": result}

    @rpc()
    async def StartEndpointServe(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Serve mode: hold a TTL lock that keeps exactly one container up
        (reference: endpoint/serve.go:16-60 + serve-mode autoscaler)."""
        stub_id = req["stub_id"]
        ttl = float(req.get("timeout", 600.0))
        key = keys.scheduler_serve_lock(auth["workspace_id"], stub_id)
        await self.gw.store.set(key, 1, ttl=ttl)
        inst = await get_instance(self.gw, stub_id)
        inst.instance.autoscaler.trigger()
        return {"ok": True}

    @rpc()
    async def StopEndpointServe(self, req: Dict[str, Any], context, auth) -> bool:
        key = keys.scheduler_serve_lock(auth["workspace_id"], req["stub_id"])
        await self.gw.store.delete(key)
        return True
