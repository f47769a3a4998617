"""Gateway HTTP API: invoke routes + REST /api/v1.

Reference surface (reference: pkg/gateway/gateway.go:227-291 echo groups,
pkg/api/v1/*.go handlers, pkg/abstractions/endpoint/http.go:17-118 invoke
routes): POST /endpoint/id/<stub> dispatches through the endpoint task path;
/api/v1/{health,containers,tasks,workers,deployments,logs,spans,metrics}
expose control-plane state. Auth: Bearer workspace token.
"""

from __future__ import annotations

import logging
import os
from typing import Any, Optional

from fastapi import Depends, FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, Response

log = logging.getLogger(__name__)


def build_http_app(gateway) -> FastAPI:
    app = FastAPI(title="beta9-amd gateway")

    async def auth_ctx(request: Request) -> dict:
        tok = request.headers.get("authorization", "")
        if tok.startswith("Bearer "):
            tok = tok[7:]
        ctx = gateway.db.authorize_token(tok)
        if ctx is None:
            raise HTTPException(status_code=401, detail="invalid token")
        return {"workspace_id": ctx["workspace_id"], "workspace_name": ctx["workspace_name"]}

    async def _public_auth(stub_id: str, request: Request) -> Optional[dict]:
        """authorized=False stubs are publicly invokable: the request is
        accounted to the stub's own workspace (reference: StubConfigV1
        .authorized + public invoke URLs)."""
        stub = gateway.db.get_stub(stub_id)
        if stub is None:
            return None
        try:
            cfg = gateway.db.get_stub_config(stub_id)
        except Exception:
            return None
        if cfg is not None and getattr(cfg, "authorized", True) is False:
            if (cfg.extra or {}).get("require_signature"):
                sig = request.headers.get("x-b9-signature", "")
                if not await _verify_signature(stub["workspace_id"], sig, request, stub_id):
                    raise HTTPException(status_code=401, detail="bad signature")
            return {"workspace_id": stub["workspace_id"], "workspace_name": "public"}
        return None

    async def _verify_signature(
        workspace_id: str, sig: str, request: Request, stub_id: str, max_age_s: float = 300.0
    ) -> bool:
        import hashlib
        import hmac as _hmac
        import time as _time

        if not sig.startswith("v1="):
            return False
        try:
            ts, mac = sig[3:].split(":", 1)
            ts_val = float(ts)
        except ValueError:
            return False  # malformed timestamp is a 401, never a 500
        if abs(_time.time() - ts_val) > max_age_s:
            return False
        key = gateway.db.get_or_create_signing_key(workspace_id).encode()
        body = await request.body()  # cached by starlette for the route
        msg = (
            ts.encode() + b"." + request.method.upper().encode() + b"."
            + stub_id.encode() + b"." + body
        )
        want = _hmac.new(key, msg, hashlib.sha256).hexdigest()
        return _hmac.compare_digest(want, mac)

    @app.get("/api/v1/health")
    async def health():
        return {"status": "ok"}

    # -- subdomain (host-style) routing -------------------------------------
    # {name}.app.{base}, {name}-latest.app.{base}, {name}-v{N}.app.{base} or
    # {stub_id}.app.{base} rewrite onto the matching invoke route before the
    # router runs (reference: pkg/gateway/middleware/subdomain.go:19-36)
    @app.middleware("http")
    async def subdomain_routing(request, call_next):
        host = request.headers.get("host", "").split(":")[0]
        suffix = ".app." + gateway.cfg.gateway.base_domain
        if host.endswith(suffix):
            sub = host[: -len(suffix)]
            path = request.scope.get("path", "/")
            if gateway.db.get_stub(sub) is not None:
                request.scope["path"] = f"/endpoint/id/{sub}" + (path if path != "/" else "/")
            else:
                name, version = sub, ""
                if "-v" in sub and sub.rsplit("-v", 1)[1].isdigit():
                    name, version = sub.rsplit("-v", 1)
                elif sub.endswith("-latest"):
                    name = sub[: -len("-latest")]
                target = f"/deployment/{name}" + (f"/v{version}" if version else "")
                request.scope["path"] = target + (path if path != "/" else "/")
        return await call_next(request)

    # -- invoke routes ------------------------------------------------------
    async def _invoke(stub_id: str, request: Request, auth: dict, path: str = "/"):
        if gateway.draining:
            return JSONResponse({"error": "gateway draining"}, status_code=503)
        from ..abstractions.endpoint import get_instance

        try:
            instance = await get_instance(gateway, stub_id)
        except KeyError:
            raise HTTPException(status_code=404, detail=f"stub {stub_id} not found")
        body = await request.body()
        task = await gateway.dispatcher.send_and_execute(
            "endpoint",
            auth["workspace_id"],
            stub_id,
            {
                "method": request.method,
                "path": path,
                "headers": {
                    "content-type": request.headers.get("content-type", "application/json"),
                    **(
                        {"x-b9-affinity": request.headers["x-b9-affinity"]}
                        if "x-b9-affinity" in request.headers
                        else {}
                    ),
                },
                "content": body,
            },
        )
        done = await gateway.dispatcher.wait(task.task_id, timeout=instance.cfg.timeout + 60.0)
        result = await gateway.dispatcher.get_result(task.task_id)
        if result is None:
            return JSONResponse({"task_id": task.task_id, "status": done.status.value}, status_code=500)
        if isinstance(result, dict) and "status_code" in result:
            return JSONResponse(
                {"task_id": task.task_id, **(result.get("body") if isinstance(result.get("body"), dict) else {"body": result.get("body")})},
                status_code=result["status_code"],
            )
        return JSONResponse({"task_id": task.task_id, "result": result})

    async def _auth_or_public(stub_id: str, request: Request) -> dict:
        pub = await _public_auth(stub_id, request)
        if pub is not None:
            return pub
        return await auth_ctx(request)

    @app.api_route("/endpoint/id/{stub_id}", methods=["GET", "POST"])
    async def invoke_endpoint(stub_id: str, request: Request):
        return await _invoke(stub_id, request, await _auth_or_public(stub_id, request))

    @app.api_route("/endpoint/id/{stub_id}/{path:path}", methods=["GET", "POST"])
    async def invoke_endpoint_path(stub_id: str, path: str, request: Request):
        return await _invoke(stub_id, request, await _auth_or_public(stub_id, request), path="/" + path)

    @app.api_route("/asgi/id/{stub_id}/{path:path}", methods=["GET", "POST"])
    async def invoke_asgi(stub_id: str, path: str, request: Request, auth: dict = Depends(auth_ctx)):
        return await _invoke(stub_id, request, auth, path="/" + path)

    # -- deployment-name routing (reference: subdomain middleware A.6:
    # {name}-latest / {name}-v{N} hosts become /deployment/{name}[/v{N}]) ---
    @app.api_route("/deployment/{name}/v{version}/{path:path}", methods=["GET", "POST"])
    async def invoke_deployment_version(
        name: str, version: int, path: str, request: Request, auth: dict = Depends(auth_ctx)
    ):
        dep = gateway.db.get_deployment(auth["workspace_id"], name, version=version)
        if dep is None:
            raise HTTPException(status_code=404, detail=f"deployment {name} v{version}")
        return await _invoke(dep["stub_id"], request, auth, path="/" + path)

    @app.api_route("/deployment/{name}/{path:path}", methods=["GET", "POST"])
    async def invoke_deployment(name: str, path: str, request: Request, auth: dict = Depends(auth_ctx)):
        dep = gateway.db.get_deployment(auth["workspace_id"], name)
        if dep is None:
            raise HTTPException(status_code=404, detail=f"deployment {name}")
        return await _invoke(dep["stub_id"], request, auth, path="/" + path)

    # -- REST API -----------------------------------------------------------
    @app.get("/api/v1/containers")
    async def list_containers(auth: dict = Depends(auth_ctx)):
        out = []
        for key in await gateway.store.keys("container:state:*"):
            d = await gateway.store.get(key)
            if d and d.get("workspace_id") == auth["workspace_id"]:
                out.append(d)
        return out

    async def _own_container_or_404(container_id: str, auth: dict) -> None:
        # workspace-scoped access: 404 on other tenants' containers
        # (reference: workspace auth on container endpoints)
        if auth.get("token_type") == "cluster_admin":
            return
        st = await gateway.container_repo.get_state(container_id)
        if st is not None:
            if st.get("workspace_id") != auth["workspace_id"]:
                raise HTTPException(status_code=404)
            return
        rows = gateway.db._query(
            "SELECT workspace_id FROM tasks WHERE container_id=? LIMIT 1", (container_id,)
        )
        if not rows or rows[0]["workspace_id"] != auth["workspace_id"]:
            raise HTTPException(status_code=404)

    @app.post("/api/v1/containers/{container_id}/stop")
    async def stop_container(container_id: str, auth: dict = Depends(auth_ctx)):
        await _own_container_or_404(container_id, auth)
        await gateway.scheduler.stop(container_id)
        return {"ok": True}

    @app.get("/api/v1/logs/{container_id}")
    async def container_logs(container_id: str, tail: int = 200, auth: dict = Depends(auth_ctx)):
        await _own_container_or_404(container_id, auth)
        return await gateway.container_repo.get_logs(container_id, tail=tail)

    @app.get("/api/v1/spans/{container_id}")
    async def container_spans(container_id: str, auth: dict = Depends(auth_ctx)):
        await _own_container_or_404(container_id, auth)
        return await gateway.container_repo.get_spans(container_id)

    @app.get("/api/v1/tasks")
    async def list_tasks(stub_id: str = "", limit: int = 100, auth: dict = Depends(auth_ctx)):
        return gateway.db.list_tasks(auth["workspace_id"], stub_id, limit)

    @app.get("/api/v1/tasks/{task_id}")
    async def get_task(task_id: str, auth: dict = Depends(auth_ctx)):
        admin = auth.get("token_type") == "cluster_admin"
        t = await gateway.dispatcher.get_task(task_id)
        if t is None:
            row = gateway.db.get_task(task_id)
            if row is None or (not admin and row.get("workspace_id") != auth["workspace_id"]):
                raise HTTPException(status_code=404)
            return row
        if not admin and t.workspace_id != auth["workspace_id"]:
            raise HTTPException(status_code=404)
        return t.model_dump()

    @app.get("/api/v1/workers")
    async def list_workers(auth: dict = Depends(auth_ctx)):
        return [w.model_dump() for w in await gateway.worker_repo.list()]

    @app.get("/api/v1/deployments")
    async def list_deployments(auth: dict = Depends(auth_ctx)):
        return gateway.db.list_deployments(auth["workspace_id"])

    @app.get("/api/v1/metrics")
    async def metrics(auth: dict = Depends(auth_ctx)):
        return {
            "scheduler": gateway.scheduler.metrics.snapshot(),
            "backlog": await gateway.scheduler.backlog_len(),
            "registry": gateway.metrics.snapshot() if getattr(gateway, "metrics", None) else {},
        }

    @app.get("/api/v1/usage")
    async def workspace_usage(auth: dict = Depends(auth_ctx)):
        from ..common import keys as _keys

        u = await gateway.store.hgetall(_keys.workspace_usage(auth["workspace_id"]))
        return {
            "workspace_id": auth["workspace_id"],
            "task_count": int(u.get("task_count", 0)),
            "task_duration_ms": float(u.get("task_duration_ms", 0.0)),
            "cost_cents": float(u.get("cost_cents", 0.0)),
        }

    @app.get("/metrics")
    async def metrics_prometheus():
        if getattr(gateway, "metrics", None) is None:
            return Response(content="", media_type="text/plain")
        return Response(content=gateway.metrics.prometheus_text(), media_type="text/plain")

    @app.get("/api/v1/events")
    async def events(limit: int = 100, prefix: str = "", auth: dict = Depends(auth_ctx)):
        if getattr(gateway, "events", None) is None:
            return []
        return await gateway.events.recent(limit=limit, name_prefix=prefix)

    @app.get("/api/v1/outputs/{rel:path}")
    async def get_output(rel: str, auth: dict = Depends(auth_ctx)):
        root = os.path.realpath(gateway.cfg.storage.root)
        path = os.path.realpath(os.path.join(root, rel))
        # commonpath (not startswith) so a sibling like <root>2 is unreachable
        if os.path.commonpath([root, path]) != root or not os.path.isfile(path):
            raise HTTPException(status_code=404)
        with open(path, "rb") as f:
            return Response(content=f.read(), media_type="application/octet-stream")

    return app
