"""Endpoint runner: HTTP server wrapping the user handler.

The in-container server the gateway's RequestBuffer dispatches to
(reference: sdk/src/beta9/runner/endpoint.py:55-330 — gunicorn+uvicorn
FastAPI app, /health, on_start before readiness, checkpoint READY signal).
GPU containers run the MI355X boot path (HIP pre-init + HBM arena +
optional hipsnap restore) BEFORE the handler import so weight streaming
lands in reserved HBM.
"""

# NOTE: no `from __future__ import annotations` here — FastAPI resolves the
# route handlers' `Request` annotations at def time from build_app's locals.

import asyncio
import logging
import os
import sys
import time
from typing import Any

from .common import (
    GPUBoot,
    unwrap_handler,
    RunnerClient,
    RunnerEnv,
    call_handler,
    load_callable,
    run_on_start,
    setup_environment,
)

log = logging.getLogger("b9.runner.endpoint")


def build_app(env: RunnerEnv, gpu: GPUBoot, rdzv=None):
    from contextlib import asynccontextmanager

    from fastapi import FastAPI, Request
    from fastapi.responses import JSONResponse

    state: dict = {"ready": False, "on_start_value": None, "handler": None, "client": None, "dispatcher": None}
    if rdzv is not None:
        from .mgpu import Rank0Dispatcher

        state["dispatcher"] = Rank0Dispatcher(rdzv)

    async def run_handler(args):
        # multi-GPU: broadcast args so every rank executes the same handler
        # call (RCCL collectives inside the handler line up across ranks)
        if state["dispatcher"] is not None:
            return await state["dispatcher"].dispatch(state["handler"], args)
        return await call_handler(state["handler"], args)

    async def startup() -> None:
        client = RunnerClient(env)
        state["client"] = client
        state["on_start_value"] = await run_on_start(env)
        state["handler"] = load_callable(env.handler)
        state["ready"] = True
        if env.checkpoint_enabled and not gpu.restored:
            # on_start done -> take the stub checkpoint if granted
            # (reference: READY sentinel -> worker startAutoCheckpoint)
            try:
                await client.take_checkpoint_if_granted(gpu)
            except Exception:
                log.warning("checkpoint flow failed", exc_info=True)

    @asynccontextmanager
    async def lifespan(app):
        await startup()
        yield

    app = FastAPI(lifespan=lifespan)

    @app.get("/health")
    async def health():
        if not state["ready"]:
            return JSONResponse({"status": "starting"}, status_code=503)
        return {"status": "ok", "gpu_phases": gpu.phases, "restored": gpu.restored}

    @app.post("/{path:path}")
    async def invoke(request: Request, path: str = ""):
        try:
            body = await request.json() if (await request.body()) else {}
        except Exception:
            body = {}
        try:
            result = await asyncio.wait_for(run_handler(body), timeout=env.timeout)
            return JSONResponse({"result": result})
        except asyncio.TimeoutError:
            return JSONResponse({"error": "timeout"}, status_code=504)
        except Exception as e:
            log.exception("handler error")
            return JSONResponse({"error": str(e)}, status_code=500)

    @app.get("/{path:path}")
    async def invoke_get(request: Request, path: str = ""):
        try:
            result = await asyncio.wait_for(
                run_handler(dict(request.query_params)), timeout=env.timeout
            )
            return JSONResponse({"result": result})
        except Exception as e:
            return JSONResponse({"error": str(e)}, status_code=500)

    return app


async def run_rpc_server(env: RunnerEnv, gpu: GPUBoot, rdzv=None) -> int:
    """The standard endpoint runner: a msgpack-gRPC "Runner" service
    (Invoke/Health) instead of an HTTP stack — the gateway's RequestBuffer
    speaks this protocol for internal dispatch (~10x cheaper per request
    than pure-python HTTP in this image). ASGI stubs still get the HTTP app
    (build_app) for raw-protocol user code."""
    from ..rpc.channel import RpcServer, rpc as rpc_deco

    state: dict = {"ready": False, "handler": None, "dispatcher": None,
                   "batcher": None}
    _inputs_raw = os.environ.get("B9_INPUTS", "")
    if _inputs_raw:
        import json as _json

        try:
            state["inputs_spec"] = _json.loads(_inputs_raw)
        except ValueError:
            state["inputs_spec"] = None
    if rdzv is not None:
        from .mgpu import Rank0Dispatcher

        state["dispatcher"] = Rank0Dispatcher(rdzv)

    class RunnerSvc:
        service_name = "Runner"

        @rpc_deco()
        async def Health(self, req, context, auth):
            return {
                "ok": state["ready"],
                "gpu_phases": gpu.phases,
                "restored": gpu.restored,
            }

        @rpc_deco()
        async def SyncFiles(self, req, context, auth):
            """Serve-mode hot reload: write pushed files into the workspace
            and reload the handler (reference: ContainerSyncWorkspace +
            sdk sync.py watchdog loop)."""
            base = env.workspace_dir or os.getcwd()
            written = 0
            for f in (req or {}).get("files", []):
                path = os.path.normpath(os.path.join(base, f["path"].lstrip("/")))
                if not path.startswith(os.path.realpath(base)) and not path.startswith(base):
                    continue
                os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
                with open(path, "wb") as fh:
                    fh.write(f.get("data", b""))
                # same-size writes within one mtime second would pass the
                # bytecode cache's (mtime, size) check — force a newer stamp
                bump = time.time() + 1.5
                os.utime(path, (bump, bump))
                written += 1
            # reload handler module so the next request sees the new code
            try:
                import importlib

                importlib.invalidate_caches()

                mod_name = env.handler.split(":")[0]
                import sys as _sys

                if mod_name in _sys.modules:
                    importlib.reload(_sys.modules[mod_name])
                state["handler"] = load_callable(env.handler)
            except Exception as e:
                return {"written": written, "reloaded": False, "error": str(e)}
            return {"written": written, "reloaded": True}

        @rpc_deco()
        async def Invoke(self, req, context, auth):
            if not state["ready"]:
                return {"status_code": 503, "body": {"error": "starting"}}
            args = (req or {}).get("args") or {}
            if state.get("inputs_spec"):
                from ..schema import Schema

                try:
                    Schema.validate_payload(state["inputs_spec"], args)
                except ValueError as ve:
                    return {"status_code": 422, "body": {"error": str(ve)}}
            try:
                if state["dispatcher"] is not None:
                    result = await asyncio.wait_for(
                        state["dispatcher"].dispatch(state["handler"], args),
                        timeout=env.timeout,
                    )
                elif state["batcher"] is not None:
                    result = await asyncio.wait_for(
                        state["batcher"].submit(args), timeout=env.timeout
                    )
                else:
                    result = await asyncio.wait_for(
                        call_handler(state["handler"], args), timeout=env.timeout
                    )
                return {"status_code": 200, "body": {"result": result}}
            except asyncio.TimeoutError:
                return {"status_code": 504, "body": {"error": "timeout"}}
            except Exception as e:
                log.exception("handler error")
                return {"status_code": 500, "body": {"error": str(e)}}

    boot = {}
    if os.environ.get("B9_BOOT_TRACE"):
        print(f"[boottrace] run_rpc_server enter {time.time():.6f}", flush=True)
    t0 = time.perf_counter()
    server = RpcServer(host="0.0.0.0", port=env.bind_port)
    server.register_service(RunnerSvc())
    await server.start()
    boot["rpc_bind_ms"] = round((time.perf_counter() - t0) * 1000, 2)

    if os.environ.get("B9_BOOT_TRACE"):
        print(f"[boottrace] pre RunnerClient {time.time():.6f}", flush=True)
    client = RunnerClient(env)
    if os.environ.get("B9_BOOT_TRACE"):
        print(f"[boottrace] post RunnerClient {time.time():.6f}", flush=True)
    t1 = time.perf_counter()
    await run_on_start(env)
    boot["on_start_ms"] = round((time.perf_counter() - t1) * 1000, 2)
    state["handler"] = load_callable(env.handler)
    bm = getattr(unwrap_handler(state["handler"]), "_b9_batch", None)
    if bm:
        from .batching import DynamicBatcher

        state["batcher"] = DynamicBatcher(
            state["handler"], max_size=int(bm.get("max_size", 32)),
            wait_ms=float(bm.get("wait_ms", 2.0)),
        )
    boot["total_ms"] = round((time.perf_counter() - t0) * 1000, 2)
    gpu.phases.update(boot)
    state["ready"] = True
    if os.environ.get("B9_BOOT_TRACE"):
        print(f"[boottrace] runner ready {time.time():.6f}", flush=True)
    if env.checkpoint_enabled and not gpu.restored:
        try:
            await client.take_checkpoint_if_granted(gpu)
        except Exception:
            log.warning("checkpoint flow failed", exc_info=True)
    log.info("endpoint runner (grpc) ready on :%d", env.bind_port)
    try:
        await asyncio.Event().wait()
    finally:
        if state["dispatcher"] is not None:
            state["dispatcher"].stop()
        await client.close()
    return 0


def _resolve_asgi_app(env: RunnerEnv, gpu: GPUBoot):
    """ASGI stubs: if the handler IS an ASGI application (FastAPI/Starlette
    instance or an app-factory returning one), serve the user's app directly
    with a /health wrapper; otherwise wrap the plain handler in the default
    app (reference: asgi decorator serves the user app — sdk @asgi)."""
    from .common import load_callable, unwrap_handler

    try:
        obj = unwrap_handler(load_callable(env.handler))
    except Exception:
        return build_app(env, gpu)
    candidate = obj
    # zero-arg factory convention: @asgi def make_app(): return FastAPI()
    if callable(obj) and not _looks_asgi(obj):
        try:
            import inspect

            if len(inspect.signature(obj).parameters) == 0:
                candidate = obj()
        except (TypeError, ValueError):
            pass
    if _looks_asgi(candidate):
        return _with_health(candidate)
    return build_app(env, gpu)


def _looks_asgi(obj) -> bool:
    try:
        import inspect

        if hasattr(obj, "router") and callable(obj):  # FastAPI/Starlette
            return True
        if callable(obj) and len(inspect.signature(obj).parameters) == 3:
            return True
    except (TypeError, ValueError):
        pass
    return False


def _with_health(user_app):
    async def app(scope, receive, send):
        if scope["type"] == "http" and scope.get("path") == "/health":
            await send({"type": "http.response.start", "status": 200, "headers": [(b"content-type", b"application/json")]})
            await send({"type": "http.response.body", "body": b'{"status": "ok"}'})
            return
        await user_app(scope, receive, send)

    return app


def main() -> int:
    if os.environ.get("B9_BOOT_TRACE"):
        import time as _t
        print(f"[boottrace] endpoint main start {_t.time():.6f}", flush=True)
    logging.basicConfig(level=logging.INFO)
    world = int(os.environ.get("WORLD_SIZE", "1") or 1)
    if world > 1:
        # gpu_count>1 container: RCCL-over-xGMI rendezvous, one process per
        # GPU (mgpu.py) — must run BEFORE any HIP context is created here.
        from .mgpu import run_multi_gpu

        return run_multi_gpu(world)
    env = RunnerEnv.from_env()
    setup_environment(env)
    t0 = time.perf_counter()
    gpu = GPUBoot(env)
    gpu.boot()
    if env.stub_type.split("/")[0] == "asgi":
        app = _resolve_asgi_app(env, gpu)
        import uvicorn

        # Single proc, event-loop concurrency; `workers` >1 would fork and
        # re-init HIP per proc — on MI355X one proc per GPU is the model.
        uvicorn.run(
            app,
            host="0.0.0.0",
            port=env.bind_port,
            log_level="warning",
            lifespan="on",
            access_log=False,
            # first-party RFC6455 protocol: the image has no
            # websockets/wsproto, and @realtime stubs need WS routes
            ws="beta9_amd.common.ws_server:B9WebSocketProtocol",
        )
        return 0
    try:
        return asyncio.run(run_rpc_server(env, gpu))
    except KeyboardInterrupt:
        return 0


if __name__ == "__main__":
    sys.exit(main())
