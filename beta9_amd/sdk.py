"""Python SDK: decorators and clients.

The user surface of the framework (reference: sdk/src/beta9/ —
@endpoint/@function/@task_queue decorators, Pod, Sandbox, Image, Volume,
abstractions/base/runner.py:569-811 prepare_runtime, mixins.py:42 deploy).
Sync user code drives the async RPC layer through a background event-loop
thread.

Usage:
    from beta9_amd import function, endpoint, task_queue, Image, Sandbox

    @function(cpu=1000, memory=512)
    def hello(x):
        return x * 2

    hello.remote(21)  # -> 42
"""

from __future__ import annotations

import asyncio
import atexit
import functools
import hashlib
import io
import os
import sys
import threading
import time
import zipfile
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, Iterable, List, Optional

from .rpc.channel import RpcClient
from .types import (
    AutoscalerConfig,
    AutoscalerType,
    Mount,
    MountType,
    RuntimeConfig,
    StubConfig,
    TaskPolicy,
)

# ---------------------------------------------------------------------------
# client plumbing
# ---------------------------------------------------------------------------


class _LoopThread:
    _instance: Optional["_LoopThread"] = None

    def __init__(self) -> None:
        self.loop = asyncio.new_event_loop()
        self.thread = threading.Thread(target=self._run, daemon=True, name="b9-sdk-loop")
        self.thread.start()

    def _run(self) -> None:
        asyncio.set_event_loop(self.loop)
        self.loop.run_forever()

    @classmethod
    def get(cls) -> "_LoopThread":
        if cls._instance is None or not cls._instance.thread.is_alive():
            cls._instance = cls()
        return cls._instance

    def run(self, coro):
        return asyncio.run_coroutine_threadsafe(coro, self.loop).result()


@dataclass
class ClientConfig:
    gateway_addr: str = ""
    token: str = ""

    @classmethod
    def from_env(cls) -> "ClientConfig":
        return cls(
            gateway_addr=os.environ.get("B9_GATEWAY", "127.0.0.1:1993"),
            token=os.environ.get("B9_TOKEN", ""),
        )


_default_config: Optional[ClientConfig] = None


def configure(gateway_addr: str, token: str) -> None:
    """Point the SDK at a gateway (tests/benches call this; otherwise env)."""
    global _default_config
    _default_config = ClientConfig(gateway_addr=gateway_addr, token=token)


def get_config() -> ClientConfig:
    return _default_config or ClientConfig.from_env()


class Client:
    """Sync gateway client."""

    def __init__(self, config: Optional[ClientConfig] = None):
        self.config = config or get_config()
        self._rpc: Optional[RpcClient] = None
        self._lt = _LoopThread.get()

    def _client(self) -> RpcClient:
        if self._rpc is None:
            self._rpc = RpcClient(self.config.gateway_addr, token=self.config.token)
        return self._rpc

    def call(self, service: str, method: str, payload: Any = None, timeout: Optional[float] = None) -> Any:
        return self._lt.run(self._client().call(service, method, payload, timeout=timeout))

    def stream(self, service: str, method: str, payload: Any = None) -> Iterable[Any]:
        q: "asyncio.Queue" = None  # type: ignore
        done = object()
        import queue as _queue

        out: "_queue.Queue" = _queue.Queue()

        async def pump():
            try:
                async for item in self._client().stream(service, method, payload):
                    out.put(item)
            except Exception as e:
                out.put(e)
            finally:
                out.put(done)

        fut = asyncio.run_coroutine_threadsafe(pump(), self._lt.loop)
        while True:
            item = out.get()
            if item is done:
                break
            if isinstance(item, Exception):
                raise item
            yield item
        fut.result()

    def client_stream(self, service: str, method: str, payloads: Iterable[Any], timeout: Optional[float] = None) -> Any:
        async def gen():
            for p in payloads:
                yield p

        return self._lt.run(self._client().client_stream(service, method, gen(), timeout=timeout))

    def close(self) -> None:
        if self._rpc is not None:
            self._lt.run(self._rpc.close())
            self._rpc = None


# ---------------------------------------------------------------------------
# Image
# ---------------------------------------------------------------------------


@dataclass
class Image:
    """Build recipe for a container image (reference: sdk Image surface,
    SURVEY.md §A.7)."""

    python_version: str = ""
    python_packages: List[str] = field(default_factory=list)
    commands: List[str] = field(default_factory=list)
    base_image: str = ""
    env_vars: List[str] = field(default_factory=list)

    def add_python_packages(self, pkgs: List[str]) -> "Image":
        self.python_packages.extend(pkgs)
        return self

    def add_commands(self, cmds: List[str]) -> "Image":
        self.commands.extend(cmds)
        return self

    def with_envs(self, envs: List[str]) -> "Image":
        self.env_vars.extend(envs)
        return self

    def from_registry(self, ref: str) -> "Image":
        self.base_image = ref
        return self

    def from_dockerfile(self, path: str, context_dir: str = "") -> "Image":
        """Build from a Dockerfile (reference: sdk Image.from_dockerfile,
        SURVEY §A.7). Supported subset — FROM (local OCI layout/archive or
        imported image id), RUN, ENV, WORKDIR, COPY (from context_dir) —
        mapped onto the container build pipeline; unsupported instructions
        raise rather than silently no-op."""
        import shlex

        ctx = context_dir or os.path.dirname(os.path.abspath(path)) or "."
        with open(path) as f:
            lines = f.read().splitlines()
        # join line continuations
        joined, buf = [], ""
        for ln in lines:
            s = ln.strip()
            if not s or s.startswith("#"):
                continue
            if s.endswith("\\"):
                buf += s[:-1] + " "
                continue
            joined.append(buf + s)
            buf = ""
        workdir = ""
        for ln in joined:
            op, _, rest = ln.partition(" ")
            op = op.upper()
            rest = rest.strip()
            if op == "FROM":
                self.base_image = rest.split()[0]
            elif op == "RUN":
                cmd = rest
                if workdir:
                    cmd = f"cd {shlex.quote(workdir)} && ({cmd})"
                self.commands.append(cmd)
            elif op == "ENV":
                first = rest.split()[0] if rest.split() else ""
                if "=" in first:  # ENV K=V [K2=V2...]
                    self.env_vars.extend(rest.split())
                else:  # legacy ENV K V
                    k, _, v = rest.partition(" ")
                    self.env_vars.append(f"{k}={v.strip()}")
            elif op == "WORKDIR":
                workdir = rest
                self.commands.append(f"mkdir -p {shlex.quote(workdir)}")
            elif op == "COPY":
                parts = shlex.split(rest)
                if len(parts) < 2:
                    raise ValueError(f"COPY needs src dst: {ln}")
                srcs, dst = parts[:-1], parts[-1]
                if workdir and not dst.startswith("/"):
                    dst = os.path.join(workdir, dst)
                for src in srcs:
                    full = os.path.join(ctx, src)
                    if not os.path.exists(full):
                        raise FileNotFoundError(f"COPY source {full} not in context")
                    # context files travel as base64 into a build step (the
                    # build container cannot see the client filesystem)
                    import base64 as _b64

                    data = open(full, "rb").read() if os.path.isfile(full) else None
                    if data is None:
                        raise ValueError(f"COPY of directories not supported yet: {src}")
                    b = _b64.b64encode(data).decode()
                    self.commands.append(
                        f"mkdir -p $(dirname {shlex.quote(dst)}) && "
                        f"echo {b} | base64 -d > {shlex.quote(dst)}"
                    )
            elif op in ("CMD", "ENTRYPOINT", "EXPOSE", "LABEL", "ARG", "USER"):
                continue  # runtime-level; stubs own entrypoint/ports
            else:
                raise ValueError(f"unsupported Dockerfile instruction: {op}")
        return self

    @property
    def is_empty(self) -> bool:
        if getattr(self, "_snapshot_id", ""):
            return False
        return not (self.python_packages or self.commands or self.base_image or self.env_vars)

    def build_payload(self) -> Dict[str, Any]:
        return {
            "python_version": self.python_version,
            "python_packages": self.python_packages,
            "commands": self.commands,
            "base_image": self.base_image,
            "env_vars": self.env_vars,
        }

    @classmethod
    def from_snapshot(cls, image_id: str) -> "Image":
        """An Image pinned to an existing id (e.g. from
        Sandbox.create_image_from_filesystem())."""
        img = cls()
        img._snapshot_id = image_id
        return img

    def build(self, client: Optional[Client] = None, verbose: bool = False) -> str:
        if getattr(self, "_snapshot_id", ""):
            return self._snapshot_id
        client = client or Client()
        v = client.call("ImageService", "VerifyImageBuild", self.build_payload())
        if v["exists"]:
            return v["image_id"]
        image_id = ""
        for msg in client.stream("ImageService", "BuildImage", self.build_payload()):
            if verbose and msg.get("type") == "log":
                print(msg["line"])
            if msg.get("type") == "done":
                image_id = msg["image_id"]
            if msg.get("type") == "error":
                raise RuntimeError(f"image build failed: {msg.get('line')}")
        if not image_id:
            raise RuntimeError("image build did not complete")
        return image_id


# ---------------------------------------------------------------------------
# Volumes / outputs / autoscalers
# ---------------------------------------------------------------------------


@dataclass
class Volume:
    name: str
    mount_path: str

    def mount(self) -> Mount:
        return Mount(local_path=self.name, mount_path=self.mount_path, mount_type=MountType.volume)


@dataclass
class PricingPolicy:
    """Per-task billing on public stubs (reference: sdk type.py:437)."""

    def __init__(self, max_in_flight: int = 10, cost_model: str = "task",
                 cost_per_task: float = 0.0, cost_per_task_duration_ms: float = 0.0):
        self.max_in_flight = max_in_flight
        self.cost_model = cost_model
        self.cost_per_task = cost_per_task
        self.cost_per_task_duration_ms = cost_per_task_duration_ms

    def to_config(self):
        from .types import PricingPolicy as _PP

        return _PP(
            max_in_flight=self.max_in_flight,
            cost_model=self.cost_model,
            cost_per_task=self.cost_per_task,
            cost_per_task_duration_ms=self.cost_per_task_duration_ms,
        )


class QueueDepthAutoscaler:
    max_containers: int = 1
    tasks_per_container: int = 1
    min_containers: int = 0

    def to_config(self) -> AutoscalerConfig:
        return AutoscalerConfig(
            type=AutoscalerType.queue_depth,
            max_containers=self.max_containers,
            tasks_per_container=self.tasks_per_container,
            min_containers=self.min_containers,
        )


@dataclass
class LLMTokenPressureAutoscaler:
    """Scale LLM serving containers by reported token pressure instead of
    request depth (reference: SDK type.py LLMTokenPressureAutoscaler +
    pod/llm.go). Containers report via RunnerClient.report_pressure()."""

    max_containers: int = 1
    pressure_per_container: float = 1.0
    min_containers: int = 0

    def to_config(self) -> AutoscalerConfig:
        return AutoscalerConfig(
            type=AutoscalerType.llm_token_pressure,
            max_containers=self.max_containers,
            tasks_per_container=max(1, int(self.pressure_per_container)),
            min_containers=self.min_containers,
        )


# ---------------------------------------------------------------------------
# runtime prep shared by all decorators
# ---------------------------------------------------------------------------


def _parse_gpu(gpu: str) -> tuple:
    """"mi355x" -> ("mi355x", 1.0); "mi355x:0.5" -> ("mi355x", 0.5)."""
    if ":" not in gpu:
        return gpu, 1.0
    name, _, frac_s = gpu.partition(":")
    try:
        frac = float(frac_s)
    except ValueError:
        raise ValueError(f"bad gpu spec {gpu!r}: fraction must be a number")
    if not (0.0 < frac <= 1.0):
        raise ValueError(f"bad gpu spec {gpu!r}: fraction must be in (0, 1]")
    return name, frac


def _handler_spec(fn: Callable) -> str:
    module = fn.__module__
    if module == "__main__":
        main_file = getattr(sys.modules.get("__main__"), "__file__", "")
        module = os.path.splitext(os.path.basename(main_file))[0] if main_file else "__main__"
    return f"{module}:{fn.__qualname__}"


def _zip_workspace(root: str) -> bytes:
    buf = io.BytesIO()
    skip_dirs = {".git", "__pycache__", ".pytest_cache", "node_modules", ".venv"}
    with zipfile.ZipFile(buf, "w", zipfile.ZIP_DEFLATED) as z:
        for dirpath, dirnames, filenames in os.walk(root):
            dirnames[:] = [d for d in dirnames if d not in skip_dirs]
            for fname in filenames:
                if fname.endswith((".pyc", ".so")):
                    continue
                fp = os.path.join(dirpath, fname)
                if os.path.getsize(fp) > 64 * (1 << 20):
                    continue
                z.write(fp, os.path.relpath(fp, root))
    return buf.getvalue()


class RunnerAbstraction:
    """Shared deploy/runtime-prep machinery (reference:
    abstractions/base/runner.py prepare_runtime + mixins.py deploy)."""

    stub_kind = "function"

    def __init__(
        self,
        func: Optional[Callable] = None,
        cpu: int = 1000,
        memory: int = 1024,
        gpu: str = "",
        gpu_count: int = 0,
        image: Optional[Image] = None,
        timeout: float = 3600.0,
        retries: int = 3,
        workers: int = 1,
        concurrent_requests: int = 1,
        keep_warm_seconds: float = 10.0,
        max_pending_tasks: int = 100,
        on_start: Optional[Callable] = None,
        volumes: Optional[List[Volume]] = None,
        secrets: Optional[List[str]] = None,
        env: Optional[Dict[str, str]] = None,
        autoscaler: Optional[QueueDepthAutoscaler] = None,
        checkpoint_enabled: bool = False,
        pool: str = "",
        name: str = "",
        sync_workspace: bool = False,
        client: Optional[Client] = None,
        inputs=None,  # Schema
        outputs=None,  # Schema
        pricing: Optional["PricingPolicy"] = None,
        authorized: bool = True,
    ):
        self.func = func
        self.cpu = cpu
        self.memory = memory
        # "mi355x:0.5" = half a shared device (Thunder-analog fractional
        # GPU — pkg/worker/thunder.go; our worker packs shares onto one
        # GCD and fences the HBM budget per process)
        self.gpu, self.gpu_fraction = _parse_gpu(gpu)
        self.gpu_count = gpu_count if gpu_count else (1 if self.gpu else 0)
        if self.gpu_fraction < 1.0 and self.gpu_count > 1:
            raise ValueError("fractional gpu requires gpu_count=1")
        self.image = image or Image()
        self.timeout = timeout
        self.retries = retries
        self.workers = workers
        self.concurrent_requests = concurrent_requests
        self.keep_warm_seconds = keep_warm_seconds
        self.max_pending_tasks = max_pending_tasks
        self.on_start = on_start
        self.volumes = volumes or []
        self.secrets = secrets or []
        self.pricing = pricing
        self.authorized = authorized
        self.env = env or {}
        self.autoscaler = autoscaler
        self.checkpoint_enabled = checkpoint_enabled
        self.pool = pool
        self.name = name
        self.sync_workspace = sync_workspace
        self.inputs = inputs
        self.outputs = outputs
        self._client = client
        self._stub_id: Optional[str] = None
        if func is not None:
            functools.update_wrapper(self, func)

    @property
    def client(self) -> Client:
        if self._client is None:
            self._client = Client()
        return self._client

    def __call__(self, *args, **kwargs):
        if self.func is None and len(args) == 1 and callable(args[0]) and not kwargs:
            # decorator used with arguments: @function(cpu=..)
            self.func = args[0]
            functools.update_wrapper(self, self.func)
            return self
        return self.func(*args, **kwargs)

    def local(self, *args, **kwargs):
        return self.func(*args, **kwargs)

    def _stub_config(self) -> StubConfig:
        cfg = StubConfig(
            runtime=RuntimeConfig(
                cpu=self.cpu,
                memory_mb=self.memory,
                gpu=self.gpu,
                gpus=[self.gpu] if self.gpu else [],
                gpu_count=self.gpu_count,
                gpu_fraction=self.gpu_fraction,
                image_id="",
            ),
            handler=_handler_spec(self.func) if self.func else "",
            on_start=_handler_spec(self.on_start) if self.on_start else "",
            keep_warm_seconds=self.keep_warm_seconds,
            max_pending_tasks=self.max_pending_tasks,
            task_policy=TaskPolicy(max_retries=self.retries, timeout=self.timeout),
            workers=self.workers,
            concurrent_requests=self.concurrent_requests,
            volumes=[v.mount() for v in self.volumes],
            secrets=list(self.secrets),
            env=dict(self.env),
            autoscaler=self.autoscaler.to_config() if self.autoscaler else AutoscalerConfig(),
            checkpoint_enabled=self.checkpoint_enabled,
            pool=self.pool,
            timeout=self.timeout,
            inputs=self.inputs.to_spec() if self.inputs is not None else None,
            outputs=self.outputs.to_spec() if self.outputs is not None else None,
            pricing=self.pricing.to_config() if self.pricing is not None else None,
            authorized=self.authorized,
        )
        return cfg

    def prepare_runtime(self, stub_type: Optional[str] = None, force: bool = False) -> str:
        if self._stub_id is not None and not force:
            return self._stub_id
        cfg = self._stub_config()
        if not self.image.is_empty:
            cfg.runtime.image_id = self.image.build(self.client)
        if self.sync_workspace:
            data = _zip_workspace(os.getcwd())
            r = self.client.client_stream(
                "GatewayService",
                "PutObjectStream",
                [{"data": data[i : i + (1 << 20)]} for i in range(0, len(data), 1 << 20)] or [{"data": b""}],
            )
            cfg.extra["object_id"] = r["object_id"]
        name = self.name or (self.func.__name__ if self.func else "anon")
        r = self.client.call(
            "GatewayService",
            "GetOrCreateStub",
            {
                "name": name,
                "stub_type": stub_type or self.stub_kind,
                "config": cfg.model_dump(),
                "app_name": name,
            },
        )
        self._stub_id = r["stub_id"]
        return self._stub_id

    def deploy(self, name: str = "") -> Dict[str, Any]:
        stub_id = self.prepare_runtime(stub_type=f"{self.stub_kind}/deployment", force=True)
        return self.client.call(
            "GatewayService",
            "DeployStub",
            {"stub_id": stub_id, "name": name or self.name or self.func.__name__},
        )


# ---------------------------------------------------------------------------
# @function
# ---------------------------------------------------------------------------


class Function(RunnerAbstraction):
    stub_kind = "function"

    def remote(self, *args, **kwargs) -> Any:
        stub_id = self.prepare_runtime()
        result = None
        status = "ERROR"
        for msg in self.client.stream(
            "FunctionService",
            "FunctionInvoke",
            {"stub_id": stub_id, "args": {"args": list(args), "kwargs": kwargs}},
        ):
            if msg.get("type") == "result":
                status = msg.get("status")
                result = msg.get("result")
        if status != "COMPLETE":
            raise RuntimeError(f"function failed: status={status} result={result!r}")
        return result

    def map(self, inputs: Iterable[Any]) -> List[Any]:
        """Parallel fan-out over inputs (reference: function .map())."""
        stub_id = self.prepare_runtime()
        lt = _LoopThread.get()

        async def one(x):
            rpc = RpcClient(self.client.config.gateway_addr, token=self.client.config.token)
            try:
                result, status = None, "ERROR"
                async for msg in rpc.stream(
                    "FunctionService",
                    "FunctionInvoke",
                    {"stub_id": stub_id, "args": {"args": [x], "kwargs": {}}},
                ):
                    if msg.get("type") == "result":
                        status, result = msg.get("status"), msg.get("result")
                if status != "COMPLETE":
                    raise RuntimeError(f"map item failed: {result!r}")
                return result
            finally:
                await rpc.close()

        async def run_all(xs):
            return await asyncio.gather(*(one(x) for x in xs))

        return lt.run(run_all(list(inputs)))


def function(*dargs, **dkwargs):
    if dargs and callable(dargs[0]) and not dkwargs:
        return Function(dargs[0])
    return lambda fn: Function(fn, **dkwargs)


def schedule(*dargs, **dkwargs):
    interval = dkwargs.pop("interval_s", 60.0)

    def deco(fn):
        f = Function(fn, **dkwargs)
        f.stub_kind = "schedule"
        f.interval_s = interval
        return f

    if dargs and callable(dargs[0]):
        return deco(dargs[0])
    return deco


# ---------------------------------------------------------------------------
# @endpoint / @asgi
# ---------------------------------------------------------------------------


class Endpoint(RunnerAbstraction):
    stub_kind = "endpoint"

    def remote(self, timeout: float = 300.0, **kwargs) -> Any:
        stub_id = self.prepare_runtime()
        r = self.client.call(
            "EndpointService",
            "EndpointInvoke",
            {"stub_id": stub_id, "args": kwargs, "timeout": timeout},
            timeout=timeout + 30.0,
        )
        if r.get("status") != "COMPLETE":
            raise RuntimeError(f"endpoint invoke failed: {r!r}")
        result = r.get("result")
        if isinstance(result, dict) and "body" in result:
            body = result["body"]
            return body.get("result", body) if isinstance(body, dict) else body
        return result

    def serve(self, timeout: float = 600.0) -> str:
        stub_id = self.prepare_runtime(stub_type=f"{self.stub_kind}/serve", force=True)
        self.client.call("EndpointService", "StartEndpointServe", {"stub_id": stub_id, "timeout": timeout})
        return stub_id


def endpoint(*dargs, **dkwargs):
    if dargs and callable(dargs[0]) and not dkwargs:
        return Endpoint(dargs[0])
    return lambda fn: Endpoint(fn, **dkwargs)


def asgi(*dargs, **dkwargs):
    def deco(fn):
        e = Endpoint(fn, **dkwargs)
        e.stub_kind = "asgi"
        return e

    if dargs and callable(dargs[0]):
        return deco(dargs[0])
    return deco


def realtime(*dargs, **dkwargs):
    # realtime = asgi with websocket dispatch (reference groups them)
    return asgi(*dargs, **dkwargs)


# ---------------------------------------------------------------------------
# @task_queue
# ---------------------------------------------------------------------------


class TaskQueue(RunnerAbstraction):
    stub_kind = "taskqueue"

    def put(self, *args, **kwargs) -> str:
        stub_id = self.prepare_runtime()
        r = self.client.call(
            "TaskQueueService",
            "TaskQueuePut",
            {"stub_id": stub_id, "args": {"args": list(args), "kwargs": kwargs}},
        )
        if not r.get("ok"):
            raise RuntimeError(f"put failed: {r.get('error')}")
        return r["task_id"]

    def get_task(self, task_id: str) -> Dict[str, Any]:
        return self.client.call("GatewayService", "GetTaskResult", {"task_id": task_id})

    def wait(self, task_id: str, timeout: float = 300.0, poll: float = 0.05) -> Any:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            r = self.get_task(task_id)
            if r["status"] in ("COMPLETE", "ERROR", "CANCELLED", "TIMEOUT"):
                if r["status"] != "COMPLETE":
                    raise RuntimeError(f"task {task_id} {r['status']}: {r.get('result')!r}")
                return r.get("result")
            time.sleep(poll)
        raise TimeoutError(task_id)


def task_queue(*dargs, **dkwargs):
    if dargs and callable(dargs[0]) and not dkwargs:
        return TaskQueue(dargs[0])
    return lambda fn: TaskQueue(fn, **dkwargs)


# ---------------------------------------------------------------------------
# Pod / Sandbox
# ---------------------------------------------------------------------------


class Pod(RunnerAbstraction):
    stub_kind = "pod"

    def __init__(self, entrypoint: Optional[List[str]] = None, ports: Optional[List[int]] = None, **kwargs):
        super().__init__(func=None, **kwargs)
        self.entrypoint = entrypoint or []
        self.ports = ports or []
        self.name = kwargs.get("name") or "pod"

    def _stub_config(self) -> StubConfig:
        cfg = super()._stub_config()
        cfg.entry_point = list(self.entrypoint)
        cfg.ports = list(self.ports)
        if self.keep_warm_seconds == 10.0:
            cfg.keep_warm_seconds = -1  # pods default to run-until-stopped
        return cfg

    def create(self, timeout: float = 60.0) -> "PodInstance":
        stub_id = self.prepare_runtime(stub_type="pod/run", force=True)
        r = self.client.call("PodService", "CreatePod", {"stub_id": stub_id, "timeout": timeout}, timeout=timeout + 10.0)
        if not r.get("ok"):
            raise RuntimeError("pod did not become ready")
        return PodInstance(self.client, r["container_id"], r["addr"])


@dataclass
class LLMConfig:
    """LLM serving metadata for Service pods (reference: type.py:314
    LLMConfig): drives OpenAI-protocol routing (prefix affinity) and the
    token-pressure autoscaler target."""

    model_id: str = ""
    engine: str = ""
    served_model_name: str = ""
    context_length: int = 0
    metrics_path: str = "/metrics"
    slo_tier: str = ""


class Service(Pod):
    """Long-running serving pod (reference: abstractions/service.py:161
    Service(Pod)): a pod with a serving port, readiness semantics, and
    optional LLM config. The platform applies prefix-affinity routing and
    the LLM token-pressure autoscaler to llm-kind services.

    Usage::

        svc = Service(entrypoint=["python", "-m", "my_server"],
                      port=8000, llm=LLMConfig(model_id="my-model"),
                      gpu="MI355X")
        inst = svc.create()
    """

    stub_kind = "pod"

    def __init__(self, entrypoint: Optional[List[str]] = None, port: int = 8000,
                 llm: Optional[LLMConfig] = None, app_kind: str = "",
                 keep_warm_seconds: float = 600.0, **kwargs):
        super().__init__(entrypoint=entrypoint, ports=[port],
                         keep_warm_seconds=keep_warm_seconds, **kwargs)
        self.port = port
        self.llm = llm
        self.app_kind = app_kind or ("llm" if llm else "")
        self.name = kwargs.get("name") or "service"

    def _stub_config(self) -> StubConfig:
        cfg = super()._stub_config()
        cfg.keep_warm_seconds = self.keep_warm_seconds
        serving = {"app_kind": self.app_kind, "port": self.port}
        if self.llm is not None:
            serving["llm"] = {
                "model_id": self.llm.model_id,
                "engine": self.llm.engine,
                "served_model_name": self.llm.served_model_name,
                "context_length": self.llm.context_length,
                "metrics_path": self.llm.metrics_path,
                "slo_tier": self.llm.slo_tier,
            }
        cfg.extra["serving"] = serving
        return cfg


@dataclass
class PodInstance:
    client: Client
    container_id: str
    addr: str

    def terminate(self) -> None:
        self.client.call("PodService", "StopPod", {"container_id": self.container_id})


class Sandbox(Pod):
    stub_kind = "sandbox"

    def __init__(self, **kwargs):
        kwargs.setdefault("name", "sandbox")
        super().__init__(**kwargs)
        self.entrypoint = []  # sandbox always runs the PM

    def create(self, timeout: float = 60.0) -> "SandboxInstance":
        stub_id = self.prepare_runtime(stub_type="sandbox", force=True)
        r = self.client.call("PodService", "CreatePod", {"stub_id": stub_id, "timeout": timeout}, timeout=timeout + 10.0)
        if not r.get("ok"):
            raise RuntimeError("sandbox did not become ready")
        return SandboxInstance(self.client, r["container_id"], r["addr"])


class SandboxInstance:
    def __init__(self, client: Client, container_id: str, addr: str):
        self.client = client
        self.container_id = container_id
        self.addr = addr
        self.fs = SandboxFS(self)
        self.process = SandboxProcessAPI(self)

    def _call(self, method: str, payload: Dict[str, Any]) -> Any:
        payload = {"container_id": self.container_id, **payload}
        return self.client.call("PodService", method, payload, timeout=float(payload.get("timeout", 120.0)) + 15.0)

    def exec(self, command, timeout: float = 120.0, background: bool = False, cwd: str = "", env: Optional[Dict] = None) -> Dict:
        return self._call("SandboxExec", {"command": command, "timeout": timeout, "background": background, "cwd": cwd, "env": env})

    def run_code(self, code: str, timeout: float = 120.0) -> Dict:
        return self._call("SandboxRunCode", {"code": code, "timeout": timeout})

    def status(self) -> Dict:
        return self._call("SandboxStatus", {})

    def expose_port(self, port: int) -> str:
        return self._call("SandboxExposePort", {"port": port})["url"]

    def create_image_from_filesystem(self, path: str = ".", timeout: float = 300.0) -> str:
        """Snapshot the sandbox filesystem into a new image id usable by
        future stubs via Image.from_snapshot() (reference:
        sandbox.py:527-549)."""
        r = self._call("SandboxCreateImageFromFilesystem", {"path": path, "timeout": timeout})
        if r.get("error"):
            raise RuntimeError(r["error"])
        return r["image_id"]

    def snapshot_memory(self) -> Dict:
        return self._call("SandboxSnapshotMemory", {})

    def terminal(self, rows: int = 24, cols: int = 80) -> "SandboxTerminal":
        return SandboxTerminal(self, rows=rows, cols=cols)

    def run_container(self, command, image_id: str = "", cpu: int = 0,
                      memory_mb: int = 0, block_network: bool = False,
                      timeout: float = 120.0) -> Dict:
        """Run a nested ISOLATED container inside this sandbox (the
        Docker-in-sandbox analog): own namespaces + overlay rootfs
        (optionally from an imported image's layers)."""
        return self._call("SandboxRunContainer", {
            "command": command, "image_id": image_id, "cpu": cpu,
            "memory_mb": memory_mb, "block_network": block_network,
            "timeout": timeout,
        })

    def terminate(self) -> None:
        self.client.call("PodService", "StopPod", {"container_id": self.container_id})


class SandboxTerminal:
    """Interactive PTY session in a sandbox (reference: shell-over-dropbear;
    here the PM owns the pty — `b9 shell` uses the same RPCs)."""

    def __init__(self, inst: "SandboxInstance", rows: int = 24, cols: int = 80):
        self.inst = inst
        r = inst._call("SandboxTerminalOpen", {"rows": rows, "cols": cols})
        self.term_id = r["term_id"]

    def write(self, data) -> None:
        self.inst._call("SandboxTerminalWrite", {"term_id": self.term_id, "data": data})

    def read(self, timeout: float = 5.0):
        r = self.inst._call("SandboxTerminalRead", {"term_id": self.term_id, "timeout": timeout})
        return r.get("data") or b"", bool(r.get("closed"))

    def resize(self, rows: int, cols: int) -> None:
        self.inst._call("SandboxTerminalResize", {"term_id": self.term_id, "rows": rows, "cols": cols})

    def close(self) -> None:
        self.inst._call("SandboxTerminalClose", {"term_id": self.term_id})


class AsyncSandboxInstance:
    """Asyncio variant of SandboxInstance (reference: the SDK's sandbox
    async APIs, sdk sandbox.py sync+asyncio surfaces). Wraps the same
    PodService RPCs on the caller's event loop."""

    def __init__(self, config: Optional["ClientConfig"], container_id: str, addr: str):
        from .rpc.channel import RpcClient

        cfg = config or get_config()
        self._rpc = RpcClient(cfg.gateway_addr, token=cfg.token)
        self.container_id = container_id
        self.addr = addr

    async def _call(self, method: str, payload: Dict[str, Any]) -> Any:
        payload = {"container_id": self.container_id, **payload}
        return await self._rpc.call(
            "PodService", method, payload,
            timeout=float(payload.get("timeout", 120.0)) + 15.0,
        )

    async def exec(self, command, timeout: float = 120.0, background: bool = False,
                   cwd: str = "", env: Optional[Dict] = None) -> Dict:
        return await self._call("SandboxExec", {
            "command": command, "timeout": timeout, "background": background,
            "cwd": cwd, "env": env,
        })

    async def run_code(self, code: str, timeout: float = 120.0) -> Dict:
        return await self._call("SandboxRunCode", {"code": code, "timeout": timeout})

    async def status(self) -> Dict:
        return await self._call("SandboxStatus", {})

    async def upload(self, path: str, data: bytes) -> Dict:
        return await self._call("SandboxUploadFile", {"path": path, "data": data})

    async def download(self, path: str) -> bytes:
        return (await self._call("SandboxDownloadFile", {"path": path}))["data"]

    async def ls(self, path: str = ".") -> list:
        return await self._call("SandboxListFiles", {"path": path})

    async def expose_port(self, port: int, tls: bool = False) -> Dict:
        return await self._call("SandboxExposePort", {"port": port, "tls": tls})

    async def terminate(self) -> None:
        await self._rpc.call("PodService", "StopPod", {"container_id": self.container_id})

    async def close(self) -> None:
        await self._rpc.close()


class AsyncSandbox:
    """`await AsyncSandbox(...).create()` — async twin of Sandbox."""

    def __init__(self, **kwargs):
        self._sync = Sandbox(**kwargs)

    async def create(self, timeout: float = 60.0) -> AsyncSandboxInstance:
        import asyncio as _a

        def _mk():
            stub_id = self._sync.prepare_runtime(stub_type="sandbox", force=True)
            return self._sync.client.call(
                "PodService", "CreatePod", {"stub_id": stub_id, "timeout": timeout},
                timeout=timeout + 10.0,
            )

        r = await _a.to_thread(_mk)
        if not r.get("ok"):
            raise RuntimeError("sandbox did not become ready")
        return AsyncSandboxInstance(self._sync.client.config, r["container_id"], r["addr"])


class SandboxFS:
    def __init__(self, sb: SandboxInstance):
        self._sb = sb

    def upload(self, path: str, data: bytes) -> Dict:
        return self._sb._call("SandboxUploadFile", {"path": path, "data": data})

    def download(self, path: str) -> bytes:
        return self._sb._call("SandboxDownloadFile", {"path": path})["data"]

    def ls(self, path: str = ".") -> List[Dict]:
        return self._sb._call("SandboxListFiles", {"path": path})

    def stat(self, path: str) -> Dict:
        return self._sb._call("SandboxStatFile", {"path": path})

    def rm(self, path: str) -> None:
        self._sb._call("SandboxDeleteFile", {"path": path})

    def replace_in_files(self, path: str, old: str, new: str, pattern: str = "*") -> Dict:
        return self._sb._call("SandboxReplaceInFiles", {"path": path, "old": old, "new": new, "pattern": pattern})

    def find_in_files(self, path: str, pattern: str) -> List[Dict]:
        return self._sb._call("SandboxFindInFiles", {"path": path, "pattern": pattern})


class SandboxProcessAPI:
    def __init__(self, sb: SandboxInstance):
        self._sb = sb

    def list(self) -> List[Dict]:
        return self._sb._call("SandboxListProcesses", {})

    def kill(self, pid: int, sig: int = 9) -> bool:
        return self._sb._call("SandboxKillProcess", {"pid": pid, "signal": sig})

    def status(self, pid: int) -> Dict:
        return self._sb._call("SandboxStatus", {"pid": pid})


# ---------------------------------------------------------------------------
# Map / Queue / Signal client objects
# ---------------------------------------------------------------------------


class Bot:
    """Petri-net agent orchestration (reference: experimental/bot — SDK Bot
    with locations + transitions; each transition firing runs as a real
    function container).

    Usage::

        bot = Bot(name="research")
        bot.location("queries", marker={"q": "str"})
        bot.location("answers", marker={"text": "str"})

        @bot.transition(inputs={"queries": 1}, outputs=["answers"])
        def answer(markers):
            q = markers["queries"][0]["q"]
            return {"markers": {"answers": [{"text": f"answer to {q}"}]}}

        session = bot.create_session()
        session.push("queries", {"q": "why is the sky blue"})
        for ev in session.events(timeout=30):
            ...
    """

    def __init__(self, name: str = "bot", model: str = "", welcome_message: str = "",
                 client: Optional[Client] = None):
        self.name = name
        self.model = model
        self.welcome_message = welcome_message
        self.client = client
        self._locations: Dict[str, Dict[str, Any]] = {}
        self._transitions: Dict[str, Dict[str, Any]] = {}

    def location(self, name: str, marker: Optional[Dict[str, str]] = None, expose: bool = True):
        self._locations[name] = {"name": name, "marker": marker or {}, "expose": expose}
        return self

    def transition(self, inputs: Dict[str, int], outputs: Optional[List[str]] = None,
                   cpu: int = 1000, memory: int = 1024, gpu: str = "",
                   timeout: float = 600.0, description: str = "", confirm: bool = False):
        def deco(fn):
            self._transitions[fn.__name__] = {
                "name": fn.__name__,
                "handler": _handler_spec(fn),
                "inputs": inputs,
                "outputs": outputs or [],
                "cpu": cpu, "memory": memory, "gpu": gpu,
                "timeout": timeout, "description": description, "confirm": confirm,
            }
            return fn
        return deco

    def create_session(self, keep_warm_seconds: float = 10.0) -> "BotSessionClient":
        client = self.client or Client()
        r = client.call("BotService", "CreateBotSession", {
            "name": self.name,
            "keep_warm_seconds": keep_warm_seconds,
            "config": {
                "model": self.model,
                "welcome_message": self.welcome_message,
                "locations": self._locations,
                "transitions": self._transitions,
            },
        })
        return BotSessionClient(client, r["session_id"])


class BotSessionClient:
    def __init__(self, client: Client, session_id: str):
        self.client = client
        self.session_id = session_id
        self._offset = 0

    def push(self, location: str, *markers: Dict[str, Any]) -> None:
        r = self.client.call("BotService", "PushBotMarkers",
                             {"session_id": self.session_id, "markers": {location: list(markers)}})
        if not r.get("ok"):
            raise RuntimeError(r.get("error", "push failed"))

    def send_event(self, event: Dict[str, Any]) -> None:
        self.client.call("BotService", "PushBotEvent",
                         {"session_id": self.session_id, "event": event})

    def confirm(self, transition: str) -> None:
        self.send_event({"type": "confirm_transition", "transition": transition})

    def state(self) -> Dict[str, Any]:
        return self.client.call("BotService", "GetBotState", {"session_id": self.session_id})

    def events(self, timeout: float = 10.0):
        """Yield events from the current offset; returns when a poll window
        passes with no new events."""
        import time as _time

        deadline = _time.monotonic() + timeout
        while _time.monotonic() < deadline:
            r = self.client.call(
                "BotService", "PopBotEvents",
                {"session_id": self.session_id, "offset": self._offset,
                 "timeout": min(5.0, max(0.1, deadline - _time.monotonic()))},
                timeout=40.0,
            )
            evs = r["events"]
            self._offset = r["next_offset"]
            for ev in evs:
                yield ev
            if evs:
                deadline = _time.monotonic() + timeout


class MapClient:
    def __init__(self, name: str, client: Optional[Client] = None):
        self.name = name
        self.client = client or Client()

    def set(self, key: str, value: Any) -> None:
        self.client.call("MapService", "MapSet", {"name": self.name, "key": key, "value": value})

    def get(self, key: str) -> Any:
        return self.client.call("MapService", "MapGet", {"name": self.name, "key": key})

    def delete(self, key: str) -> None:
        self.client.call("MapService", "MapDelete", {"name": self.name, "key": key})

    def keys(self) -> List[str]:
        return self.client.call("MapService", "MapKeys", {"name": self.name})

    def __len__(self) -> int:
        return self.client.call("MapService", "MapCount", {"name": self.name})

    __setitem__ = set
    __getitem__ = get


class QueueClient:
    def __init__(self, name: str, client: Optional[Client] = None):
        self.name = name
        self.client = client or Client()

    def put(self, value: Any) -> None:
        self.client.call("SimpleQueueService", "SimpleQueuePut", {"name": self.name, "value": value})

    def pop(self, timeout: float = 0.0) -> Any:
        r = self.client.call(
            "SimpleQueueService", "SimpleQueuePop", {"name": self.name, "timeout": timeout},
            timeout=timeout + 10.0,
        )
        return r["value"] if r["ok"] else None

    def empty(self) -> bool:
        return self.client.call("SimpleQueueService", "SimpleQueueEmpty", {"name": self.name})

    def __len__(self) -> int:
        return self.client.call("SimpleQueueService", "SimpleQueueSize", {"name": self.name})


class Output:
    """Task output files: save -> retrieval URL (reference:
    pkg/abstractions/output — SDK Output surface)."""

    def __init__(self, task_id: str = "", client: Optional[Client] = None):
        self.task_id = task_id or os.environ.get("TASK_ID", "")
        self.client = client or Client()

    def save(self, data: bytes, filename: str) -> Dict[str, Any]:
        chunks = [
            {"task_id": self.task_id, "filename": filename, "data": data[i : i + (1 << 20)]}
            for i in range(0, max(len(data), 1), 1 << 20)
        ] or [{"task_id": self.task_id, "filename": filename, "data": b""}]
        return self.client.client_stream("OutputService", "OutputSaveStream", chunks)

    def save_file(self, path: str, filename: str = "") -> Dict[str, Any]:
        with open(path, "rb") as f:
            return self.save(f.read(), filename or os.path.basename(path))

    def public_url(self, path: str, ttl: float = 3600.0) -> str:
        return self.client.call("OutputService", "OutputPublicURL", {"path": path, "ttl": ttl})["url"]

    def stat(self, path: str) -> Dict[str, Any]:
        return self.client.call("OutputService", "OutputStat", {"path": path})


class SignalClient:
    def __init__(self, name: str, client: Optional[Client] = None):
        self.name = name
        self.client = client or Client()

    def set(self, ttl: float = 0.0) -> None:
        self.client.call("SignalService", "SignalSet", {"name": self.name, "ttl": ttl})

    def clear(self) -> None:
        self.client.call("SignalService", "SignalClear", {"name": self.name})

    def is_set(self) -> bool:
        return self.client.call("SignalService", "SignalMonitor", {"name": self.name})
