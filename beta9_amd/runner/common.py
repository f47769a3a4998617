"""Runner shared machinery: env contract, gateway client, handler loading,
and the MI355X GPU boot path.

Reference behavior (reference: sdk/src/beta9/runner/common.py —
FunctionHandler user-code loader, env contract SURVEY.md §A.10, checkpoint
READY sentinel :618-640). Containers are gateway clients: they dial the
gateway gRPC with B9_TOKEN for task claim/args/result (the reference's
runners do exactly this).

MI355X boot path (new vs reference): when the container owns GPUs, the
runner pre-initializes the HIP context and pre-allocates the HBM arena
(hipsnap, beta9_amd/ops) BEFORE importing the handler, so model-weight
loading streams straight into reserved HBM through pinned staging — and on a
checkpoint restore the arena is refilled from the snapshot instead of
re-running handler setup.
"""

from __future__ import annotations

import asyncio
import importlib
import logging
import os
import sys
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from ..rpc.channel import RpcClient

log = logging.getLogger("b9.runner")


@dataclass
class RunnerEnv:
    token: str = ""
    gateway_host: str = "127.0.0.1"
    gateway_port: int = 0
    stub_id: str = ""
    stub_type: str = ""
    workspace_id: str = ""
    container_id: str = ""
    task_id: str = ""
    handler: str = ""
    on_start: str = ""
    workers: int = 1
    concurrent_requests: int = 1
    keep_warm_seconds: float = 10.0
    bind_port: int = 0
    timeout: float = 3600.0
    checkpoint_enabled: bool = False
    checkpoint_key: str = ""
    workspace_dir: str = ""
    gpu_devices: List[int] = field(default_factory=list)
    callback_url: str = ""

    @property
    def gateway_addr(self) -> str:
        return f"{self.gateway_host}:{self.gateway_port}"

    @classmethod
    def from_env(cls) -> "RunnerEnv":
        e = os.environ
        gpus = [int(x) for x in e.get("WORKER_GPU_DEVICES", "").split(",") if x != ""]
        return cls(
            token=e.get("B9_TOKEN", ""),
            gateway_host=e.get("B9_GATEWAY_HOST", "127.0.0.1"),
            gateway_port=int(e.get("B9_GATEWAY_PORT", "0") or 0),
            stub_id=e.get("STUB_ID", ""),
            stub_type=e.get("STUB_TYPE", ""),
            workspace_id=e.get("WORKSPACE_ID", ""),
            container_id=e.get("CONTAINER_ID", ""),
            task_id=e.get("TASK_ID", ""),
            handler=e.get("HANDLER", ""),
            on_start=e.get("ON_START", ""),
            workers=int(e.get("WORKERS", "1") or 1),
            concurrent_requests=int(e.get("CONCURRENT_REQUESTS", "1") or 1),
            keep_warm_seconds=float(e.get("KEEP_WARM_SECONDS", "10") or 10),
            bind_port=int(e.get("BIND_PORT", "0") or 0),
            timeout=float(e.get("TIMEOUT", "3600") or 3600),
            checkpoint_enabled=e.get("CHECKPOINT_ENABLED", "0") == "1",
            checkpoint_key=e.get("B9_CHECKPOINT_KEY", ""),
            workspace_dir=e.get("B9_WORKSPACE_DIR", ""),
            gpu_devices=gpus,
            callback_url=e.get("CALLBACK_URL", ""),
        )


def setup_environment(env: RunnerEnv) -> None:
    """Workspace code + image prefix onto sys.path; chdir to workspace."""
    site = os.environ.get("B9_IMAGE_SITE", "")
    if site and site not in sys.path:
        sys.path.insert(0, site)
    if env.workspace_dir and os.path.isdir(env.workspace_dir):
        if env.workspace_dir not in sys.path:
            sys.path.insert(0, env.workspace_dir)
        os.chdir(env.workspace_dir)
    elif os.getcwd() == "/":
        # native runtime pivots to "/": give the runner a private scratch
        # workspace instead of the container root
        ws = "/tmp/workspace"
        os.makedirs(ws, exist_ok=True)
        os.chdir(ws)
        sys.path.insert(0, ws)
    else:
        sys.path.insert(0, os.getcwd())


_gpu_boot: Optional["GPUBoot"] = None


def get_gpu_boot() -> Optional["GPUBoot"]:
    """The container's GPU boot context (arena + restore state); handlers use
    this to rebuild models from a restored HBM snapshot."""
    return _gpu_boot


class GPUBoot:
    """HIP pre-init + HBM arena + snapshot restore for GPU containers.

    The phases here are the MI355X cold-start levers the whole design aims
    at: hip_preinit (context creation overlapped with handler import),
    hbm_prealloc (one big reserved arena instead of incremental hipMalloc),
    hbm_restore (pinned double-buffered stream of a hipsnap snapshot).
    """

    def __init__(self, env: RunnerEnv):
        self.env = env
        self.arena = None
        self.restored = False
        self.phases: Dict[str, float] = {}

    def boot(self) -> None:
        global _gpu_boot
        _gpu_boot = self
        if not self.env.gpu_devices:
            return
        t0 = time.perf_counter()
        try:
            import torch

            if not torch.cuda.is_available():
                log.warning("GPU devices assigned but torch.cuda unavailable")
                return
            torch.cuda.init()
            self.phases["hip_preinit_ms"] = (time.perf_counter() - t0) * 1000
            # Shared-GPU (fractional) container: enforce the HBM budget the
            # worker assigned — HIP time-shares the GCD natively, memory is
            # the resource that needs fencing (Thunder-analog; see
            # GPUManager._assign_fraction).
            frac_s = os.environ.get("B9_GPU_MEMORY_FRACTION", "")
            if frac_s:
                frac = max(0.01, min(1.0, float(frac_s)))
                torch.cuda.set_per_process_memory_fraction(frac, 0)
                self.phases["gpu_memory_fraction"] = frac
            t1 = time.perf_counter()
            from ..ops import hipsnap

            ckpt_dir = os.environ.get("B9_CHECKPOINT_DIR", "")
            size = 0
            snap_path = ""
            if self.env.checkpoint_key and ckpt_dir:
                snap_path = hipsnap.snapshot_path(ckpt_dir, self.env.checkpoint_key)
                meta = hipsnap.peek_meta(snap_path)
                if meta is not None:
                    total = sum(r["nbytes"] for r in meta["regions"])
                    size = int(total * 1.25) + (64 << 20)
            if frac_s:
                # the arena is raw hipMalloc (outside the torch allocator
                # that set_per_process_memory_fraction fences) — cap it to
                # the budgeted share of HBM explicitly
                total = torch.cuda.get_device_properties(0).total_memory
                budget = int(total * float(self.phases["gpu_memory_fraction"]) * 0.5)
                size = min(size or hipsnap.DEFAULT_ARENA_BYTES, budget)
            # prefer the prewarm shell's pre-created arena (hipMalloc +
            # pinned staging already paid before the container started)
            self.arena = hipsnap.take_prewarmed(min_size=size)
            if self.arena is None:
                self.arena = hipsnap.Arena(device=0, size_bytes=size)
            else:
                self.phases["arena_prewarmed"] = 1.0
            self.phases["hbm_prealloc_ms"] = (time.perf_counter() - t1) * 1000
            if snap_path:
                t2 = time.perf_counter()
                self.restored = hipsnap.restore_snapshot(self.arena, snap_path)
                self.phases["hbm_restore_ms"] = (time.perf_counter() - t2) * 1000
                if self.restored:
                    t3 = time.perf_counter()
                    self._load_object_state(snap_path)
                    self.phases["obj_restore_ms"] = (time.perf_counter() - t3) * 1000
        except Exception:
            log.exception("GPU boot failed; continuing CPU-only")

    def _load_object_state(self, snap_path: str) -> None:
        """Rehydrate the pickled handler objects over the restored arena so
        on_start is skipped (objsnap; VERDICT round-1 item #4)."""
        global _restored_state
        obj_path = snap_path + ".obj"
        if not os.path.exists(obj_path):
            return
        try:
            from ..ops import objsnap

            with open(obj_path, "rb") as f:
                state = objsnap.load_state(f.read(), self.arena)
            if isinstance(state, dict):
                _restored_state = state
                self.phases["obj_restored"] = float(len(state))
        except Exception:
            log.exception("object-state restore failed; on_start will run")


class HandlerError(Exception):
    pass


def load_callable(spec: str) -> Callable:
    """Load "module:function" (reference: FunctionHandler in runner/common.py)."""
    if ":" not in spec:
        raise HandlerError(f"handler spec {spec!r} must be module:function")
    module_name, fn_name = spec.split(":", 1)
    module = importlib.import_module(module_name)
    fn = module
    for part in fn_name.split("."):
        fn = getattr(fn, part)
    if not callable(fn):
        raise HandlerError(f"{spec} is not callable")
    return fn


def unwrap_handler(fn: Callable) -> Callable:
    """SDK decorators wrap the user function; executing in-container we want
    the raw callable (the decorator objects expose .func)."""
    seen = set()
    while hasattr(fn, "func") and id(fn) not in seen:
        seen.add(id(fn))
        fn = fn.func
    return fn


async def call_handler(fn: Callable, args: Any, context: Optional[Dict] = None) -> Any:
    """Invoke a user handler with the SDK calling convention:
    payload {"args": [...], "kwargs": {...}} or a plain dict of kwargs."""
    fn = unwrap_handler(fn)
    if isinstance(args, dict) and ("args" in args or "kwargs" in args):
        pos = args.get("args", []) or []
        kw = args.get("kwargs", {}) or {}
    elif isinstance(args, dict):
        pos, kw = [], args
    elif args is None:
        pos, kw = [], {}
    else:
        pos, kw = [args], {}
    if asyncio.iscoroutinefunction(fn):
        return await fn(*pos, **kw)
    loop = asyncio.get_event_loop()
    import functools

    return await loop.run_in_executor(None, functools.partial(fn, *pos, **kw))


class RunnerClient:
    """Gateway client for the runner loop."""

    def __init__(self, env: RunnerEnv):
        self.env = env
        self.rpc = RpcClient(env.gateway_addr, token=env.token)
        self._hb_task: Optional[asyncio.Task] = None

    async def close(self) -> None:
        await self.rpc.close()

    async def start_task(self, task_id: str) -> bool:
        r = await self.rpc.call(
            "RunnerService", "StartTask", {"task_id": task_id, "container_id": self.env.container_id}
        )
        return bool(r.get("ok"))

    async def get_args(self, task_id: str) -> Any:
        return await self.rpc.call("RunnerService", "GetTaskArgs", {"task_id": task_id})

    async def end_task(self, task_id: str, result: Any, status: str = "COMPLETE") -> None:
        await self.rpc.call(
            "RunnerService", "EndTask", {"task_id": task_id, "result": result, "status": status}
        )

    async def heartbeat(self, task_id: str) -> None:
        await self.rpc.call("RunnerService", "Heartbeat", {"task_id": task_id})

    async def is_cancelled(self, task_id: str) -> bool:
        return bool(await self.rpc.call("RunnerService", "IsCancelled", {"task_id": task_id}))

    async def set_keep_warm(self) -> None:
        await self.rpc.call(
            "RunnerService",
            "SetKeepWarm",
            {
                "stub_id": self.env.stub_id,
                "container_id": self.env.container_id,
                "ttl": self.env.keep_warm_seconds,
            },
        )

    async def report_pressure(self, pressure: float) -> None:
        """LLM containers report token pressure for the
        llm_token_pressure autoscaler."""
        await self.rpc.call(
            "RunnerService",
            "ReportPressure",
            {
                "stub_id": self.env.stub_id,
                "container_id": self.env.container_id,
                "pressure": pressure,
            },
        )

    async def checkpoint_ready(self) -> Dict[str, Any]:
        """Signal on_start done; gateway may grant this container the
        checkpoint (single-flight per stub)."""
        return await self.rpc.call(
            "RunnerService",
            "CheckpointReady",
            {"stub_id": self.env.stub_id, "container_id": self.env.container_id},
        )

    async def commit_checkpoint(self, key: str, size_bytes: int, gpu_count: int = 1) -> None:
        await self.rpc.call(
            "RunnerService",
            "CommitCheckpoint",
            {
                "stub_id": self.env.stub_id,
                "container_id": self.env.container_id,
                "key": key,
                "size_bytes": size_bytes,
                "gpu_count": gpu_count,
            },
        )

    async def take_checkpoint_if_granted(self, gpu_boot: "GPUBoot") -> bool:
        """The full runner-side checkpoint flow: ask, snapshot the arena,
        commit. Returns True if a checkpoint was written."""
        try:
            grant = await self.checkpoint_ready()
        except Exception:
            log.warning("checkpoint-ready signal failed", exc_info=True)
            return False
        if not isinstance(grant, dict) or not grant.get("take"):
            return False
        arena = gpu_boot.arena if gpu_boot else None
        if arena is None or not getattr(arena, "native", False):
            return False
        from ..ops import hipsnap

        info = hipsnap.snapshot(arena, grant["dir"], grant["key"])
        # object-level state rides next to the arena snapshot so restore
        # can skip on_start entirely
        if _checkpoint_state:
            try:
                from ..ops import objsnap

                data = objsnap.save_state(dict(_checkpoint_state), arena)
                with open(
                    hipsnap.snapshot_path(grant["dir"], grant["key"]) + ".obj", "wb"
                ) as f:
                    f.write(data)
            except Exception:
                log.exception("object-state snapshot failed (arena-only checkpoint)")
        await self.commit_checkpoint(grant["key"], info["bytes"])
        log.info(
            "checkpoint %s written: %.1f GB at %.1f GB/s",
            grant["key"], info["bytes"] / 1e9, info["gbps"],
        )
        return True

    def start_heartbeat(self, task_id: str, interval: float = 5.0) -> None:
        async def loop() -> None:
            while True:
                try:
                    await self.heartbeat(task_id)
                except Exception:
                    pass
                await asyncio.sleep(interval)

        self._hb_task = asyncio.create_task(loop())

    def stop_heartbeat(self) -> None:
        if self._hb_task is not None:
            self._hb_task.cancel()
            self._hb_task = None


async def run_on_start(env: RunnerEnv) -> Any:
    """Run on_start and return its context value (reference: on_start result
    passed to handlers via context). After an object-level checkpoint
    restore (objsnap) on_start is SKIPPED entirely — the model objects and
    the on_start return value came back with the snapshot (reference flow
    this replaces: CRIU restore resumes past on_start, criu.go:512-651)."""
    if _restored_state is not None:
        return _restored_state.get("__on_start_value__")
    if not env.on_start:
        return None
    fn = unwrap_handler(load_callable(env.on_start))
    if asyncio.iscoroutinefunction(fn):
        value = await fn()
    else:
        value = fn()
    register_checkpoint_state("__on_start_value__", value)
    return value


# -- object-level checkpoint state (objsnap) --------------------------------
# Handlers register their GPU-resident objects (models, KV caches); the
# checkpoint pickles them with arena-tensor refs; restore hands the SAME
# objects back and on_start never runs.
_checkpoint_state: Dict[str, Any] = {}
_restored_state: Optional[Dict[str, Any]] = None


def register_checkpoint_state(name: str, obj: Any) -> None:
    _checkpoint_state[name] = obj


def get_checkpoint_state(name: str) -> Any:
    if _restored_state is not None and name in _restored_state:
        return _restored_state[name]
    return _checkpoint_state.get(name)


def state_restored() -> bool:
    return _restored_state is not None
