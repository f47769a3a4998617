"""Core domain types — the data contracts shared by SDK, gateway, scheduler
and worker.

These reproduce the reference's field-for-field contracts (reference:
pkg/types/scheduler.go:255-298 ContainerRequest, pkg/types/backend.go:526-567
StubConfigV1, pkg/types/scheduler.go:58-80 Worker, pkg/types/worker.go:52-60
Mount, pkg/types/backend.go:827-839 Autoscaler, backend.go:444-464 Checkpoint,
backend.go:889-911 stub types) re-expressed for the MI355X device plane:
GPU requests name AMD device sets, checkpoints record the hipsnap accelerator
payload, and multi-GPU requests carry an RCCL rendezvous spec.
"""

from __future__ import annotations

import time
import uuid
from enum import Enum
from typing import Any, Dict, List, Optional

from pydantic import BaseModel, Field

GPU_TYPE_MI355X = "MI355X"
GPU_TYPE_ANY = "any"

# GPU preference table (reference: pkg/types/gpu.go:80-140 keeps a priced
# priority order of NVIDIA SKUs; ours is the AMD Instinct line).
GPU_PRIORITY_ORDER = ["MI355X", "MI350X", "MI325X", "MI300X", "any"]


def new_id(prefix: str = "") -> str:
    v = uuid.uuid4().hex[:12]
    return f"{prefix}{v}" if prefix else v


class StubType(str, Enum):
    function = "function"
    function_deployment = "function/deployment"
    function_serve = "function/serve"
    container = "container"
    shell = "shell"
    taskqueue = "taskqueue"
    taskqueue_deployment = "taskqueue/deployment"
    taskqueue_serve = "taskqueue/serve"
    endpoint = "endpoint"
    endpoint_deployment = "endpoint/deployment"
    endpoint_serve = "endpoint/serve"
    asgi = "asgi"
    asgi_deployment = "asgi/deployment"
    asgi_serve = "asgi/serve"
    schedule = "schedule"
    schedule_deployment = "schedule/deployment"
    pod = "pod"
    pod_deployment = "pod/deployment"
    pod_run = "pod/run"
    sandbox = "sandbox"
    image_build = "image/build"
    bot = "bot"
    bot_deployment = "bot/deployment"

    @property
    def kind(self) -> str:
        return self.value.split("/")[0]

    @property
    def is_deployment(self) -> bool:
        return self.value.endswith("/deployment")

    @property
    def is_serve(self) -> bool:
        return self.value.endswith("/serve")


class ContainerStatus(str, Enum):
    pending = "PENDING"
    running = "RUNNING"
    stopping = "STOPPING"


class WorkerStatus(str, Enum):
    available = "available"
    pending = "pending"
    disabled = "disabled"


class TaskStatus(str, Enum):
    pending = "PENDING"
    running = "RUNNING"
    complete = "COMPLETE"
    error = "ERROR"
    cancelled = "CANCELLED"
    timeout = "TIMEOUT"
    retry = "RETRY"

    @property
    def is_terminal(self) -> bool:
        return self in (
            TaskStatus.complete,
            TaskStatus.error,
            TaskStatus.cancelled,
            TaskStatus.timeout,
        )


class MountType(str, Enum):
    volume = "volume"
    bind = "bind"
    secret = "secret"


class Mount(BaseModel):
    local_path: str
    mount_path: str
    link_path: str = ""
    read_only: bool = False
    mount_type: MountType = MountType.volume


class AutoscalerType(str, Enum):
    queue_depth = "queue_depth"
    llm_token_pressure = "llm_token_pressure"


class AutoscalerConfig(BaseModel):
    type: AutoscalerType = AutoscalerType.queue_depth
    max_containers: int = 1
    tasks_per_container: int = 1
    min_containers: int = 0


class TaskPolicy(BaseModel):
    max_retries: int = 3
    timeout: float = 3600.0
    ttl: float = 0.0


class RendezvousSpec(BaseModel):
    """RCCL-over-xGMI rendezvous for gpu_count>1 containers.

    The worker provisions this before exec: one process per GPU inside the
    container, torch.distributed over RCCL ("nccl" backend on ROCm), loopback
    rendezvous. (The reference has no equivalent — multi-GPU containers just
    see N devices)
    """

    master_addr: str = "127.0.0.1"
    master_port: int = 0  # allocated by the worker
    world_size: int = 1
    device_ordinals: List[int] = Field(default_factory=list)
    xgmi_connected: bool = True
    env: Dict[str, str] = Field(default_factory=dict)


class RuntimeConfig(BaseModel):
    cpu: int = 1000  # millicores
    memory_mb: int = 1024
    gpu: str = ""  # requested GPU type, "" = CPU-only
    gpus: List[str] = Field(default_factory=list)  # ordered preference
    gpu_count: int = 0
    # Fractional GPU sharing (reference: Thunder virtualized pools,
    # pkg/worker/thunder.go:31-138 + selection worker.go:128-137). On
    # MI355X there is no external virtualization service: HIP natively
    # time-shares one GCD across processes (HWS firmware scheduler), so
    # the worker packs fractional containers onto a shared device and
    # enforces the HBM budget per process (torch per-process memory
    # fraction + arena sizing). 1.0 = exclusive whole-GPU.
    gpu_fraction: float = 1.0
    image_id: str = ""


class CheckpointStatus(str, Enum):
    available = "available"
    checkpoint_in_progress = "in_progress"
    checkpoint_failed = "failed"
    restore_failed = "restore_failed"
    not_found = "not_found"


class CheckpointState(BaseModel):
    """Checkpoint row (reference: pkg/types/backend.go:444-464). The
    accelerator payload is a hipsnap HBM image, not CRIU/cuda-checkpoint."""

    checkpoint_id: str = ""
    stub_id: str = ""
    workspace_id: str = ""
    source_container_id: str = ""
    status: CheckpointStatus = CheckpointStatus.not_found
    remote_key: str = ""
    stub_type: str = ""
    accelerator: str = ""  # GPU type recorded — host-compat gate for restore
    gpu_count: int = 0
    size_bytes: int = 0
    created_at: float = 0.0
    # durable tier: content-store digests so any node can materialize the
    # snapshot over the cache transport (reference: checkpoint cache_hash,
    # pkg/types/backend.go:444-464)
    cache_digest: str = ""
    obj_digest: str = ""



class PricingPolicy(BaseModel):
    """Per-task billing policy on public stubs (reference:
    sdk type.py:431-441 PricingPolicy, pkg/abstractions/common/usage.go
    TrackTaskCost): cost per task or per task-duration-ms, metered into
    workspace usage counters at task end."""

    max_in_flight: int = 10
    cost_model: str = "task"  # task | duration
    cost_per_task: float = 0.0
    cost_per_task_duration_ms: float = 0.0


class StubConfig(BaseModel):
    """SDK <-> gateway stub contract, stored as JSON on the stub row
    (reference: pkg/types/backend.go:526-567 StubConfigV1)."""

    runtime: RuntimeConfig = Field(default_factory=RuntimeConfig)
    handler: str = ""  # "module:function"
    on_start: str = ""
    on_deploy: str = ""
    python_version: str = ""
    keep_warm_seconds: float = 10.0
    max_pending_tasks: int = 100
    callback_url: str = ""
    task_policy: TaskPolicy = Field(default_factory=TaskPolicy)
    workers: int = 1
    concurrent_requests: int = 1
    authorized: bool = True
    volumes: List[Mount] = Field(default_factory=list)
    secrets: List[str] = Field(default_factory=list)
    env: Dict[str, str] = Field(default_factory=dict)
    autoscaler: AutoscalerConfig = Field(default_factory=AutoscalerConfig)
    extra: Dict[str, Any] = Field(default_factory=dict)
    checkpoint_enabled: bool = False
    work_dir: str = ""
    entry_point: List[str] = Field(default_factory=list)
    ports: List[int] = Field(default_factory=list)
    inputs: Optional[Dict[str, Any]] = None
    outputs: Optional[Dict[str, Any]] = None
    block_network: bool = False
    pool: str = ""
    timeout: float = 3600.0
    pricing: Optional[PricingPolicy] = None


class BuildSpec(BaseModel):
    """Build-container payload. Build steps execute INSIDE a scheduled
    container — never on the gateway host (reference:
    pkg/abstractions/image/build.go:62-343 build containers into the build
    pool). mode "prefix" = host-base image (site/files dirs bound at
    /b9img); mode "rootfs" = layered OCI-style build (base image layers as
    overlayfs lowerdirs, host-backed upper captured as the new layer)."""

    image_id: str = ""
    mode: str = "prefix"  # prefix | rootfs
    base_image_id: str = ""
    build_dir: str = ""  # host dir receiving prefix content / upper delta
    steps: List[Dict[str, Any]] = Field(default_factory=list)


class ContainerRequest(BaseModel):
    """Scheduler unit of work (reference: pkg/types/scheduler.go:255-298)."""

    container_id: str = ""
    entry_point: List[str] = Field(default_factory=list)
    env: Dict[str, str] = Field(default_factory=dict)
    cpu: int = 1000  # millicores
    memory_mb: int = 1024
    gpu: str = ""
    gpu_request: List[str] = Field(default_factory=list)  # ordered preference
    gpu_count: int = 0
    gpu_fraction: float = 1.0  # <1.0 = share a device (see RuntimeConfig)
    image_id: str = ""
    stub_id: str = ""
    workspace_id: str = ""
    stub_type: str = ""
    timestamp: float = Field(default_factory=time.time)
    mounts: List[Mount] = Field(default_factory=list)
    retry_count: int = 0
    pool_selector: str = ""
    preemptable: bool = False
    checkpoint_enabled: bool = False
    ports: List[int] = Field(default_factory=list)
    app_id: str = ""
    checkpoint: Optional[CheckpointState] = None
    block_network: bool = False
    machine_id: str = ""  # pin
    task_id: str = ""
    delivery_token: str = ""
    hostname: str = ""
    rendezvous: Optional[RendezvousSpec] = None
    build: Optional[BuildSpec] = None

    def requires_gpu(self) -> bool:
        return bool(self.gpu) or bool(self.gpu_request) or self.gpu_count > 0

    def gpu_units(self) -> float:
        """Scheduling demand in GPU units: whole devices, or the shared
        fraction for gpu_fraction<1 requests (only valid with one GPU)."""
        if not self.requires_gpu():
            return 0.0
        count = max(1, self.gpu_count)
        if self.gpu_fraction < 1.0 and count == 1:
            return self.gpu_fraction
        return float(count)


class Worker(BaseModel):
    """Worker state snapshot (reference: pkg/types/scheduler.go:58-80)."""

    id: str = ""
    status: WorkerStatus = WorkerStatus.pending
    total_cpu: int = 0
    total_memory_mb: int = 0
    total_gpu_count: int = 0
    free_cpu: int = 0
    free_memory_mb: int = 0
    # float: fractional (shared-device) containers subtract their share
    free_gpu_count: float = 0
    gpu: str = ""  # GPU type on this worker
    pool_name: str = ""
    machine_id: str = ""
    resource_version: int = 0
    requires_pool_selector: bool = False
    priority: int = 0
    preemptable: bool = False
    active_containers: int = 0
    cordoned: bool = False
    addr: str = ""  # worker ContainerService address


class ContainerState(BaseModel):
    container_id: str = ""
    stub_id: str = ""
    workspace_id: str = ""
    status: ContainerStatus = ContainerStatus.pending
    worker_id: str = ""
    started_at: float = 0.0
    scheduled_at: float = 0.0
    gpu: str = ""
    gpu_count: int = 0
    exit_code: Optional[int] = None


class Task(BaseModel):
    task_id: str = ""
    stub_id: str = ""
    workspace_id: str = ""
    status: TaskStatus = TaskStatus.pending
    container_id: str = ""
    created_at: float = Field(default_factory=time.time)
    started_at: float = 0.0
    ended_at: float = 0.0
    retries: int = 0
    policy: TaskPolicy = Field(default_factory=TaskPolicy)
    executor: str = ""  # stub kind that dispatches this task


class LifecycleSpan(BaseModel):
    """Cold-start phase instrumentation (reference: pkg/types/event.go:548-700
    container-lifecycle spans — the de-facto profiler, and our benchmark's
    phase decomposition)."""

    name: str
    container_id: str = ""
    start: float = 0.0
    end: float = 0.0
    parent: str = ""
    meta: Dict[str, Any] = Field(default_factory=dict)

    @property
    def duration_ms(self) -> float:
        return (self.end - self.start) * 1000.0
