"""Typed application config tree.

Single AppConfig for every subsystem, loaded from embedded defaults and
overlaid by CONFIG_PATH (YAML file) or CONFIG_JSON (inline JSON), mirroring the
reference's config system (reference: pkg/common/config.go,
pkg/common/config.default.yaml) so dev/test/prod installs stay uniform.
"""

from __future__ import annotations

import json
import os
from typing import Any, Dict, List, Optional

import yaml
from pydantic import BaseModel, Field


class DatabaseConfig(BaseModel):
    # sqlite path; ":memory:" for tests. (The reference uses Postgres; this
    # framework embeds its backend store so a single-node MI355X deployment
    # has no external dependencies.) Lives in its own subdir so the worker
    # can tmpfs-mask it out of containers' view of the host rootfs.
    path: str = "/var/lib/b9/db/backend.db"


class GatewayConfig(BaseModel):
    host: str = "127.0.0.1"
    grpc_port: int = 1993
    http_port: int = 1994
    shutdown_timeout_s: float = 10.0
    invoke_url_type: str = "path"  # path | host
    # host-style invoke URLs resolve {name}[-v{N}|-latest].app.{base_domain}
    # (reference: pkg/gateway/middleware/subdomain.go, config invokeURLType)
    base_domain: str = "b9.local"
    stub_limits_cpu: int = 128000
    stub_limits_memory_mb: int = 32768
    stub_limits_max_replicas: int = 64
    stub_limits_max_gpu_count: int = 8


class ImageServiceConfig(BaseModel):
    # Content-addressed image store root (OCI layers + prefix images).
    store_path: str = "/var/lib/b9/images"
    chunk_size: int = 4 * 1024 * 1024
    build_container_ttl_s: float = 60.0
    # build containers (steps NEVER run on the gateway host)
    build_cpu: int = 4000
    build_memory_mb: int = 8192
    build_timeout_s: float = 900.0


class StorageConfig(BaseModel):
    # Workspace storage root (volumes, outputs, task results).
    root: str = "/var/lib/b9/workspaces"


class CacheConfig(BaseModel):
    # Distributed content cache (pages on disk, raw sendfile transport).
    enabled: bool = True
    root: str = "/var/lib/b9/cache"
    page_size: int = 4 * 1024 * 1024
    disk_quota_bytes: int = 200 * (1 << 30)
    eviction_watermark: float = 0.80
    transport_port: int = 1996
    inflight_bytes_cap: int = 512 * (1 << 20)


class PoolConfig(BaseModel):
    name: str = "default"
    gpu_type: str = ""  # "" = CPU pool; "MI355X" for GPU pools
    runtime: str = "proc"  # proc | native
    min_free_cpu: int = 0
    min_free_gpu: int = 0
    min_free_memory_mb: int = 0
    priority: int = 0
    preemptable: bool = False
    requires_pool_selector: bool = False
    criu_enabled: bool = False  # GPU snapshot/restore (hipsnap)


class WorkerConfig(BaseModel):
    # Defaults for spawned workers (the reference reads these from env set by
    # the pool controller; we keep the same env contract).
    cpu_limit: int = 64000  # millicores
    memory_limit_mb: int = 131072
    gpu_type: str = ""
    gpu_count: int = 0
    heartbeat_interval_s: float = 2.0
    keepalive_ttl_s: float = 15.0
    idle_shutdown_s: float = 0.0  # 0 = never
    container_start_limit: int = 32
    network_prealloc_slots: int = 32
    runtime: str = "proc"
    images_path: str = "/var/lib/b9/worker/images"
    scratch_path: str = "/var/lib/b9/worker/scratch"


class SchedulerConfig(BaseModel):
    # GPU failover chains: requested GPU -> ranked alternatives appended to
    # the preference list when the primary has no capacity (reference:
    # scheduling.failover config + pkg/scheduler/failover.go)
    gpu_failover: Dict[str, List[str]] = Field(default_factory=dict)
    backlog_poll_interval_s: float = 0.01
    batch_size: int = 10
    max_retries: int = 120
    max_schedule_age_s: float = 1200.0
    retry_backoff_base_s: float = 0.25
    retry_backoff_max_s: float = 10.0
    pools: List[PoolConfig] = Field(default_factory=lambda: [PoolConfig()])


class TaskConfig(BaseModel):
    heartbeat_ttl_s: float = 60.0
    monitor_interval_s: float = 1.0
    default_retries: int = 3
    default_timeout_s: float = 3600.0
    result_ttl_s: float = 600.0


class GPUConfig(BaseModel):
    # AMD device plane.
    kfd_topology_path: str = "/sys/class/kfd/kfd/topology/nodes"
    drm_path: str = "/sys/class/drm"
    visible_devices_env: List[str] = Field(
        default_factory=lambda: ["HIP_VISIBLE_DEVICES", "ROCR_VISIBLE_DEVICES"]
    )
    # RCCL rendezvous defaults injected for gpu_count>1 containers.
    rccl_env: Dict[str, str] = Field(
        default_factory=lambda: {
            "NCCL_SOCKET_IFNAME": "lo",
            "GLOO_SOCKET_IFNAME": "lo",
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        }
    )


class CheckpointConfig(BaseModel):
    # First-party GPU snapshot/restore (hipsnap): the runner dumps HBM-resident
    # state through the arena to the content cache; restore streams it back
    # through pinned staging before handler import completes.
    enabled: bool = True
    store_path: str = "/var/lib/b9/checkpoints"
    ready_sentinel: str = "READY_FOR_CHECKPOINT"
    staging_buffer_mb: int = 256


class StateConfig(BaseModel):
    # Control-plane state durability: "" = in-memory (dies with the
    # gateway); a path = WAL + snapshot persistence so backlog/claims
    # survive gateway restarts (reference keeps this in Redis).
    persist_path: str = "/var/lib/b9/state"


class MonitoringConfig(BaseModel):
    metrics_enabled: bool = True
    # lifecycle span sink: "state" (queryable via API) and/or "jsonl" file
    span_sink_path: str = ""


class AppConfig(BaseModel):
    database: DatabaseConfig = Field(default_factory=DatabaseConfig)
    gateway: GatewayConfig = Field(default_factory=GatewayConfig)
    image_service: ImageServiceConfig = Field(default_factory=ImageServiceConfig)
    storage: StorageConfig = Field(default_factory=StorageConfig)
    cache: CacheConfig = Field(default_factory=CacheConfig)
    worker: WorkerConfig = Field(default_factory=WorkerConfig)
    scheduler: SchedulerConfig = Field(default_factory=SchedulerConfig)
    task: TaskConfig = Field(default_factory=TaskConfig)
    gpu: GPUConfig = Field(default_factory=GPUConfig)
    checkpoint: CheckpointConfig = Field(default_factory=CheckpointConfig)
    state: StateConfig = Field(default_factory=StateConfig)
    monitoring: MonitoringConfig = Field(default_factory=MonitoringConfig)

    def with_root(self, root: str) -> "AppConfig":
        """Re-root all on-disk paths under `root` (tests, per-rank benches)."""
        cfg = self.model_copy(deep=True)
        cfg.database.path = os.path.join(root, "db/backend.db")
        cfg.image_service.store_path = os.path.join(root, "images")
        cfg.storage.root = os.path.join(root, "workspaces")
        cfg.cache.root = os.path.join(root, "cache")
        cfg.checkpoint.store_path = os.path.join(root, "checkpoints")
        cfg.worker.images_path = os.path.join(root, "worker/images")
        cfg.worker.scratch_path = os.path.join(root, "worker/scratch")
        cfg.state.persist_path = os.path.join(root, "state")
        return cfg


def _deep_merge(base: Dict[str, Any], overlay: Dict[str, Any]) -> Dict[str, Any]:
    out = dict(base)
    for k, v in overlay.items():
        if k in out and isinstance(out[k], dict) and isinstance(v, dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = v
    return out


def load_config(
    path: Optional[str] = None, inline: Optional[str] = None
) -> AppConfig:
    """Load AppConfig: defaults <- CONFIG_PATH yaml <- CONFIG_JSON json."""
    data: Dict[str, Any] = {}
    path = path if path is not None else os.environ.get("CONFIG_PATH")
    inline = inline if inline is not None else os.environ.get("CONFIG_JSON")
    if path and os.path.exists(path):
        with open(path) as f:
            loaded = yaml.safe_load(f) or {}
        data = _deep_merge(data, loaded)
    if inline:
        data = _deep_merge(data, json.loads(inline))
    return AppConfig.model_validate(data)
