"""Container runtime abstraction.

Reference interface (reference: pkg/runtime/runtime.go:87-127 — Name /
Capabilities / Run / Exec / Kill / Delete / State / Checkpoint / Restore,
with runc / gVisor / docker impls). This framework ships two first-party
runtimes:

- ProcRuntime (proc.py): direct process launch with cgroup-v2 resource
  limits when available — the dev/test runtime and the lowest-latency GPU
  path (no rootfs pivot; device isolation via *_VISIBLE_DEVICES pinning).
- NativeRuntime (native.py): the b9launch C launcher — Linux namespaces
  (PID/UTS/IPC/NS), overlayfs rootfs, device-node injection (/dev/kfd +
  /dev/dri/renderD*), cgroup v2 cpu/memory/device controllers. This is the
  runc-equivalent, built from source in native/launcher/.

GPU checkpoint/restore is NOT a runtime capability here (no CRIU in the
image): it is a workload-level hipsnap snapshot (beta9_amd/worker/snapshot.py)
— see SURVEY.md §5.4 for what the reference does instead.
"""

from __future__ import annotations

import abc
from dataclasses import dataclass, field
from enum import Enum
from typing import Awaitable, Callable, Dict, List, Optional


class ContainerRunState(str, Enum):
    created = "created"
    running = "running"
    exited = "exited"
    unknown = "unknown"


@dataclass
class ContainerSpec:
    container_id: str
    entry_point: List[str]
    env: Dict[str, str] = field(default_factory=dict)
    cwd: str = ""
    rootfs: str = ""  # "" = share host rootfs (proc runtime)
    cpu_millicores: int = 0  # 0 = unlimited
    memory_mb: int = 0
    devices: List[str] = field(default_factory=list)  # /dev nodes to allow
    mounts: List[Dict] = field(default_factory=list)
    hostname: str = ""
    block_network: bool = False
    # OCI-layer rootfs: overlayfs lowerdir chain (topmost first) from the
    # content-addressed image store; empty = host-rootfs overlay mode.
    image_lowerdirs: List[str] = field(default_factory=list)
    # host-backed upper/work (build capture: the fs delta survives exit)
    upper_dir: str = ""
    work_dir: str = ""
    # host paths hidden behind empty tmpfs inside the container (gateway
    # DB / workspace state must not leak through the host-rootfs lowerdir)
    mask_paths: List[str] = field(default_factory=list)
    # preallocated network-namespace slot to join (fd into nsfs); -1 = host
    # network (or a fresh empty netns when block_network is set)
    netns_fd: int = -1


@dataclass
class ContainerHandle:
    container_id: str
    pid: int
    state: ContainerRunState = ContainerRunState.running


LogCallback = Callable[[str, str], None]  # (stream, line)


class Runtime(abc.ABC):
    name: str = "base"

    @abc.abstractmethod
    def capabilities(self) -> set:
        ...

    @abc.abstractmethod
    async def run(self, spec: ContainerSpec, on_log: Optional[LogCallback] = None) -> ContainerHandle:
        ...

    @abc.abstractmethod
    async def wait(self, container_id: str) -> int:
        """Block until exit; return exit code."""
        ...

    @abc.abstractmethod
    async def kill(self, container_id: str, sig: int = 15) -> None:
        ...

    @abc.abstractmethod
    async def delete(self, container_id: str) -> None:
        ...

    @abc.abstractmethod
    def state(self, container_id: str) -> ContainerRunState:
        ...


# Exit code the OOM watcher / killers normalize to (reference:
# pkg/worker/lifecycle.go:1927-1973 exit-code normalization).
EXIT_OOM = 137

# Distributed-launcher env that must NEVER leak from the worker process into
# containers (a torchrun-launched worker would otherwise make every endpoint
# runner think it is rank N of a world): the platform sets these itself for
# gpu_count>1 containers (worker/gpu.py provision_rendezvous).
_DIST_ENV = (
    "RANK",
    "LOCAL_RANK",
    "WORLD_SIZE",
    "LOCAL_WORLD_SIZE",
    "GROUP_RANK",
    "GROUP_WORLD_SIZE",
    "ROLE_RANK",
    "ROLE_WORLD_SIZE",
    "ROLE_NAME",
    "MASTER_ADDR",
    "MASTER_PORT",
    "OMP_NUM_THREADS",
)


def container_base_env(spec_env: Dict[str, str]) -> Dict[str, str]:
    """os.environ minus launcher-distributed vars, overlaid with the spec."""
    import os as _os

    env = dict(_os.environ)
    for k in list(env):
        if k in _DIST_ENV or k.startswith("TORCHELASTIC_") or k.startswith("PET_"):
            del env[k]
    env.update(spec_env)
    return env
