"""NativeRuntime: containers via the first-party b9launch C launcher.

Real isolation (mount/PID/UTS/IPC namespaces, overlayfs root, minimal /dev
with only the assigned GPU nodes, cgroup-v2 limits, optional no-network) —
the runc-equivalent (reference: pkg/runtime/runc.go; built first-party
because the target image ships no runc). Source: native/launcher/b9launch.c,
built by `make -C native` into bin/b9launch.
"""

from __future__ import annotations

import asyncio
import os
import signal
from typing import Dict, Optional

from .base import (
    ContainerHandle,
    ContainerRunState,
    ContainerSpec,
    LogCallback,
    Runtime,
)
from .proc import ProcRuntime

REPO_ROOT = os.path.dirname(
    os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
)
LAUNCHER = os.path.join(REPO_ROOT, "bin", "b9launch")


def launcher_available() -> bool:
    return os.access(LAUNCHER, os.X_OK)


_NS_AVAILABLE: Optional[bool] = None


def namespaces_available() -> bool:
    """Probe whether this host permits the namespace/overlay setup
    (cached: one b9launch round-trip per process)."""
    global _NS_AVAILABLE
    if _NS_AVAILABLE is not None:
        return _NS_AVAILABLE
    if not launcher_available():
        _NS_AVAILABLE = False
        return False
    import subprocess

    try:
        r = subprocess.run(
            [LAUNCHER, "--rootfs-overlay", "/tmp/b9-probe-ov", "--", "true"],
            capture_output=True,
            timeout=15,
        )
        _NS_AVAILABLE = r.returncode == 0
    except (subprocess.TimeoutExpired, OSError):
        _NS_AVAILABLE = False
    return _NS_AVAILABLE


class NativeRuntime(ProcRuntime):
    """b9launch-backed containers; shares process bookkeeping with
    ProcRuntime (wait/kill/state), overrides the launch."""

    name = "native"

    def __init__(self, scratch_path: str = "/var/lib/b9/worker/scratch"):
        super().__init__()
        self.scratch_path = scratch_path
        os.makedirs(scratch_path, exist_ok=True)

    def capabilities(self) -> set:
        return {"gpu", "cgroup_limits", "rootfs_isolation", "device_isolation", "network_block"}

    async def run(self, spec: ContainerSpec, on_log: Optional[LogCallback] = None) -> ContainerHandle:
        overlay_dir = os.path.join(self.scratch_path, spec.container_id)
        argv = [
            LAUNCHER,
            "--rootfs-overlay",
            overlay_dir,
            "--hostname",
            spec.hostname or spec.container_id,
            "--cgroup-parent",
            "b9",
        ]
        if spec.image_lowerdirs:
            # rootfs image: pure layer chain (host "/" never mixed in —
            # overlayfs rejects nested lowerdirs)
            argv += ["--lower", ":".join(spec.image_lowerdirs)]
        if spec.upper_dir and spec.work_dir:
            argv += ["--upper", spec.upper_dir, "--work", spec.work_dir]
        if not spec.image_lowerdirs:
            # host-rootfs mode: mask the platform's data roots
            for mask in spec.mask_paths:
                argv += ["--mask", mask]
        if spec.cpu_millicores > 0:
            argv += ["--cpu", str(spec.cpu_millicores)]
        if spec.memory_mb > 0:
            argv += ["--mem", str(spec.memory_mb)]
        for dev in spec.devices:
            argv += ["--dev", dev]
        for m in spec.mounts:
            local = m.get("local_path") if isinstance(m, dict) else m.local_path
            target = m.get("mount_path") if isinstance(m, dict) else m.mount_path
            if local and target:
                argv += ["--bind", f"{local}:{target}"]
        if spec.block_network:
            argv += ["--block-network"]
        pass_fds = ()
        if spec.netns_fd >= 0 and not spec.block_network:
            argv += ["--netns-fd", str(spec.netns_fd)]
            pass_fds = (spec.netns_fd,)
        if spec.cwd:
            argv += ["--cwd", spec.cwd]
        argv += ["--"] + list(spec.entry_point)

        from .base import container_base_env

        env = container_base_env(spec.env)
        if not spec.image_lowerdirs:
            # host-rootfs mode: the runner package must be importable even
            # though the container's cwd is / after pivot (the worker may
            # rely on cwd for imports; containers cannot)
            extra = env.get("PYTHONPATH", "")
            env["PYTHONPATH"] = (extra + os.pathsep if extra else "") + REPO_ROOT
        proc = await asyncio.create_subprocess_exec(
            *argv,
            env=env,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
            start_new_session=True,
            pass_fds=pass_fds,
        )
        from .proc import _Proc

        entry = _Proc(proc, "")
        self._containers[spec.container_id] = entry
        if on_log is not None:
            asyncio.create_task(self._pump(proc.stdout, "stdout", on_log))
            asyncio.create_task(self._pump(proc.stderr, "stderr", on_log))
        else:
            asyncio.create_task(self._drain(proc.stdout))
            asyncio.create_task(self._drain(proc.stderr))
        return ContainerHandle(container_id=spec.container_id, pid=proc.pid)

    async def delete(self, container_id: str) -> None:
        await super().delete(container_id)
        overlay_dir = os.path.join(self.scratch_path, container_id)
        # the tmpfs inside unmounts with the namespace; remove the dir
        try:
            os.rmdir(overlay_dir)
        except OSError:
            pass
