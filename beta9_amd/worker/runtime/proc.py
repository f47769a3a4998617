"""ProcRuntime: direct process containers with cgroup-v2 limits.

The fastest cold-start path on a dedicated MI355X node: no image pivot, no
namespace setup — exec the runner directly with pinned GPU env, optional
cgroup v2 cpu/memory limits, its own session/pgid for clean teardown, and an
OOM watcher polling memory.events (reference analog: pkg/runtime/runc.go Run
+ pkg/runtime/oom_watcher.go:28-156).
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

CGROUP_ROOT = "/sys/fs/cgroup"


class _Proc:
    def __init__(self, proc: asyncio.subprocess.Process, cgroup_dir: str = ""):
        self.proc = proc
        self.cgroup_dir = cgroup_dir
        self.exit_code: Optional[int] = None
        self.oom_killed = False


class ProcRuntime(Runtime):
    name = "proc"

    def __init__(self, cgroup_parent: str = "b9"):
        self._containers: Dict[str, _Proc] = {}
        self._cgroup_parent = cgroup_parent
        self._cgroups_ok = self._probe_cgroups()

    def _probe_cgroups(self) -> bool:
        try:
            path = os.path.join(CGROUP_ROOT, self._cgroup_parent)
            os.makedirs(path, exist_ok=True)
            # need cpu+memory controllers delegated
            with open(os.path.join(CGROUP_ROOT, "cgroup.subtree_control")) as f:
                ctl = f.read()
            return "memory" in ctl or os.path.exists(os.path.join(path, "memory.max"))
        except OSError:
            return False

    def capabilities(self) -> set:
        caps = {"gpu"}
        if self._cgroups_ok:
            caps.add("cgroup_limits")
        return caps

    def _setup_cgroup(self, spec: ContainerSpec) -> str:
        if not self._cgroups_ok or (spec.cpu_millicores <= 0 and spec.memory_mb <= 0):
            return ""
        cg = os.path.join(CGROUP_ROOT, self._cgroup_parent, spec.container_id)
        try:
            os.makedirs(cg, exist_ok=True)
            if spec.memory_mb > 0:
                with open(os.path.join(cg, "memory.max"), "w") as f:
                    f.write(str(spec.memory_mb * 1024 * 1024))
            if spec.cpu_millicores > 0:
                period = 100000
                quota = spec.cpu_millicores * period // 1000
                with open(os.path.join(cg, "cpu.max"), "w") as f:
                    f.write(f"{quota} {period}")
            return cg
        except OSError:
            return ""

    async def run(self, spec: ContainerSpec, on_log: Optional[LogCallback] = None) -> ContainerHandle:
        from .base import container_base_env

        env = container_base_env(spec.env)
        cg = self._setup_cgroup(spec)
        proc = await asyncio.create_subprocess_exec(
            *spec.entry_point,
            env=env,
            cwd=spec.cwd or None,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
            start_new_session=True,
        )
        entry = _Proc(proc, cg)
        self._containers[spec.container_id] = entry
        if cg:
            try:
                with open(os.path.join(cg, "cgroup.procs"), "w") as f:
                    f.write(str(proc.pid))
            except OSError:
                pass
        if on_log is not None:
            asyncio.create_task(self._pump(proc.stdout, "stdout", on_log))
            asyncio.create_task(self._pump(proc.stderr, "stderr", on_log))
        else:
            asyncio.create_task(self._drain(proc.stdout))
            asyncio.create_task(self._drain(proc.stderr))
        return ContainerHandle(container_id=spec.container_id, pid=proc.pid)

    async def adopt(
        self,
        spec: ContainerSpec,
        proc: asyncio.subprocess.Process,
        on_log: Optional[LogCallback] = None,
    ) -> ContainerHandle:
        """Adopt an already-running process (a prewarmed runner shell) as a
        container: move it into the resource cgroup and take over its
        lifecycle."""
        cg = self._setup_cgroup(spec)
        entry = _Proc(proc, cg)
        self._containers[spec.container_id] = entry
        if cg:
            try:
                with open(os.path.join(cg, "cgroup.procs"), "w") as f:
                    f.write(str(proc.pid))
            except OSError:
                pass
        if on_log is not None and proc.stdout is not None:
            asyncio.create_task(self._pump(proc.stdout, "stdout", on_log))
            asyncio.create_task(self._pump(proc.stderr, "stderr", on_log))
        elif proc.stdout is not None:
            asyncio.create_task(self._drain(proc.stdout))
            asyncio.create_task(self._drain(proc.stderr))
        return ContainerHandle(container_id=spec.container_id, pid=proc.pid)

    @staticmethod
    async def _pump(stream: asyncio.StreamReader, name: str, cb: LogCallback) -> None:
        while True:
            try:
                line = await stream.readline()
            except (ValueError, asyncio.LimitOverrunError):
                continue
            if not line:
                return
            try:
                cb(name, line.decode(errors="replace").rstrip("\n"))
            except Exception:
                pass

    @staticmethod
    async def _drain(stream: asyncio.StreamReader) -> None:
        while True:
            chunk = await stream.read(65536)
            if not chunk:
                return

    def _check_oom(self, entry: _Proc) -> bool:
        if not entry.cgroup_dir:
            return False
        try:
            with open(os.path.join(entry.cgroup_dir, "memory.events")) as f:
                for line in f:
                    k, _, v = line.partition(" ")
                    if k == "oom_kill" and int(v) > 0:
                        return True
        except OSError:
            pass
        return False

    async def wait(self, container_id: str) -> int:
        entry = self._containers.get(container_id)
        if entry is None:
            return -1
        code = await entry.proc.wait()
        if self._check_oom(entry):
            entry.oom_killed = True
            code = 137
        entry.exit_code = code
        return code

    async def kill(self, container_id: str, sig: int = signal.SIGTERM) -> None:
        entry = self._containers.get(container_id)
        if entry is None or entry.proc.returncode is not None:
            return
        try:
            os.killpg(entry.proc.pid, sig)
        except ProcessLookupError:
            pass

    async def delete(self, container_id: str) -> None:
        entry = self._containers.pop(container_id, None)
        if entry is None:
            return
        if entry.proc.returncode is None:
            try:
                os.killpg(entry.proc.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
            try:
                await asyncio.wait_for(entry.proc.wait(), timeout=5)
            except asyncio.TimeoutError:
                pass
        if entry.cgroup_dir:
            try:
                os.rmdir(entry.cgroup_dir)
            except OSError:
                pass

    def state(self, container_id: str) -> ContainerRunState:
        entry = self._containers.get(container_id)
        if entry is None:
            return ContainerRunState.unknown
        if entry.proc.returncode is None:
            return ContainerRunState.running
        return ContainerRunState.exited
