"""Worker-side prewarm pool: keeps runner shells (torch+HIP pre-initialized)
per GPU ordinal and hands them to the lifecycle on container start.

See beta9_amd/runner/prewarm.py for the shell side. Pool shape: one shell
per GPU ordinal (the GPU is pinned at spawn — HIP visibility cannot be
changed after context creation) plus a small CPU pool. A shell consumed by a
container start is replenished in the background.
"""

from __future__ import annotations

import asyncio
import json
import logging
import os
import sys
import tempfile
import time
from typing import Dict, List, Optional, Tuple

log = logging.getLogger(__name__)

_RUNNER_MODULES = {
    "beta9_amd.runner.endpoint",
    "beta9_amd.runner.function",
    "beta9_amd.runner.taskqueue",
    "beta9_amd.runner.sandbox",
    "beta9_amd.runner.pod",
}


class PrewarmedShell:
    def __init__(self, proc: asyncio.subprocess.Process, sock_path: str, key: Tuple[int, ...]):
        self.proc = proc
        self.sock_path = sock_path
        self.key = key
        self.ready = False

    async def wait_ready(self, timeout: float = 120.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if self.proc.returncode is not None:
                return False
            if os.path.exists(self.sock_path):
                # socket exists once the shell is parked
                self.ready = True
                return True
            await asyncio.sleep(0.02)
        return False

    async def activate(
        self, env: Dict[str, str], module: str, netns_fd: Optional[int] = None
    ) -> bool:
        """Send the spec; on success the shell becomes the container. A
        netns fd travels over the unix socket (SCM_RIGHTS) and the shell
        setns()es into the slot before exec'ing the runner — prewarm hits
        get per-container networking too."""
        import socket as _socket

        def _sync() -> bool:
            s = _socket.socket(_socket.AF_UNIX, _socket.SOCK_STREAM)
            s.settimeout(10.0)
            try:
                s.connect(self.sock_path)
                spec = {"env": env, "module": module, "want_netns": netns_fd is not None}
                s.sendall((json.dumps(spec) + "\n").encode())
                if netns_fd is not None:
                    _socket.send_fds(s, [b"F"], [netns_fd])
                resp = b""
                while not resp.endswith(b"\n"):
                    chunk = s.recv(64)
                    if not chunk:
                        return False
                    resp += chunk
                return resp.strip() == b"ok"
            except (OSError, _socket.timeout):
                return False
            finally:
                s.close()

        try:
            return await asyncio.wait_for(asyncio.to_thread(_sync), timeout=12.0)
        except asyncio.TimeoutError:
            return False


class PrewarmPool:
    def __init__(
        self,
        gpu_ordinals: List[int],
        cpu_shells: int = 2,
        base_env: Optional[Dict[str, str]] = None,
        visible_env_names: Optional[List[str]] = None,
        shells_per_gpu: int = 2,
        isolate: bool = False,
        device_paths: Optional[Dict[int, List[str]]] = None,
        mask_paths: Optional[List[str]] = None,
        bind_paths: Optional[List[str]] = None,
        zygotes: int = 2,
    ):
        """isolate=True parks each shell INSIDE a b9launch container
        (namespaces + overlay root + only its own GPU nodes in /dev), so a
        prewarm hit is an isolated container too — the cold-start headline
        never bypasses the isolation default. The shell's unix socket dir is
        bind-mounted through; activation works unchanged."""
        self.gpu_ordinals = gpu_ordinals
        self.cpu_shells = cpu_shells
        self.shells_per_gpu = max(0, shells_per_gpu)
        self.base_env = base_env or {}
        self.visible_env_names = visible_env_names or [
            "HIP_VISIBLE_DEVICES",
            "ROCR_VISIBLE_DEVICES",
        ]
        self.isolate = isolate
        self.device_paths = device_paths or {}
        self.mask_paths = mask_paths or []
        self.bind_paths = bind_paths or []
        # zygote tier (the MISS path): shells with every import warmed but
        # NO HIP context — GPU-agnostic, activated onto ANY ordinal (HIP
        # reads *_VISIBLE_DEVICES at the first cuda.init after activation)
        self.zygotes = zygotes if gpu_ordinals else 0
        self._shells: Dict[Tuple[int, ...], List[PrewarmedShell]] = {}
        self._dir = tempfile.mkdtemp(prefix="b9-prewarm-")
        self._scratch = tempfile.mkdtemp(prefix="b9-prewarm-ov-")
        self._replenish_lock = asyncio.Lock()
        self._n = 0
        self._stopping = False

    ZYGOTE_KEY = ("zygote",)

    async def start(self) -> None:
        tasks = []
        for o in self.gpu_ordinals:
            tasks += [self._spawn((o,)) for _ in range(self.shells_per_gpu)]
        tasks += [self._spawn(()) for _ in range(self.cpu_shells)]
        tasks += [self._spawn(self.ZYGOTE_KEY) for _ in range(self.zygotes)]
        await asyncio.gather(*tasks)

    async def _spawn(self, key: Tuple[int, ...]) -> None:
        if self._stopping:
            return
        self._n += 1
        sock_path = os.path.join(self._dir, f"shell-{self._n}.sock")
        from .runtime.base import container_base_env

        env = container_base_env(self.base_env)
        env["B9_PREWARM_SOCK"] = sock_path
        is_zygote = key == self.ZYGOTE_KEY
        if is_zygote:
            env["B9_PREWARM_ZYGOTE"] = "1"
            env.pop("WORKER_GPU_DEVICES", None)
        elif key:
            vis = ",".join(str(o) for o in key)
            for name in self.visible_env_names:
                env[name] = vis
            env["WORKER_GPU_DEVICES"] = vis
        else:
            env.pop("WORKER_GPU_DEVICES", None)
        argv = [sys.executable, "-u", "-m", "beta9_amd.runner.prewarm"]
        if self.isolate:
            from .runtime.native import LAUNCHER, REPO_ROOT

            # cwd pivots to "/" inside the namespace; the package must come
            # from PYTHONPATH, not the worker's cwd
            extra = env.get("PYTHONPATH", "")
            env["PYTHONPATH"] = (extra + os.pathsep if extra else "") + REPO_ROOT

            wrap = [
                LAUNCHER,
                "--rootfs-overlay",
                os.path.join(self._scratch, f"shell-{self._n}"),
                "--hostname",
                f"prewarm-{self._n}",
                "--bind",
                f"{self._dir}:{self._dir}",  # the park socket crosses the ns
            ]
            for mask in self.mask_paths:
                wrap += ["--mask", mask]
            for b in self.bind_paths:
                wrap += ["--bind", f"{b}:{b}"]
            if is_zygote:
                # GPU-agnostic: all device nodes in /dev (env pins the
                # actual GPU at activation)
                seen = set()
                for devs in self.device_paths.values():
                    for dev in devs:
                        if dev not in seen:
                            seen.add(dev)
                            wrap += ["--dev", dev]
            else:
                for o in key:
                    for dev in self.device_paths.get(o, []):
                        wrap += ["--dev", dev]
            argv = wrap + ["--"] + argv
        proc = await asyncio.create_subprocess_exec(
            *argv,
            env=env,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
            start_new_session=True,
        )
        shell = PrewarmedShell(proc, sock_path, key)
        self._shells.setdefault(key, []).append(shell)

    def sees_path(self, path: str) -> bool:
        """Whether a parked shell's mount namespace can see `path` — shells
        bind a fixed set of host dirs at spawn; activation cannot add
        binds, so containers needing other paths skip the prewarm pool."""
        if not path or not self.isolate:
            return True
        return any(
            os.path.commonpath([path, b]) == b for b in self.bind_paths if b
        )

    def _take_ready(self, key) -> Optional[PrewarmedShell]:
        shells = self._shells.get(key, [])
        for shell in list(shells):
            if shell.proc.returncode is not None:
                shells.remove(shell)
                continue
            if not shell.ready and not os.path.exists(shell.sock_path):
                continue
            shells.remove(shell)
            asyncio.create_task(self._spawn(key))  # replenish in background
            return shell
        return None

    async def acquire(self, ordinals: List[int]) -> Optional[PrewarmedShell]:
        """Take a ready shell matching the GPU set (or CPU for []).
        Single-GPU misses fall back to a ZYGOTE (imports warm, HIP context
        created after activation on the pinned ordinal) — the miss pays
        HIP init + model load, not python+torch import."""
        key = tuple(ordinals)
        shell = self._take_ready(key)
        if shell is not None:
            return shell
        if len(ordinals) == 1 and self.zygotes:
            return self._take_ready(self.ZYGOTE_KEY)
        return None

    async def shutdown(self) -> None:
        self._stopping = True
        for shells in self._shells.values():
            for shell in shells:
                if shell.proc.returncode is None:
                    try:
                        shell.proc.terminate()
                    except ProcessLookupError:
                        pass
        for shells in self._shells.values():
            for shell in shells:
                try:
                    await asyncio.wait_for(shell.proc.wait(), timeout=3.0)
                except asyncio.TimeoutError:
                    try:
                        shell.proc.kill()
                    except ProcessLookupError:
                        pass


def runner_module_from_entry(entry_point: List[str]) -> Optional[str]:
    """Extract the runner module from a standard entrypoint, or None if the
    entrypoint is custom (no prewarm for those)."""
    if len(entry_point) >= 3 and entry_point[-2] == "-m" and entry_point[-1] in _RUNNER_MODULES:
        return entry_point[-1]
    return None
