"""Sandbox process manager — the in-container exec/fs server.

First-party replacement for the reference's goproc binary (reference:
beam-cloud/goproc, proxied via pkg/worker/sandbox.go:468-573 and
container_server.go:671-1403): a msgpack-gRPC ProcessManager serving
exec/run_code/process mgmt/file ops on the container's BIND_PORT. The
gateway's PodService proxies Sandbox* RPCs here directly.
"""

from __future__ import annotations

import asyncio
import base64
import fnmatch
import logging
import os
import re
import signal
import sys
import time
from typing import Any, Dict, List, Optional

from ..rpc.channel import RpcServer, invalid, not_found, rpc
from .common import GPUBoot, RunnerEnv, setup_environment

log = logging.getLogger("b9.runner.sandbox")


class _Process:
    def __init__(self, pid: int, proc: asyncio.subprocess.Process, cmd: str):
        self.pid = pid
        self.proc = proc
        self.cmd = cmd
        self.stdout = b""
        self.stderr = b""
        self.exit_code: Optional[int] = None
        self.started_at = time.time()


class _Terminal:
    """A PTY-backed shell session (the dropbear/SSH terminal-semantics
    equivalent — reference: pkg/abstractions/shell + in-container dropbear;
    here the PM owns the pty directly, no SSH daemon needed)."""

    def __init__(self, term_id: int, pid: int, master_fd: int):
        self.term_id = term_id
        self.pid = pid
        self.master_fd = master_fd
        self.buf = b""
        self.closed = False

    def pump(self) -> None:
        """Drain available pty output into the buffer (non-blocking)."""
        import errno

        while True:
            try:
                data = os.read(self.master_fd, 65536)
            except BlockingIOError:
                return
            except OSError as e:
                if e.errno in (errno.EIO, errno.EBADF):  # shell exited
                    self.closed = True
                return
            if not data:
                self.closed = True
                return
            self.buf += data
            if len(self.buf) > (4 << 20):
                self.buf = self.buf[-(4 << 20):]


class ProcessManager:
    service_name = "ProcessManager"

    def __init__(self, root: str = "/"):
        self.root = root
        self.procs: Dict[int, _Process] = {}
        self.terms: Dict[int, _Terminal] = {}
        self._next_term = 0

    def _path(self, p: str) -> str:
        if not os.path.isabs(p):
            p = os.path.join(os.getcwd(), p)
        return p

    # -- exec ----------------------------------------------------------------
    async def _spawn(self, argv: List[str], cwd: str = "", env: Optional[Dict] = None) -> _Process:
        full_env = dict(os.environ)
        if env:
            full_env.update({str(k): str(v) for k, v in env.items()})
        proc = await asyncio.create_subprocess_exec(
            *argv,
            cwd=cwd or None,
            env=full_env,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
            start_new_session=True,
        )
        entry = _Process(proc.pid, proc, " ".join(argv))
        self.procs[proc.pid] = entry

        async def collect() -> None:
            out, err = await proc.communicate()
            entry.stdout, entry.stderr = out, err
            entry.exit_code = proc.returncode

        asyncio.create_task(collect())
        return entry

    @rpc()
    async def Exec(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        cmd = req.get("command")
        if isinstance(cmd, str):
            argv = ["bash", "-c", cmd]
        elif isinstance(cmd, list):
            argv = cmd
        else:
            raise invalid("command must be str or list")
        entry = await self._spawn(argv, cwd=req.get("cwd", ""), env=req.get("env"))
        if not req.get("background"):
            timeout = float(req.get("timeout", 120.0))
            try:
                await asyncio.wait_for(entry.proc.wait(), timeout=timeout)
                # give the collector a tick to store output
                for _ in range(100):
                    if entry.exit_code is not None:
                        break
                    await asyncio.sleep(0.005)
            except asyncio.TimeoutError:
                return {"pid": entry.pid, "running": True, "timeout": True}
        return {
            "pid": entry.pid,
            "running": entry.exit_code is None,
            "exit_code": entry.exit_code,
            "stdout": entry.stdout.decode(errors="replace"),
            "stderr": entry.stderr.decode(errors="replace"),
        }

    # -- nested containers (the Docker-in-sandbox analog) -------------------
    @rpc()
    async def RunContainer(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Run an ISOLATED nested container inside this sandbox via
        b9launch (reference: Docker-in-sandbox, pkg/worker/sandbox.go
        :148-465 — dockerd in the sandbox; here the first-party launcher
        nests directly: namespaces-in-namespaces, own overlay rootfs,
        optional image layers, optional no-network)."""
        import shlex

        launcher = os.environ.get("B9_LAUNCHER", "")
        if not launcher:
            for cand in ("/opt/b9/bin/b9launch",):
                if os.access(cand, os.X_OK):
                    launcher = cand
                    break
        if not launcher:
            # host-rootfs sandboxes see the repo tree
            import beta9_amd

            repo = os.path.dirname(os.path.dirname(os.path.abspath(beta9_amd.__file__)))
            launcher = os.path.join(repo, "bin", "b9launch")
        if not os.access(launcher, os.X_OK):
            return {"error": "b9launch unavailable in this sandbox"}
        cmd = req.get("command")
        argv_user = ["sh", "-c", cmd] if isinstance(cmd, str) else list(cmd or ["sh"])
        import uuid

        overlay = f"/tmp/b9-nested-{uuid.uuid4().hex[:8]}"
        argv = [launcher, "--rootfs-overlay", overlay,
                "--hostname", req.get("hostname", "nested")]
        if req.get("block_network"):
            argv.append("--block-network")
        if int(req.get("cpu", 0)) > 0:
            argv += ["--cpu", str(int(req["cpu"]))]
        if int(req.get("memory_mb", 0)) > 0:
            argv += ["--mem", str(int(req["memory_mb"]))]
        image_id = req.get("image_id", "")
        if image_id:
            store_path = os.environ.get("B9_IMAGE_STORE", "")
            if store_path:
                from ..images.store import ImageStore

                try:
                    dirs = ImageStore(store_path).lowerdirs(image_id)
                    argv += ["--lower", ":".join(dirs)]
                except Exception as e:  # noqa: BLE001
                    return {"error": f"image {image_id}: {e}"}
        argv += ["--"] + argv_user
        proc = await asyncio.create_subprocess_exec(
            *argv,
            stdout=asyncio.subprocess.PIPE,
            stderr=asyncio.subprocess.PIPE,
        )
        try:
            out, err = await asyncio.wait_for(
                proc.communicate(), timeout=float(req.get("timeout", 120.0))
            )
        except asyncio.TimeoutError:
            proc.kill()
            return {"error": "nested container timed out", "timeout": True}
        return {
            "exit_code": proc.returncode,
            "stdout": out.decode(errors="replace"),
            "stderr": err.decode(errors="replace"),
            "cmd": " ".join(shlex.quote(a) for a in argv_user),
        }

    # -- interactive terminal (PTY) -----------------------------------------
    @rpc()
    async def TerminalOpen(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        import fcntl
        import pty
        import shutil
        import struct
        import termios

        shell = req.get("command") or shutil.which("bash") or "/bin/sh"
        rows = int(req.get("rows", 24))
        cols = int(req.get("cols", 80))
        pid, master_fd = pty.fork()
        if pid == 0:  # the shell
            env = dict(os.environ)
            env["TERM"] = req.get("term", "xterm-256color")
            try:
                os.chdir(req.get("cwd") or os.getcwd())
            except OSError:
                pass
            os.execvpe(shell, [shell], env)
            os._exit(127)
        os.set_blocking(master_fd, False)
        fcntl.ioctl(master_fd, termios.TIOCSWINSZ, struct.pack("HHHH", rows, cols, 0, 0))
        self._next_term += 1
        t = _Terminal(self._next_term, pid, master_fd)
        self.terms[t.term_id] = t
        return {"term_id": t.term_id, "pid": pid}

    def _term(self, req) -> _Terminal:
        t = self.terms.get(int(req.get("term_id", 0)))
        if t is None:
            raise invalid(f"no terminal {req.get('term_id')}")
        return t

    @rpc()
    async def TerminalRead(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Long-poll read: returns buffered pty output (waits up to
        `timeout` for the first byte)."""
        t = self._term(req)
        deadline = time.time() + float(req.get("timeout", 10.0))
        while True:
            t.pump()
            if t.buf or t.closed or time.time() >= deadline:
                data, t.buf = t.buf, b""
                return {"data": data, "closed": t.closed}
            await asyncio.sleep(0.02)

    @rpc()
    async def TerminalWrite(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        t = self._term(req)
        data = req.get("data", b"")
        if isinstance(data, str):
            data = data.encode()
        try:
            os.write(t.master_fd, data)
        except OSError:
            t.closed = True
        return {"ok": not t.closed}

    @rpc()
    async def TerminalResize(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        import fcntl
        import struct
        import termios

        t = self._term(req)
        fcntl.ioctl(
            t.master_fd, termios.TIOCSWINSZ,
            struct.pack("HHHH", int(req.get("rows", 24)), int(req.get("cols", 80)), 0, 0),
        )
        return {"ok": True}

    @rpc()
    async def TerminalClose(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        t = self.terms.pop(int(req.get("term_id", 0)), None)
        if t is not None:
            import signal as _signal

            try:
                os.kill(t.pid, _signal.SIGHUP)
            except ProcessLookupError:
                pass
            try:
                os.close(t.master_fd)
            except OSError:
                pass
        return {"ok": True}

    @rpc()
    async def RunCode(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._run_code(req)

    async def _run_code(self, req: Dict[str, Any]) -> Dict[str, Any]:
        entry = await self._spawn([sys.executable, "-c", req.get("code", "")], cwd=req.get("cwd", ""), env=req.get("env"))
        try:
            await asyncio.wait_for(entry.proc.wait(), timeout=float(req.get("timeout", 120.0)))
            for _ in range(100):
                if entry.exit_code is not None:
                    break
                await asyncio.sleep(0.005)
        except asyncio.TimeoutError:
            return {"pid": entry.pid, "running": True, "timeout": True}
        return {
            "pid": entry.pid,
            "running": False,
            "exit_code": entry.exit_code,
            "stdout": entry.stdout.decode(errors="replace"),
            "stderr": entry.stderr.decode(errors="replace"),
        }

    @rpc()
    async def Status(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        pid = req.get("pid")
        if pid is not None:
            entry = self.procs.get(int(pid))
            if entry is None:
                raise not_found(f"pid {pid}")
            return {
                "pid": entry.pid,
                "running": entry.exit_code is None,
                "exit_code": entry.exit_code,
                "stdout": entry.stdout.decode(errors="replace"),
                "stderr": entry.stderr.decode(errors="replace"),
            }
        return {"ok": True, "pid": os.getpid(), "uptime": time.time()}

    @rpc()
    async def ListProcesses(self, req: Dict[str, Any], context, auth) -> List[Dict]:
        return [
            {"pid": p.pid, "cmd": p.cmd, "running": p.exit_code is None, "exit_code": p.exit_code}
            for p in self.procs.values()
        ]

    @rpc()
    async def KillProcess(self, req: Dict[str, Any], context, auth) -> bool:
        entry = self.procs.get(int(req["pid"]))
        if entry is None or entry.exit_code is not None:
            return False
        try:
            os.killpg(entry.pid, int(req.get("signal", signal.SIGKILL)))
        except ProcessLookupError:
            pass
        return True

    # -- files ----------------------------------------------------------------
    @rpc()
    async def UploadFile(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        path = self._path(req["path"])
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        data = req.get("data", b"")
        if isinstance(data, str):
            data = base64.b64decode(data)
        mode = "ab" if req.get("append") else "wb"
        with open(path, mode) as f:
            f.write(data)
        return {"path": path, "size": len(data)}

    @rpc()
    async def DownloadFile(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        path = self._path(req["path"])
        if not os.path.isfile(path):
            raise not_found(path)
        offset = int(req.get("offset", 0))
        length = int(req.get("length", 0)) or None
        with open(path, "rb") as f:
            f.seek(offset)
            data = f.read(length) if length else f.read()
        return {"data": data, "size": len(data)}

    @rpc()
    async def ListFiles(self, req: Dict[str, Any], context, auth) -> List[Dict]:
        path = self._path(req.get("path", "."))
        if not os.path.isdir(path):
            raise not_found(path)
        out = []
        for name in sorted(os.listdir(path)):
            fp = os.path.join(path, name)
            try:
                st = os.stat(fp)
                out.append(
                    {
                        "name": name,
                        "is_dir": os.path.isdir(fp),
                        "size": st.st_size,
                        "mtime": st.st_mtime,
                    }
                )
            except OSError:
                pass
        return out

    @rpc()
    async def StatFile(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        path = self._path(req["path"])
        if not os.path.exists(path):
            return {"exists": False}
        st = os.stat(path)
        return {"exists": True, "is_dir": os.path.isdir(path), "size": st.st_size, "mtime": st.st_mtime}

    @rpc()
    async def DeleteFile(self, req: Dict[str, Any], context, auth) -> bool:
        path = self._path(req["path"])
        if os.path.isdir(path):
            import shutil

            shutil.rmtree(path, ignore_errors=True)
        elif os.path.exists(path):
            os.unlink(path)
        return True

    @rpc()
    async def ReplaceInFiles(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        root = self._path(req["path"])
        old, new = req["old"], req["new"]
        pattern = req.get("pattern", "*")
        changed = 0
        for dirpath, _dirs, files in os.walk(root):
            for fname in files:
                if not fnmatch.fnmatch(fname, pattern):
                    continue
                fp = os.path.join(dirpath, fname)
                try:
                    with open(fp, "r", errors="ignore") as f:
                        content = f.read()
                except OSError:
                    continue
                if old in content:
                    with open(fp, "w") as f:
                        f.write(content.replace(old, new))
                    changed += 1
        return {"files_changed": changed}

    @rpc()
    async def SnapshotFilesystem(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Archive a subtree of the sandbox filesystem (tar.gz bytes) so the
        gateway can register it as a new image (reference:
        sandbox.py:527-549 create_image_from_filesystem -> worker archives
        the container fs into a CLIP image)."""
        import io
        import tarfile

        # default: the sandbox workspace (cwd). "/" only makes sense under
        # the native runtime where it is the container overlay root.
        sub = os.path.normpath(self._path(req.get("path") or "."))
        max_bytes = int(req.get("max_bytes", 512 * 1024 * 1024))
        skip = {"/proc", "/sys", "/dev", "/run", "/.oldroot"}
        buf = io.BytesIO()
        total = 0
        with tarfile.open(fileobj=buf, mode="w:gz") as tf:
            for dirpath, dirnames, filenames in os.walk(sub):
                # never descend into kernel/virtual trees
                dirnames[:] = [
                    d for d in dirnames
                    if os.path.normpath(os.path.join(dirpath, d)) not in skip
                ]
                for fn in filenames:
                    full = os.path.join(dirpath, fn)
                    try:
                        st = os.lstat(full)
                    except OSError:
                        continue
                    if not (os.path.isfile(full) and not os.path.islink(full)):
                        continue
                    total += st.st_size
                    if total > max_bytes:
                        return {"error": f"subtree exceeds max_bytes={max_bytes}", "size": total}
                    try:
                        tf.add(full, arcname=os.path.relpath(full, sub))
                    except (PermissionError, OSError):
                        continue
        data = buf.getvalue()
        return {"archive": data, "size": len(data), "files_root": sub}

    @rpc()
    async def FindInFiles(self, req: Dict[str, Any], context, auth) -> List[Dict]:
        root = self._path(req["path"])
        regex = re.compile(req["pattern"])
        out: List[Dict] = []
        for dirpath, _dirs, files in os.walk(root):
            for fname in files:
                fp = os.path.join(dirpath, fname)
                try:
                    with open(fp, "r", errors="ignore") as f:
                        for i, line in enumerate(f, 1):
                            if regex.search(line):
                                out.append({"path": fp, "line": i, "text": line.rstrip()})
                                if len(out) >= int(req.get("limit", 1000)):
                                    return out
                except OSError:
                    continue
        return out


async def main() -> int:
    env = RunnerEnv.from_env()
    setup_environment(env)
    gpu = GPUBoot(env)
    gpu.boot()
    pm = ProcessManager()
    server = RpcServer(host="0.0.0.0", port=env.bind_port)
    server.register_service(pm)
    await server.start()
    log.info("sandbox process manager up on :%d", env.bind_port)
    await asyncio.Event().wait()  # run until killed
    return 0


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    try:
        sys.exit(asyncio.run(main()))
    except KeyboardInterrupt:
        sys.exit(0)
