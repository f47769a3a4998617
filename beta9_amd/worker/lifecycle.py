"""Container lifecycle: the cold-start hot path.

RunContainer mirrors the reference's fully phase-instrumented path
(reference: pkg/worker/lifecycle.go:581 RunContainer, :1532 spawn, :1975
runContainer): port reservation -> spec from request -> GPU assignment +
device/env injection (+ RCCL rendezvous for gpu_count>1) -> checkpoint
attach -> runtime exec -> address publication -> exit normalization. Every
phase emits a lifecycle span (common/spans.py) — these are the numbers
bench.py reports as the cold-start decomposition.

MI355X specifics vs the reference: GPU assignment is xGMI-topology-aware
(gpu.py), the runner pre-initializes HIP + pre-allocates its HBM arena before
handler import (runner/common.py + ops hipsnap), and scale-to-zero restore
streams an HBM snapshot through pinned staging instead of CRIU
(ops/hipsnap.py + the grant/commit flow in gateway/services/core.py).
"""

from __future__ import annotations

import asyncio
import logging
import os
import socket
import sys
import time
from typing import Dict, Optional

from ..common.spans import SpanRecorder
from ..types import ContainerRequest, ContainerState, ContainerStatus, StubType
from .gpu import AllocationError, GPUManager
from .runtime.base import ContainerSpec, Runtime

log = logging.getLogger(__name__)

# Stub kinds whose containers serve HTTP and need an address published.
_HTTP_KINDS = {"endpoint", "asgi", "pod", "sandbox"}

_RUNNER_MODULES = {
    "endpoint": "beta9_amd.runner.endpoint",
    "asgi": "beta9_amd.runner.endpoint",
    "function": "beta9_amd.runner.function",
    "schedule": "beta9_amd.runner.function",
    "taskqueue": "beta9_amd.runner.taskqueue",
    "pod": "beta9_amd.runner.pod",
    "sandbox": "beta9_amd.runner.sandbox",
    "container": "beta9_amd.runner.function",
    "shell": "beta9_amd.runner.pod",
}


def free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class ContainerInstance:
    def __init__(self, request: ContainerRequest):
        self.request = request
        self.port: int = 0
        self.exit_code: Optional[int] = None
        self.started_at = 0.0
        self.done = asyncio.Event()
        self.net_ip: str = ""  # per-container netns IP ("" = host network)
        self.exposed_ports: list = []  # (host_port, container_port)


class Lifecycle:
    def __init__(
        self,
        worker_id: str,
        repo,  # DirectRepoClient | RemoteRepoClient
        runtime: Runtime,
        gpu_manager: Optional[GPUManager],
        gateway_host: str,
        gateway_grpc_port: int,
        worker_host: str = "127.0.0.1",
        prewarm=None,  # PrewarmPool
        image_store_path: str = "",
        isolation: str = "native",
        mask_paths=None,
        netplane=None,  # NetworkPlane
        data_root: str = "",
    ):
        self.worker_id = worker_id
        self.repo = repo
        self.runtime = runtime
        self.gpu = gpu_manager
        self.gateway_host = gateway_host
        self.gateway_grpc_port = gateway_grpc_port
        self.worker_host = worker_host
        self.prewarm = prewarm
        self.image_store_path = image_store_path
        self.images = None  # ImageStore, lazy (store may not exist yet)
        self.isolation = isolation
        self.mask_paths = list(mask_paths or [])
        self.netplane = netplane
        self.data_root = data_root
        self.containers: Dict[str, ContainerInstance] = {}
        from .usage import UsageTracker

        self.usage = UsageTracker(repo)
        # b9launch-backed runtime: THE DEFAULT for user code when the host
        # permits namespaces (reference: runc is the default; bare host
        # processes are the explicit opt-out via isolation="proc")
        self.native_runtime = None
        if type(runtime).__name__ == "NativeRuntime":
            self.native_runtime = runtime
        else:
            try:
                from .runtime.native import NativeRuntime, namespaces_available

                if namespaces_available():
                    self.native_runtime = NativeRuntime()
            except Exception:
                self.native_runtime = None
        self._container_runtime: Dict[str, object] = {}

    def _native_default(self) -> bool:
        return self.native_runtime is not None and self.isolation == "native"

    def _runtime_for(self, spec) -> "Runtime":
        needs_native = (
            spec.mounts
            or spec.block_network
            or spec.image_lowerdirs
            or spec.upper_dir
            or self._native_default()
        )
        if self.native_runtime is not None and needs_native:
            self._container_runtime[spec.container_id] = self.native_runtime
            return self.native_runtime
        self._container_runtime[spec.container_id] = self.runtime
        return self.runtime

    def runtime_of(self, container_id: str):
        return self._container_runtime.get(container_id, self.runtime)

    async def run_container(self, request: ContainerRequest) -> int:
        cid = request.container_id
        inst = ContainerInstance(request)
        self.containers[cid] = inst

        def sink(span: Dict) -> None:
            asyncio.ensure_future(self.repo.push_span(cid, span))

        rec = SpanRecorder(cid, sink=sink)
        start_total = time.time()
        gpu_assignment = None
        try:
            with rec.span("worker.spec_from_request"):
                spec, needs_addr = self._spec_from_request(request, inst)

            if any(isinstance(m, dict) and not m.get("system") for m in spec.mounts):
                with rec.span("worker.setup_mounts"):
                    await self._materialize_mounts(spec)

            if request.requires_gpu():
                with rec.span("worker.gpu_assignment"):
                    if self.gpu is None or not self.gpu.can_fit(
                        max(1, request.gpu_count), request.gpu_fraction
                    ):
                        raise AllocationError("no GPUs available on worker")
                    gpu_assignment = self.gpu.assign(
                        cid, max(1, request.gpu_count), fraction=request.gpu_fraction
                    )
                    spec.env.update(gpu_assignment.env)
                    spec.devices.extend(gpu_assignment.devices)
                    rdzv_env = self.gpu.provision_rendezvous(gpu_assignment)
                    spec.env.update(rdzv_env)
                    if request.rendezvous is not None:
                        spec.env.update(request.rendezvous.env)

            if request.checkpoint is not None and request.checkpoint_enabled:
                # Attach the hipsnap checkpoint location; the runner streams
                # HBM state back through pinned staging during boot.
                spec.env["B9_CHECKPOINT_KEY"] = request.checkpoint.remote_key
                spec.env["B9_CHECKPOINT_ID"] = request.checkpoint.checkpoint_id
                with rec.span("worker.checkpoint_materialize"):
                    local_dir = await self._materialize_checkpoint(request.checkpoint, spec.env)
                    if local_dir:
                        spec.env["B9_CHECKPOINT_DIR"] = local_dir

            # per-container network namespace (pods/sandboxes + any request
            # exposing ports): preallocated slot with veth/bridge/IP
            # (reference: network.go Setup + the slot pool :542-624)
            net_slot = None
            kind = StubType(request.stub_type).kind if request.stub_type else "function"
            if (
                self.netplane is not None
                and self._native_default()
                and not request.block_network
                and (request.ports or kind in ("pod", "sandbox"))
            ):
                with rec.span("worker.network_setup"):
                    net_slot, pool_hit = await self.netplane.acquire(cid)
                    spec.netns_fd = net_slot.ns_fd()
                    inst.net_ip = net_slot.ip
                    if pool_hit:
                        rec.record("worker.network_slot_hit", time.time(), time.time())
                if request.ports:
                    with rec.span("worker.network_expose_ports"):
                        for cport in request.ports:
                            hport = free_port()
                            await self.netplane.expose_port(hport, net_slot.ip, cport)
                            inst.exposed_ports.append((hport, cport))
                if inst.port and self.worker_host != "127.0.0.1":
                    # remote gateways can't route to the bridge subnet:
                    # proxy the runner's bind port at the same host port
                    await self.netplane.expose_port(inst.port, net_slot.ip, inst.port)
                    inst.exposed_ports.append((inst.port, inst.port))

            with rec.span("worker.runtime_start"):
                on_log = self._make_log_cb(cid)
                handle = None
                runtime = self._runtime_for(spec)
                # Prewarmed shell path: torch+HIP context already up on the
                # assigned GPU; the shell becomes the container. (Shells are
                # themselves b9launch containers when isolation is on, so a
                # prewarm hit never bypasses the isolation default.)
                user_mounts = [m for m in spec.mounts if not m.get("system")]
                prewarm_ok = (
                    self.prewarm is not None
                    and hasattr(runtime, "adopt")
                    and not user_mounts
                    and not spec.image_lowerdirs
                    and not spec.upper_dir
                    and not spec.block_network
                    and spec.env.get("B9_DISABLE_PREWARM") != "1"
                    # shared-GPU containers boot cold: exact-GPU shells have
                    # a full-size arena already reserved, which would bust
                    # the fractional HBM budget the runner must enforce
                    and request.gpu_fraction >= 1.0
                    and self.prewarm.sees_path(spec.env.get("B9_WORKSPACE_DIR", ""))
                )
                if prewarm_ok:
                    from .prewarm import runner_module_from_entry

                    module = runner_module_from_entry(spec.entry_point)
                    ordinals = gpu_assignment.ordinals if gpu_assignment else []
                    if module is not None and len(ordinals) <= 1:
                        shell = await self.prewarm.acquire(ordinals)
                        if shell is not None:
                            nfd = net_slot.ns_fd() if net_slot is not None else None
                            if await shell.activate(spec.env, module, netns_fd=nfd):
                                handle = await runtime.adopt(spec, shell.proc, on_log)
                                rec.record("worker.prewarm_hit", time.time(), time.time())
                            else:
                                try:
                                    shell.proc.kill()
                                except ProcessLookupError:
                                    pass
                if handle is None:
                    handle = await runtime.run(spec, on_log=on_log)
                inst.started_at = time.time()

            await self.repo.update_container_status(cid, ContainerStatus.running.value)
            try:
                await self.repo.push_event(
                    "container.started", container_id=cid, stub_id=request.stub_id,
                    worker_id=self.worker_id, gpu=request.gpu, gpu_count=request.gpu_count,
                )
            except Exception:
                pass
            self.usage.container_started(
                cid, gpu=request.gpu, gpu_count=request.gpu_count,
                cpu=request.cpu, memory_mb=request.memory_mb,
                gpu_fraction=request.gpu_fraction,
            )
            if needs_addr:
                with rec.span("worker.publish_address"):
                    # netns containers are addressed by their bridge IP on
                    # this host; remote gateways reach them through the
                    # exposed-port proxies
                    host = inst.net_ip if (inst.net_ip and self.worker_host == "127.0.0.1") else self.worker_host
                    addr = f"{host}:{inst.port}"
                    await self.repo.set_container_address(cid, request.stub_id, addr)

            rec.record("worker.cold_start", start_total, time.time())

            stop_watch = asyncio.create_task(self._watch_stop(cid))
            try:
                exit_code = await runtime.wait(cid)
            finally:
                stop_watch.cancel()
            inst.exit_code = exit_code
            return exit_code
        except AllocationError as e:
            log.warning("gpu allocation failed for %s: %s", cid, e)
            inst.exit_code = 125
            return 125
        except Exception:
            log.exception("container %s failed", cid)
            inst.exit_code = 1
            return 1
        finally:
            usage_rec = self.usage.container_stopped(cid, inst.exit_code)
            if usage_rec is not None:
                rec.record("worker.usage", start_total, time.time(), **usage_rec)
            try:
                name = "container.oom" if inst.exit_code == 137 else "container.stopped"
                await self.repo.push_event(
                    name, container_id=cid, stub_id=request.stub_id,
                    worker_id=self.worker_id, exit_code=inst.exit_code,
                    duration_s=(usage_rec or {}).get("duration_s"),
                )
            except Exception:
                pass
            with rec.span("worker.cleanup"):
                if self.netplane is not None:
                    for hport, cport in inst.exposed_ports:
                        try:
                            await self.netplane.unexpose_port(hport, cport)
                        except Exception:
                            pass
                    try:
                        await self.netplane.release(cid)
                    except Exception:
                        pass
                if gpu_assignment is not None and self.gpu is not None:
                    self.gpu.unassign(cid)
                await self.runtime_of(cid).delete(cid)
                self._container_runtime.pop(cid, None)
                if inst.port:
                    await self.repo.remove_container_address(cid, request.stub_id)
                await self.repo.set_exit_code(cid, inst.exit_code if inst.exit_code is not None else 1)
                await self.repo.delete_container_state(cid)
            inst.done.set()
            self.containers.pop(cid, None)

    async def _materialize_checkpoint(self, cp, env: Dict[str, str]) -> str:
        """Make the hipsnap snapshot (+objsnap) present on THIS node.

        Single-node: the gateway-provided B9_CHECKPOINT_DIR is a shared
        path and the files already exist — no work. Remote workers pull
        the content-store digests over the gateway's CacheService (the
        durable tier — reference: checkpoint pulled from cache/S3 on
        restore, pkg/worker/criu.go:147-233)."""
        gw_dir = env.get("B9_CHECKPOINT_DIR", "")
        snap_name = f"{cp.remote_key}.hipsnap"
        if gw_dir and os.path.exists(os.path.join(gw_dir, snap_name)):
            return ""  # shared filesystem: nothing to do
        if not cp.cache_digest:
            return ""
        base = self.data_root or "/var/lib/b9/worker"
        local_dir = os.path.join(base, "pulled-checkpoints")
        os.makedirs(local_dir, exist_ok=True)
        dst = os.path.join(local_dir, snap_name)
        if not os.path.exists(dst):
            ok = await self._cache_pull(cp.cache_digest, dst)
            if not ok:
                return ""
        if cp.obj_digest and not os.path.exists(dst + ".obj"):
            await self._cache_pull(cp.obj_digest, dst + ".obj")
        return local_dir

    async def _materialize_mounts(self, spec) -> None:
        """Make volume mounts present on THIS node. Shared-fs/single-node:
        paths exist, no-op. Remote workers: read-only volumes are snapshot
        + pulled over the cache (the offline GeeseFS/S3 stand-in — L8);
        writable volumes without shared storage fail loudly."""
        for m in spec.mounts:
            if not isinstance(m, dict) or m.get("system"):
                continue
            local = m.get("local_path", "")
            if not local or os.path.exists(local):
                continue
            if not m.get("read_only"):
                raise RuntimeError(
                    f"writable volume {local} is not on this node and has no "
                    "shared storage; mark it read_only or mount shared storage"
                )
            rpc = getattr(self.repo, "_rpc", None)
            if rpc is None:
                raise RuntimeError(f"volume {local} missing on embedded worker")
            r = await rpc.call(
                "WorkerRepositoryService", "SnapshotVolumeContent", {"path": local}
            )
            digest = r["digest"]
            base = self.data_root or "/var/lib/b9/worker"
            dst = os.path.join(base, "pulled-volumes", digest)
            if not os.path.isdir(dst):
                tarball = dst + ".tar"
                if not await self._cache_pull(digest, tarball):
                    raise RuntimeError(f"volume pull failed for {local}")
                import tarfile as _tar

                os.makedirs(dst + ".tmp", exist_ok=True)
                with _tar.open(tarball) as tf:
                    tf.extractall(dst + ".tmp")
                os.rename(dst + ".tmp", dst)
                os.unlink(tarball)
            m["local_path"] = dst

    async def _cache_pull(self, digest: str, dst: str) -> bool:
        """Stream one content-store object from the gateway's CacheService
        (over the worker's authenticated gRPC channel)."""
        rpc = getattr(self.repo, "_rpc", None)
        if rpc is None:
            return False  # embedded worker shares the gateway filesystem
        tmp = dst + ".pull"
        try:
            with open(tmp, "wb") as f:
                async for chunk in rpc.stream(
                    "CacheService", "GetContentStream", {"hash": digest}
                ):
                    if isinstance(chunk, dict):
                        chunk = chunk.get("data", b"")
                    f.write(chunk)
            os.rename(tmp, dst)
            return True
        except Exception:
            log.exception("checkpoint pull %s failed", digest)
            try:
                os.unlink(tmp)
            except OSError:
                pass
            return False

    def _image_store(self):
        if self.images is None and self.image_store_path:
            from ..images.store import ImageStore

            self.images = ImageStore(self.image_store_path)
        return self.images

    def _build_spec(self, request: ContainerRequest, inst: ContainerInstance):
        """Spec for an image-build container (reference: build containers,
        pkg/abstractions/image/build.go:89-343). Steps run inside the
        container runtime; the product leaves via bind mounts (prefix) or
        the host-backed overlay upper (rootfs)."""
        import json as _json

        b = request.build
        env = dict(request.env)
        env.setdefault("CONTAINER_ID", request.container_id)
        env.setdefault("STUB_ID", request.stub_id)
        env["B9_BUILD_STEPS"] = _json.dumps(b.steps)
        env["B9_BUILD_MODE"] = b.mode
        mounts = []
        lowerdirs: list = []
        upper = work = ""
        native_ok = self.native_runtime is not None
        if b.mode == "prefix":
            site_host = os.path.join(b.build_dir, "site")
            files_host = os.path.join(b.build_dir, "files")
            os.makedirs(site_host, exist_ok=True)
            os.makedirs(files_host, exist_ok=True)
            if native_ok:
                mounts = [
                    {"local_path": site_host, "mount_path": "/b9img/site"},
                    {"local_path": files_host, "mount_path": "/b9img/files"},
                ]
                env["IMAGE_SITE"] = "/b9img/site"
                env["PREFIX"] = "/b9img/files"
            else:
                # no namespace privileges on this host: steps still run in a
                # worker-side container process, product written directly
                env["IMAGE_SITE"] = site_host
                env["PREFIX"] = files_host
            entry = [sys.executable, "-u", "-m", "beta9_amd.runner.build"]
        else:  # rootfs
            if not native_ok:
                raise RuntimeError("rootfs builds need namespace privileges (b9launch)")
            store = self._image_store()
            lowerdirs = store.lowerdirs(b.base_image_id)
            upper = os.path.join(b.build_dir, "upper")
            work = os.path.join(b.build_dir, "work")
            os.makedirs(upper, exist_ok=True)
            os.makedirs(work, exist_ok=True)
            repo_root = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
            mounts = [{"local_path": repo_root, "mount_path": "/opt/b9", "read_only": True}]
            env["PYTHONPATH"] = "/opt/b9"
            cfg_env = (store.get_manifest(b.base_image_id) or {}).get("config", {}).get("env") or []
            for kv in cfg_env:
                k, _, v = kv.partition("=")
                env.setdefault(k, v)
            env.setdefault("PATH", "/usr/local/sbin:/usr/local/bin:/usr/sbin:/usr/bin:/sbin:/bin")
            if b.steps and all(s.get("kind") == "shell" for s in b.steps):
                # shell-only builds run with the image's own /bin/sh — no
                # python required inside minimal base images
                script = " && ".join(s.get("cmd", "true") for s in b.steps)
                entry = ["/bin/sh", "-ec", script]
            else:
                entry = ["python3", "-u", "-m", "beta9_amd.runner.build"]
        spec = ContainerSpec(
            container_id=request.container_id,
            entry_point=entry,
            env=env,
            cpu_millicores=request.cpu,
            memory_mb=request.memory_mb,
            hostname=request.container_id,
            mounts=mounts,
            image_lowerdirs=lowerdirs,
            upper_dir=upper,
            work_dir=work,
        )
        return spec, False

    def _spec_from_request(self, request: ContainerRequest, inst: ContainerInstance):
        if request.build is not None:
            return self._build_spec(request, inst)
        kind = StubType(request.stub_type).kind if request.stub_type else "function"
        needs_addr = kind in _HTTP_KINDS
        env = dict(request.env)
        env.setdefault("CONTAINER_ID", request.container_id)
        env.setdefault("STUB_ID", request.stub_id)
        env.setdefault("STUB_TYPE", request.stub_type)
        env.setdefault("WORKSPACE_ID", request.workspace_id)
        # the gateway may bind 0.0.0.0 (multi-node); containers must dial
        # the address THIS worker reaches it at
        if env.get("B9_GATEWAY_HOST", "") in ("", "0.0.0.0", "::"):
            env["B9_GATEWAY_HOST"] = self.gateway_host
        env.setdefault("B9_GATEWAY_PORT", str(self.gateway_grpc_port))
        env.setdefault("CONTAINER_HOSTNAME", request.hostname or request.container_id)
        env.setdefault("B9_LIGHT_IMPORT", "1")  # runners skip the SDK surface
        if self.image_store_path:
            env.setdefault("B9_IMAGE_STORE", self.image_store_path)
        if request.task_id:
            env.setdefault("TASK_ID", request.task_id)
        lowerdirs: list = []
        extra_mounts: list = []
        rootfs_manifest = None
        if request.image_id and self.image_store_path:
            store = self._image_store()
            m = store.get_manifest(request.image_id) if store else None
            if m is not None and m.get("kind") == "rootfs" and self.native_runtime is not None:
                # OCI-layer rootfs: overlay lowerdir chain; our runner rides
                # in via a read-only bind (reference mounts the SDK runner
                # into containers the same way)
                rootfs_manifest = m
                lowerdirs = store.lowerdirs(request.image_id)
                repo_root = os.path.dirname(
                    os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
                )
                extra_mounts.append(
                    {"local_path": repo_root, "mount_path": "/opt/b9", "read_only": True}
                )
                env.setdefault("PYTHONPATH", "/opt/b9")
                for kv in (m.get("config", {}).get("env") or []):
                    k, _, v = kv.partition("=")
                    env.setdefault(k, v)
                env.setdefault(
                    "PATH", "/usr/local/sbin:/usr/local/bin:/usr/sbin:/usr/bin:/sbin:/bin"
                )
            else:
                # prefix image: site dir onto PYTHONPATH + files tree
                from ..abstractions.image import image_env

                env.update(image_env(self.image_store_path, request.image_id))
                if self._native_default() and self.mask_paths:
                    # re-expose the image store through the data-root mask
                    # ("system" mounts don't disqualify the prewarm path)
                    extra_mounts.append(
                        {
                            "local_path": self.image_store_path,
                            "mount_path": self.image_store_path,
                            "system": True,
                        }
                    )
        wd = env.get("B9_WORKSPACE_DIR", "")
        if wd and self._native_default() and os.path.isdir(wd):
            # fresh /tmp + masks hide host paths: the synced workspace code
            # rides in on an explicit system bind
            extra_mounts.append(
                {"local_path": wd, "mount_path": wd, "system": True}
            )
        if needs_addr:
            inst.port = free_port()
            env["BIND_PORT"] = str(inst.port)
        entry = list(request.entry_point)
        if not entry and rootfs_manifest is not None:
            entry = list(rootfs_manifest.get("config", {}).get("entrypoint") or [])
        if not entry:
            module = _RUNNER_MODULES.get(kind, "beta9_amd.runner.function")
            if rootfs_manifest is not None:
                entry = ["python3", "-u", "-m", module]
            else:
                entry = [sys.executable, "-u", "-m", module]
        spec = ContainerSpec(
            container_id=request.container_id,
            entry_point=entry,
            env=env,
            cwd=(rootfs_manifest or {}).get("config", {}).get("cwd") or "",
            cpu_millicores=request.cpu,
            memory_mb=request.memory_mb,
            hostname=request.hostname or request.container_id,
            block_network=request.block_network,
            mounts=[m.model_dump() for m in request.mounts] + extra_mounts,
            image_lowerdirs=lowerdirs,
            mask_paths=self.mask_paths,
        )
        return spec, needs_addr

    def _make_log_cb(self, cid: str):
        def cb(stream: str, line: str) -> None:
            asyncio.ensure_future(self.repo.append_log(cid, line))

        return cb

    async def stop_container(self, cid: str) -> None:
        """Graceful-then-hard kill (driven by the worker's stop-event
        stream; also the slow-poll fallback's action)."""
        log.info("stop requested for %s", cid)
        rt = self.runtime_of(cid)
        await rt.kill(cid, 15)
        await asyncio.sleep(5.0)
        if cid in self.containers:
            await rt.kill(cid, 9)

    async def _watch_stop(self, cid: str) -> None:
        """SLOW-poll safety net for stop requests that raced the event
        stream (the fast path is the worker's single stop-event stream —
        reference: processStopContainerEvents via keyspace events)."""
        try:
            if await self.repo.stop_requested(cid):  # raced container start
                await self.stop_container(cid)
                return
            while True:
                await asyncio.sleep(2.0)
                if await self.repo.stop_requested(cid):
                    await self.stop_container(cid)
                    return
        except asyncio.CancelledError:
            pass
