"""Pod + Sandbox abstraction.

Pods are arbitrary long-running containers with exposed ports; sandboxes add
a process-manager surface (exec / run_code / fs ops) (reference:
pkg/abstractions/pod/ — pod.proto 24 RPCs incl. the Sandbox file/process
surface; worker-side goproc process manager, pkg/worker/sandbox.go:468-573).

This build runs the process manager INSIDE the container as a first-party
msgpack-gRPC server (beta9_amd/runner/sandbox.py — the goproc equivalent);
the gateway proxies Sandbox* RPCs straight to the container address (single
node, no per-node ContainerService hop needed).
"""

from __future__ import annotations

import asyncio
import os
import logging
import time
from typing import Any, Dict, Optional

from ..common import keys
from ..rpc.channel import RpcClient, not_found, rpc
from ..types import ContainerRequest, StubConfig, new_id

log = logging.getLogger(__name__)


class PodService:
    service_name = "PodService"

    def __init__(self, gateway):
        self.gw = gateway

    async def _schedule_pod(self, auth: Dict, stub_id: str, wait_timeout: float = 60.0) -> Dict[str, Any]:
        stub = self.gw.db.get_stub(stub_id)
        if stub is None:
            raise not_found(f"stub {stub_id}")
        cfg: StubConfig = self.gw.db.get_stub_config(stub_id)
        env = self.gw.runner_env(auth["workspace_id"], stub, cfg)
        env.update(cfg.env)
        if cfg.entry_point:
            env["POD_ENTRYPOINT"] = " ".join(cfg.entry_point)
        container_id = f"{stub['type'].split('/')[0]}-{stub_id[-8:]}-{new_id()}"
        request = ContainerRequest(
            container_id=container_id,
            env=env,
            cpu=cfg.runtime.cpu,
            memory_mb=cfg.runtime.memory_mb,
            gpu=cfg.runtime.gpu,
            gpu_request=list(cfg.runtime.gpus),
            gpu_count=cfg.runtime.gpu_count,
            gpu_fraction=cfg.runtime.gpu_fraction,
            image_id=cfg.runtime.image_id,
            stub_id=stub_id,
            workspace_id=auth["workspace_id"],
            stub_type=stub["type"],
            pool_selector=cfg.pool,
            mounts=self.gw.resolve_mounts(auth["workspace_id"], cfg.volumes),
            ports=list(cfg.ports),
            timestamp=time.time(),
        )
        await self.gw.scheduler.run(request)
        # wait for the container address, then probe the PM port until it
        # accepts (reference: waitForProcessManager pkg/worker/sandbox.go:468)
        deadline = time.monotonic() + wait_timeout
        addr = None
        while time.monotonic() < deadline:
            addr = await self.gw.container_repo.get_address(container_id)
            if addr:
                break
            code = await self.gw.container_repo.get_exit_code(container_id)
            if code is not None:
                raise not_found(f"container exited early with code {code}")
            await asyncio.sleep(0.01)
        ready = False
        if addr:
            host, _, port = addr.partition(":")
            while time.monotonic() < deadline:
                try:
                    _r, w = await asyncio.wait_for(
                        asyncio.open_connection(host, int(port)), timeout=1.0
                    )
                    w.close()
                    ready = True
                    break
                except OSError:
                    await asyncio.sleep(0.02)
                except asyncio.TimeoutError:
                    pass
        return {"container_id": container_id, "addr": addr or "", "ok": ready}

    @rpc()
    async def CreatePod(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._schedule_pod(auth, req["stub_id"], float(req.get("timeout", 60.0)))

    @rpc()
    async def StopPod(self, req: Dict[str, Any], context, auth) -> bool:
        await self.gw.scheduler.stop(req["container_id"])
        return True

    @rpc()
    async def ListPods(self, req: Dict[str, Any], context, auth) -> list:
        out = []
        for cid, status in (await self.gw.container_repo.containers_for_stub(req["stub_id"])).items():
            addr = await self.gw.container_repo.get_address(cid)
            out.append({"container_id": cid, "status": status, "addr": addr})
        return out

    # -- sandbox process-manager proxy --------------------------------------
    async def _pm_call(self, container_id: str, method: str, payload: Dict[str, Any]) -> Any:
        addr = await self.gw.container_repo.get_address(container_id)
        if not addr:
            raise not_found(f"container {container_id} has no address")
        client = RpcClient(addr)
        try:
            return await client.call("ProcessManager", method, payload, timeout=float(payload.get("timeout", 120.0)) + 10.0)
        finally:
            await client.close()

    @rpc()
    async def SandboxExec(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "Exec", req)

    @rpc()
    async def SandboxRunContainer(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Nested isolated container inside a sandbox (the Docker-in-sandbox
        analog — reference: pkg/worker/sandbox.go:148-465)."""
        return await self._pm_call(req["container_id"], "RunContainer", req)

    # -- interactive terminal (PTY in the PM; the dropbear/shell analog) ----
    @rpc()
    async def SandboxTerminalOpen(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "TerminalOpen", req)

    @rpc()
    async def SandboxTerminalRead(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "TerminalRead", req)

    @rpc()
    async def SandboxTerminalWrite(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "TerminalWrite", req)

    @rpc()
    async def SandboxTerminalResize(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "TerminalResize", req)

    @rpc()
    async def SandboxTerminalClose(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "TerminalClose", req)

    @rpc()
    async def SandboxRunCode(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "RunCode", req)

    @rpc()
    async def SandboxStatus(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        return await self._pm_call(req["container_id"], "Status", req)

    @rpc()
    async def SandboxListProcesses(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "ListProcesses", req)

    @rpc()
    async def SandboxKillProcess(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "KillProcess", req)

    @rpc()
    async def SandboxUploadFile(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "UploadFile", req)

    @rpc()
    async def SandboxDownloadFile(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "DownloadFile", req)

    @rpc()
    async def SandboxListFiles(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "ListFiles", req)

    @rpc()
    async def SandboxStatFile(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "StatFile", req)

    @rpc()
    async def SandboxDeleteFile(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "DeleteFile", req)

    @rpc()
    async def SandboxReplaceInFiles(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "ReplaceInFiles", req)

    @rpc()
    async def SandboxFindInFiles(self, req: Dict[str, Any], context, auth) -> Any:
        return await self._pm_call(req["container_id"], "FindInFiles", req)

    @rpc()
    async def SandboxCreateImageFromFilesystem(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Snapshot a live sandbox filesystem into a new content-addressed
        image usable by future stubs (reference: pod.proto
        SandboxCreateImageFromFilesystem + sandbox.py:527-549): the in-
        container PM archives the subtree; the gateway unpacks it into an
        image files tree and registers the manifest."""
        import hashlib
        import io
        import json as _json
        import tarfile

        r = await self._pm_call(req["container_id"], "SnapshotFilesystem", req)
        if r.get("error"):
            return r
        archive = r["archive"]
        image_id = "snap-" + hashlib.sha256(archive).hexdigest()[:20]
        img_dir = os.path.join(self.gw.cfg.image_service.store_path, image_id)
        files = os.path.join(img_dir, "files")
        if not os.path.exists(os.path.join(img_dir, ".manifest.json")):
            os.makedirs(files, exist_ok=True)
            with tarfile.open(fileobj=io.BytesIO(archive), mode="r:gz") as tf:
                for m in tf.getmembers():
                    # refuse path traversal
                    if m.name.startswith("/") or ".." in m.name.split("/"):
                        continue
                    tf.extract(m, files)
            manifest = {"image_id": image_id, "env_vars": [], "source": "sandbox-filesystem"}
            with open(os.path.join(img_dir, ".manifest.json"), "w") as f:
                _json.dump(manifest, f)
            self.gw.db.upsert_image(image_id, "available", _json.dumps(manifest), auth["workspace_id"])
        return {"image_id": image_id, "size": r["size"]}

    @rpc()
    async def SandboxSnapshotMemory(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Reference parity stub (pod.proto:147 SandboxSnapshotMemory uses
        CRIU, which does not exist for arbitrary processes here). GPU state
        snapshot/restore is first-party via hipsnap for checkpoint-enabled
        stubs (runner/common.py GPUBoot + CheckpointReady/CommitCheckpoint);
        arbitrary-pod CPU memory snapshots are not supported."""
        return {
            "supported": False,
            "reason": "use checkpoint_enabled stubs (hipsnap) for GPU state snapshots",
        }

    @rpc()
    async def SandboxExposePort(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Expose a container port through the gateway TCP ingress
        (reference: pod tcp.go raw TCP ingress)."""
        cid = req["container_id"]
        addr = await self.gw.container_repo.get_address(cid)
        if not addr:
            raise not_found(f"container {cid}")
        host = addr.split(":")[0]
        if getattr(self.gw, "tcp_ingress", None) is None:
            from ..gateway.tcp_ingress import TcpIngress

            self.gw.tcp_ingress = TcpIngress()
        port = await self.gw.tcp_ingress.expose(cid, host, int(req["port"]))
        out = {
            "url": f"http://{self.gw.cfg.gateway.host}:{port}",
            "host": self.gw.cfg.gateway.host,
            "port": port,
            "target": f"{host}:{req['port']}",
        }
        if req.get("tls"):
            # SNI TLS ingress (reference: pod tcp.go TLS listener with
            # per-stub SNI certs): one shared TLS port, routed by hostname
            # {container_id}.tcp.{base_domain}, self-signed per-host cert
            if getattr(self.gw, "sni_ingress", None) is None:
                from ..gateway.tcp_ingress import SniTlsIngress

                self.gw.sni_ingress = SniTlsIngress(
                    base_domain=self.gw.cfg.gateway.base_domain
                )
                await self.gw.sni_ingress.start()
            self.gw.sni_ingress.add_route(cid, host, int(req["port"]))
            sni_host = f"{cid}.tcp.{self.gw.cfg.gateway.base_domain}"
            out["tls_port"] = self.gw.sni_ingress.port
            out["tls_sni"] = sni_host
            out["tls_url"] = f"https://{sni_host}:{self.gw.sni_ingress.port}"
        return out
