"""Image service: deterministic image ids, in-container builds, OCI import.

Reference behavior (reference: pkg/abstractions/image/ — VerifyImageBuild
computes a deterministic id from the build recipe, image.go:83 +
image_id.go:17-60; BuildImage schedules build steps INTO A CONTAINER in the
build pool and streams logs, build.go:62-343; images mount lazily from a
content-addressed store, pkg/worker/image.go:278-348).

MI355X-native design — two image kinds, both served from the
content-addressed store (beta9_amd/images/store.py):

- "prefix" images (no base): the host ROCm environment is the base; the
  build produces a site-packages dir + files tree that the runner prepends
  (PYTHONPATH / bind-mounts). This keeps torch-ROCm + our HIP extensions
  zero-copy on the cold-start path.
- "rootfs" images (base_image given): OCI layers imported from a local OCI
  layout / docker-save archive (`from_registry` in the offline world),
  stacked as overlayfs lowerdirs by b9launch; builds capture the overlay
  upper as a new content-addressed layer.

Build steps NEVER run on the gateway host: BuildImage creates a
ContainerRequest (stub_type image/build) through the scheduler onto a
worker, which runs beta9_amd/runner/build.py inside the container runtime.
"""

from __future__ import annotations

import asyncio
import hashlib
import json
import os
import shutil
import sys
import time
import uuid
from typing import Any, AsyncIterator, Dict, List

from ..images.store import ImageStore
from ..rpc.channel import invalid, not_found, rpc
from ..types import BuildSpec, ContainerRequest


def compute_image_id(
    python_version: str = "",
    python_packages: List[str] = (),
    commands: List[str] = (),
    base_image: str = "",
    env_vars: List[str] = (),
) -> str:
    """Deterministic image id from the build recipe (reference:
    image_id.go:17-60 hashstructure of build opts)."""
    recipe = json.dumps(
        {
            "python_version": python_version,
            "python_packages": sorted(python_packages),
            "commands": list(commands),
            "base_image": base_image,
            "env_vars": sorted(env_vars),
        },
        sort_keys=True,
    )
    return hashlib.sha256(recipe.encode()).hexdigest()[:24]


class ImageService:
    service_name = "ImageService"

    def __init__(self, gateway):
        self.gw = gateway
        self.store_path = gateway.cfg.image_service.store_path
        os.makedirs(self.store_path, exist_ok=True)
        self.store = ImageStore(self.store_path)

    def image_dir(self, image_id: str) -> str:
        return os.path.join(self.store_path, image_id)

    def image_exists(self, image_id: str) -> bool:
        return (
            os.path.exists(os.path.join(self.image_dir(image_id), ".manifest.json"))
            or self.store.has_image(image_id)
        )

    # -- RPCs ----------------------------------------------------------------
    @rpc()
    async def VerifyImageBuild(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        image_id = compute_image_id(
            python_version=req.get("python_version", ""),
            python_packages=req.get("python_packages", []),
            commands=req.get("commands", []),
            base_image=req.get("base_image", ""),
            env_vars=req.get("env_vars", []),
        )
        return {"image_id": image_id, "exists": self.image_exists(image_id)}

    @rpc()
    async def ImportImage(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Import an OCI image layout dir / OCI-archive / docker-save tar
        into the layer store (the offline `from_registry`). Admin-gated:
        reads a host path."""
        if auth.get("token_type") != "cluster_admin":
            from ..rpc.channel import permission_denied

            raise permission_denied("image import reads host paths: admin only")
        path = req.get("path", "")
        if not path or not os.path.exists(path):
            raise invalid(f"no such path: {path}")
        image_id = self.store.import_oci(path, image_id=req.get("image_id") or None)
        manifest = self.store.get_manifest(image_id)
        self.gw.db.upsert_image(image_id, "available", json.dumps(manifest), auth["workspace_id"])
        return {"image_id": image_id, "layers": len(manifest["layers"])}

    @rpc()
    async def GetImage(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        image_id = req.get("image_id", "")
        if not self.image_exists(image_id):
            raise not_found(f"image {image_id}")
        m = self.store.get_manifest(image_id)
        if m is None:
            m = {"image_id": image_id, "kind": "prefix", "layers": []}
        return m

    @rpc(kind="unary_stream")
    async def BuildImage(self, req: Dict[str, Any], context, auth) -> AsyncIterator[Dict]:
        """Schedule build steps into a container and stream its logs
        (reference: build.go:89-343 — never on the gateway host)."""
        image_id = compute_image_id(
            python_version=req.get("python_version", ""),
            python_packages=req.get("python_packages", []),
            commands=req.get("commands", []),
            base_image=req.get("base_image", ""),
            env_vars=req.get("env_vars", []),
        )
        if self.image_exists(image_id):
            yield {"type": "done", "image_id": image_id, "cached": True}
            return

        base_image = req.get("base_image", "")
        mode = "prefix"
        base_image_id = ""
        if base_image:
            # resolve: an already-imported image id, or a local OCI path
            if self.store.has_image(base_image):
                base_image_id = base_image
            elif os.path.exists(base_image):
                base_image_id = self.store.import_oci(base_image)
                yield {"type": "log", "line": f"imported base image -> {base_image_id}"}
            else:
                yield {
                    "type": "error",
                    "line": f"base image '{base_image}' not found (offline: give a "
                    "local OCI layout dir / docker-save tar, or an imported image id)",
                }
                return
            mode = "rootfs"

        steps: List[Dict[str, Any]] = []
        pkgs = req.get("python_packages", [])
        if pkgs:
            steps.append({"kind": "pip", "packages": list(pkgs)})
        for cmd in req.get("commands", []):
            steps.append({"kind": "shell", "cmd": cmd})

        build_dir = os.path.join(self.store_path, ".building", image_id + "-" + uuid.uuid4().hex[:6])
        os.makedirs(build_dir, exist_ok=True)
        spec = BuildSpec(
            image_id=image_id,
            mode=mode,
            base_image_id=base_image_id,
            build_dir=build_dir,
            steps=steps,
        )
        cid = f"build-{image_id[:12]}-{uuid.uuid4().hex[:6]}"
        request = ContainerRequest(
            container_id=cid,
            stub_id=f"image-build-{image_id[:12]}",
            stub_type="image/build",
            workspace_id=auth["workspace_id"],
            cpu=self.gw.cfg.image_service.build_cpu,
            memory_mb=self.gw.cfg.image_service.build_memory_mb,
            build=spec,
            env={"B9_BUILD_IMAGE_ID": image_id},
        )
        try:
            await self.gw.scheduler.run(request)
        except Exception as e:  # noqa: BLE001
            yield {"type": "error", "line": f"build scheduling failed: {e}"}
            return

        # stream container logs until it exits
        sent = 0
        deadline = time.monotonic() + self.gw.cfg.image_service.build_timeout_s
        exit_code = None
        repo = self.gw.container_repo
        while time.monotonic() < deadline:
            lines = await repo.get_logs(cid, tail=100000)
            for line in lines[sent:]:
                yield {"type": "log", "line": line}
            sent = len(lines)
            code = await repo.get_exit_code(cid)
            if code is not None:
                exit_code = code
                break
            st = await repo.get_state(cid)
            if st is None and sent > 0:
                # state gone without an exit code: treat as failure below
                exit_code = await repo.get_exit_code(cid)
                break
            await asyncio.sleep(0.1)
        lines = await repo.get_logs(cid, tail=100000)
        for line in lines[sent:]:
            yield {"type": "log", "line": line}

        if exit_code != 0:
            self.gw.db.upsert_image(image_id, "failed", "", auth["workspace_id"])
            shutil.rmtree(build_dir, ignore_errors=True)
            yield {
                "type": "error",
                "line": f"build container exited with code {exit_code}",
            }
            return
        try:
            self._finalize_build(spec, req, auth["workspace_id"])
        except Exception as e:  # noqa: BLE001
            self.gw.db.upsert_image(image_id, "failed", "", auth["workspace_id"])
            yield {"type": "error", "line": f"finalize failed: {e}"}
            return
        finally:
            shutil.rmtree(build_dir, ignore_errors=True)
        yield {"type": "done", "image_id": image_id, "cached": False}

    # -- finalize ------------------------------------------------------------
    def _finalize_build(self, spec: BuildSpec, req: Dict[str, Any], ws_id: str) -> None:
        image_id = spec.image_id
        if spec.mode == "prefix":
            img = self.image_dir(image_id)
            os.makedirs(img, exist_ok=True)
            for sub in ("site", "files"):
                src = os.path.join(spec.build_dir, sub)
                dst = os.path.join(img, sub)
                if os.path.exists(dst):
                    shutil.rmtree(dst)
                if os.path.exists(src):
                    shutil.move(src, dst)
                else:
                    os.makedirs(dst, exist_ok=True)
            manifest = {
                "image_id": image_id,
                "python_version": req.get("python_version", ""),
                "env_vars": req.get("env_vars", []),
            }
            with open(os.path.join(img, ".manifest.json"), "w") as f:
                json.dump(manifest, f)
            self.store.put_manifest(image_id, [], {"env": req.get("env_vars", [])}, kind="prefix")
        else:
            upper = os.path.join(spec.build_dir, "upper")
            base = self.store.get_manifest(spec.base_image_id)
            if base is None:
                raise RuntimeError(f"base image {spec.base_image_id} vanished")
            layers = list(base["layers"])
            if os.path.exists(upper) and os.listdir(upper):
                delta = self.store.layers.adopt_layer_dir(upper, move=True)
                layers.append(delta)
            config = dict(base.get("config") or {})
            config["env"] = list(config.get("env") or []) + list(req.get("env_vars") or [])
            self.store.put_manifest(image_id, layers, config, kind="rootfs")
        self.gw.db.upsert_image(
            image_id, "available", json.dumps({"mode": spec.mode}), ws_id
        )


def image_env(store_path: str, image_id: str) -> Dict[str, str]:
    """Env injection for a container using a PREFIX image (site dir onto
    PYTHONPATH, files tree exported). Rootfs images are handled by the
    runtime (overlay lowerdirs), not env."""
    if not image_id:
        return {}
    img = os.path.join(store_path, image_id)
    site = os.path.join(img, "site")
    env: Dict[str, str] = {"B9_IMAGE_ID": image_id, "B9_IMAGE_FILES": os.path.join(img, "files")}
    if os.path.isdir(site):
        env["B9_IMAGE_SITE"] = site
    try:
        with open(os.path.join(img, ".manifest.json")) as f:
            manifest = json.load(f)
        for kv in manifest.get("env_vars", []):
            k, _, v = kv.partition("=")
            env[k] = v
    except OSError:
        pass
    return env
