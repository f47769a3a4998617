"""Image service: deterministic image ids + builds.

Reference behavior (reference: pkg/abstractions/image/ — VerifyImageBuild
computes a deterministic id from the build recipe, image.go:83 +
image_id.go:17-60; BuildImage streams build logs while steps run). Images in
this single-node MI355X build are content-addressed overlay prefixes (a
site-packages dir + files tree) that the runner prepends to its environment —
a rootfs-less equivalent of the reference's CLIP archives suited to the proc
runtime; the native runtime mounts the same prefix as an overlayfs lower
layer. Layer payloads are stored in the worker content cache so 8 concurrent
cold starts on one node read one copy (beta9_amd/worker/cache.py).
"""

from __future__ import annotations

import asyncio
import hashlib
import json
import os
import sys
from typing import Any, AsyncIterator, Dict, List

from ..rpc.channel import invalid, rpc


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

    def image_dir(self, image_id: str) -> str:
        return os.path.join(self.store_path, image_id)

    def image_exists(self, image_id: str) -> bool:
        return os.path.exists(os.path.join(self.image_dir(image_id), ".manifest.json"))

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

    @rpc(kind="unary_stream")
    async def BuildImage(self, req: Dict[str, Any], context, auth) -> AsyncIterator[Dict]:
        """Run build steps into a content-addressed prefix, streaming logs.

        Steps run with PREFIX pointing at the image files tree and pip
        installs targeting the image site dir. (The reference schedules build
        containers into a build pool — build.go:62-343; on a single node we
        execute directly with the same step semantics.)
        """
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
        img = self.image_dir(image_id)
        site = os.path.join(img, "site")
        files = os.path.join(img, "files")
        os.makedirs(site, exist_ok=True)
        os.makedirs(files, exist_ok=True)
        steps: List[List[str]] = []
        pkgs = req.get("python_packages", [])
        if pkgs:
            steps.append(
                [sys.executable, "-m", "pip", "install", "--no-index", "--target", site, *pkgs]
            )
        for cmd in req.get("commands", []):
            steps.append(["bash", "-c", cmd])
        ok = True
        for step in steps:
            yield {"type": "log", "line": f"+ {' '.join(step)}"}
            proc = await asyncio.create_subprocess_exec(
                *step,
                cwd=files,
                env={**os.environ, "PREFIX": files, "IMAGE_SITE": site},
                stdout=asyncio.subprocess.PIPE,
                stderr=asyncio.subprocess.STDOUT,
            )
            assert proc.stdout is not None
            async for line in proc.stdout:
                yield {"type": "log", "line": line.decode(errors="replace").rstrip()}
            code = await proc.wait()
            if code != 0:
                ok = False
                yield {"type": "error", "line": f"step failed with exit code {code}"}
                break
        if ok:
            manifest = {
                "image_id": image_id,
                "python_version": req.get("python_version", ""),
                "env_vars": req.get("env_vars", []),
            }
            with open(os.path.join(img, ".manifest.json"), "w") as f:
                json.dump(manifest, f)
            self.gw.db.upsert_image(image_id, "available", json.dumps(manifest), auth["workspace_id"])
            yield {"type": "done", "image_id": image_id, "cached": False}
        else:
            self.gw.db.upsert_image(image_id, "failed", "", auth["workspace_id"])


def image_env(store_path: str, image_id: str) -> Dict[str, str]:
    """Env injection for a container using image `image_id` (consumed by the
    runner: site dir onto PYTHONPATH, files tree exported)."""
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
