"""Workspace volumes: ls/cp/rm/mv + upload/download (reference:
pkg/abstractions/volume/ — 13 RPCs incl. presigned multipart; this
single-node build serves volume bytes through the gateway directly, with the
same SDK surface; workspace storage lives under storage.root/<ws>/volumes)."""

from __future__ import annotations

import base64
import os
import shutil
from typing import Any, AsyncIterator, Dict

from ..backend.db import BackendDB
from ..rpc.channel import invalid, not_found, rpc


class VolumeService:
    service_name = "VolumeService"

    def __init__(self, db: BackendDB, storage_root: str):
        self.db = db
        self.storage_root = storage_root

    def _vol_root(self, auth: Dict, volume_name: str) -> str:
        p = os.path.join(self.storage_root, auth["workspace_id"], "volumes", volume_name)
        os.makedirs(p, exist_ok=True)
        return p

    def _safe(self, root: str, rel: str) -> str:
        p = os.path.realpath(os.path.join(root, rel.lstrip("/")))
        if not p.startswith(os.path.realpath(root)):
            raise invalid(f"path escapes volume: {rel}")
        return p

    @rpc()
    async def GetOrCreateVolume(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        root = self._vol_root(auth, req["name"])
        return {"name": req["name"], "path": root}

    @rpc()
    async def ListPath(self, req: Dict[str, Any], context, auth) -> list:
        root = self._vol_root(auth, req["volume_name"])
        p = self._safe(root, req.get("path", ""))
        if not os.path.exists(p):
            return []
        out = []
        if os.path.isfile(p):
            st = os.stat(p)
            return [{"name": os.path.basename(p), "is_dir": False, "size": st.st_size}]
        for name in sorted(os.listdir(p)):
            fp = os.path.join(p, name)
            st = os.stat(fp)
            out.append({"name": name, "is_dir": os.path.isdir(fp), "size": st.st_size})
        return out

    @rpc()
    async def DeletePath(self, req: Dict[str, Any], context, auth) -> bool:
        root = self._vol_root(auth, req["volume_name"])
        p = self._safe(root, req["path"])
        if os.path.isdir(p):
            shutil.rmtree(p, ignore_errors=True)
        elif os.path.exists(p):
            os.unlink(p)
        return True

    @rpc()
    async def MovePath(self, req: Dict[str, Any], context, auth) -> bool:
        root = self._vol_root(auth, req["volume_name"])
        src = self._safe(root, req["src"])
        dst = self._safe(root, req["dst"])
        os.makedirs(os.path.dirname(dst), exist_ok=True)
        shutil.move(src, dst)
        return True

    @rpc()
    async def CopyPath(self, req: Dict[str, Any], context, auth) -> bool:
        root = self._vol_root(auth, req["volume_name"])
        src = self._safe(root, req["src"])
        dst = self._safe(root, req["dst"])
        os.makedirs(os.path.dirname(dst), exist_ok=True)
        if os.path.isdir(src):
            shutil.copytree(src, dst, dirs_exist_ok=True)
        else:
            shutil.copy2(src, dst)
        return True

    @rpc(kind="stream_unary")
    async def UploadFile(self, request_iter, context, auth) -> Dict[str, Any]:
        """Chunked upload: first message carries volume_name+path, the rest
        carry data chunks."""
        f = None
        path = ""
        written = 0
        async for msg in request_iter:
            if f is None:
                root = self._vol_root(auth, msg["volume_name"])
                path = self._safe(root, msg["path"])
                os.makedirs(os.path.dirname(path), exist_ok=True)
                f = open(path + ".tmp", "wb")
            data = msg.get("data", b"")
            if data:
                f.write(data)
                written += len(data)
        if f is None:
            raise invalid("empty upload stream")
        f.close()
        os.rename(path + ".tmp", path)
        return {"path": path, "size": written}

    @rpc(kind="unary_stream")
    async def DownloadFile(self, req: Dict[str, Any], context, auth) -> AsyncIterator[Dict]:
        root = self._vol_root(auth, req["volume_name"])
        p = self._safe(root, req["path"])
        if not os.path.isfile(p):
            raise not_found(req["path"])
        with open(p, "rb") as f:
            while True:
                chunk = f.read(1 << 20)
                if not chunk:
                    return
                yield {"data": chunk}
