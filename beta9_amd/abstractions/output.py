"""Task output files: save -> public URL (reference: pkg/abstractions/output/
— 3 RPCs: save, presigned public URL, stat)."""

from __future__ import annotations

import os
import time
from typing import Any, AsyncIterator, Dict

from ..backend.db import BackendDB
from ..rpc.channel import invalid, not_found, rpc
from ..types import new_id


class OutputService:
    service_name = "OutputService"

    def __init__(self, db: BackendDB, storage_root: str):
        self.db = db
        self.storage_root = storage_root

    def _out_root(self, auth: Dict, task_id: str) -> str:
        p = os.path.join(self.storage_root, auth["workspace_id"], "outputs", task_id)
        os.makedirs(p, exist_ok=True)
        return p

    @rpc(kind="stream_unary")
    async def OutputSaveStream(self, request_iter, context, auth) -> Dict[str, Any]:
        f = None
        output_id = new_id("out-")
        path = ""
        async for msg in request_iter:
            if f is None:
                root = self._out_root(auth, msg["task_id"])
                fname = os.path.basename(msg.get("filename", output_id))
                path = os.path.join(root, fname)
                f = open(path, "wb")
            data = msg.get("data", b"")
            if data:
                f.write(data)
        if f is None:
            raise invalid("empty output stream")
        f.close()
        return {"output_id": output_id, "path": path}

    @rpc()
    async def OutputPublicURL(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Mint a retrieval URL served by the gateway HTTP API."""
        path = req["path"]
        if not os.path.isfile(path):
            raise not_found(path)
        rel = os.path.relpath(path, self.storage_root)
        return {"url": f"/api/v1/outputs/{rel}", "expires": time.time() + float(req.get("ttl", 3600))}

    @rpc()
    async def OutputStat(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        path = req["path"]
        if not os.path.isfile(path):
            return {"exists": False}
        st = os.stat(path)
        return {"exists": True, "size": st.st_size, "mtime": st.st_mtime}
