"""Durable-disk abstraction service (reference: pkg/abstractions/disk/ — 4
RPCs + REST; rows in Postgres, volumes attached to stubs)."""

from __future__ import annotations

from typing import Any, Dict

from ..rpc.channel import invalid, not_found, rpc
from ..worker.disk import DiskManager


class DiskService:
    service_name = "DiskService"

    def __init__(self, gateway, disk_manager: DiskManager):
        self.gw = gateway
        self.disks = disk_manager

    @rpc()
    async def CreateDisk(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        ws = auth["workspace_id"]
        name = req["name"]
        size_gb = int(req.get("size_gb", 10))
        if self.gw.db.get_disk(ws, name) is not None:
            raise invalid(f"disk {name} exists")
        info = self.disks.create(ws, name, size_gb)
        self.gw.db.create_disk(ws, name, size_gb, info["path"])
        return info

    @rpc()
    async def ListDisks(self, req: Dict[str, Any], context, auth) -> list:
        return self.gw.db.list_disks(auth["workspace_id"])

    @rpc()
    async def DeleteDisk(self, req: Dict[str, Any], context, auth) -> bool:
        ws = auth["workspace_id"]
        self.disks.delete(ws, req["name"])
        self.gw.db.delete_disk(ws, req["name"])
        return True

    @rpc()
    async def SnapshotDisk(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        ws = auth["workspace_id"]
        if self.gw.db.get_disk(ws, req["name"]) is None:
            raise not_found(req["name"])
        return self.disks.snapshot(ws, req["name"], req.get("snapshot_name", "snap"))

    @rpc()
    async def AttachDisk(self, req: Dict[str, Any], context, auth) -> Dict[str, Any]:
        """Attach and return the host path (bind-mounted into containers
        that list this disk)."""
        ws = auth["workspace_id"]
        if self.gw.db.get_disk(ws, req["name"]) is None:
            raise not_found(req["name"])
        path = self.disks.attach(ws, req["name"])
        return {"path": path}
