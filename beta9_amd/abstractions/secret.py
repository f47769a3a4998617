"""Workspace secrets CRUD; injected as env into containers (reference:
pkg/abstractions/secret/ — 5 RPCs)."""

from __future__ import annotations

from typing import Any, Dict

from ..backend.db import BackendDB
from ..rpc.channel import not_found, rpc


class SecretService:
    service_name = "SecretService"

    def __init__(self, db: BackendDB):
        self.db = db

    @rpc()
    async def SecretCreate(self, req: Dict[str, Any], context, auth) -> bool:
        self.db.set_secret(auth["workspace_id"], req["name"], req["value"])
        return True

    @rpc()
    async def SecretGet(self, req: Dict[str, Any], context, auth) -> str:
        v = self.db.get_secret(auth["workspace_id"], req["name"])
        if v is None:
            raise not_found(f"secret {req['name']}")
        return v

    @rpc()
    async def SecretUpdate(self, req: Dict[str, Any], context, auth) -> bool:
        self.db.set_secret(auth["workspace_id"], req["name"], req["value"])
        return True

    @rpc()
    async def SecretList(self, req: Dict[str, Any], context, auth) -> list:
        return self.db.list_secrets(auth["workspace_id"])

    @rpc()
    async def SecretDelete(self, req: Dict[str, Any], context, auth) -> bool:
        self.db.delete_secret(auth["workspace_id"], req["name"])
        return True
