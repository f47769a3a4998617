"""Typed event pipeline.

(reference: pkg/types/event.go schemas + pkg/repository/events*.go sinks —
container lifecycle, task lifecycle, worker lifecycle events to S2/HTTP
sinks). This build keeps a ring in the state store (queryable via
/api/v1/events) plus an optional JSONL file sink; the schema field names
match the reference's event envelope (name, schema_version, created_at,
payload).
"""

from __future__ import annotations

import asyncio
import json
import time
from typing import Any, Dict, List, Optional

from . import keys
from .state import StateStore

SCHEMA_VERSION = "1.0"

# event names (reference: pkg/types/event.go catalog)
CONTAINER_SCHEDULED = "container.scheduled"
CONTAINER_STARTED = "container.started"
CONTAINER_STOPPED = "container.stopped"
CONTAINER_OOM = "container.oom"
TASK_CREATED = "task.created"
TASK_COMPLETED = "task.completed"
WORKER_REGISTERED = "worker.registered"
WORKER_REMOVED = "worker.removed"
CHECKPOINT_CREATED = "checkpoint.created"
DEPLOYMENT_CREATED = "deployment.created"


class EventSink:
    def __init__(self, store: StateStore, jsonl_path: str = "", ring_cap: int = 5000):
        self.store = store
        self.jsonl_path = jsonl_path
        self.ring_cap = ring_cap
        self._file = open(jsonl_path, "a") if jsonl_path else None

    async def emit(self, name: str, **payload: Any) -> None:
        event = {
            "name": name,
            "schema_version": SCHEMA_VERSION,
            "created_at": time.time(),
            "payload": payload,
        }
        key = keys.events_stream()
        await self.store.rpush(key, event)
        if await self.store.llen(key) > self.ring_cap:
            await self.store.lpop(key)
        if self._file is not None:
            self._file.write(json.dumps(event, default=str) + "\n")
            self._file.flush()

    def emit_nowait(self, name: str, **payload: Any) -> None:
        asyncio.ensure_future(self.emit(name, **payload))

    async def recent(self, limit: int = 100, name_prefix: str = "") -> List[Dict]:
        events = await self.store.lrange(keys.events_stream(), 0, -1)
        if name_prefix:
            events = [e for e in events if e.get("name", "").startswith(name_prefix)]
        return events[-limit:]

    def close(self) -> None:
        if self._file is not None:
            self._file.close()
            self._file = None
