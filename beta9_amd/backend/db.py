"""Backend metadata store (workspaces, tokens, stubs, deployments, tasks,
secrets, objects).

The reference keeps these rows in Postgres with embedded migrations
(reference: pkg/repository/backend_postgres.go + backend_postgres_migrations/).
This framework embeds sqlite so a single MI355X node runs with zero external
services; the row shapes and the deployment-versioning semantics are kept.
"""

from __future__ import annotations

import json
import os
import sqlite3
import threading
import time
import uuid
from typing import Any, Dict, List, Optional

from ..types import StubConfig, StubType, new_id

_SCHEMA = """
CREATE TABLE IF NOT EXISTS workspaces (
  id TEXT PRIMARY KEY,
  name TEXT UNIQUE NOT NULL,
  created_at REAL NOT NULL,
  concurrency_limit_cpu INTEGER DEFAULT 0,
  concurrency_limit_gpu INTEGER DEFAULT 0,
  multi_gpu_enabled INTEGER DEFAULT 1
);
CREATE TABLE IF NOT EXISTS tokens (
  key TEXT PRIMARY KEY,
  workspace_id TEXT NOT NULL,
  active INTEGER DEFAULT 1,
  created_at REAL NOT NULL,
  token_type TEXT DEFAULT 'workspace'
);
CREATE TABLE IF NOT EXISTS apps (
  id TEXT PRIMARY KEY,
  workspace_id TEXT NOT NULL,
  name TEXT NOT NULL,
  created_at REAL NOT NULL,
  UNIQUE(workspace_id, name)
);
CREATE TABLE IF NOT EXISTS stubs (
  id TEXT PRIMARY KEY,
  workspace_id TEXT NOT NULL,
  app_id TEXT DEFAULT '',
  name TEXT NOT NULL,
  type TEXT NOT NULL,
  config TEXT NOT NULL,
  created_at REAL NOT NULL
);
CREATE TABLE IF NOT EXISTS deployments (
  id TEXT PRIMARY KEY,
  workspace_id TEXT NOT NULL,
  app_id TEXT DEFAULT '',
  name TEXT NOT NULL,
  stub_id TEXT NOT NULL,
  stub_type TEXT NOT NULL,
  version INTEGER NOT NULL,
  active INTEGER DEFAULT 1,
  subdomain TEXT DEFAULT '',
  created_at REAL NOT NULL
);
CREATE TABLE IF NOT EXISTS tasks (
  id TEXT PRIMARY KEY,
  workspace_id TEXT NOT NULL,
  stub_id TEXT NOT NULL,
  status TEXT NOT NULL,
  container_id TEXT DEFAULT '',
  created_at REAL NOT NULL,
  started_at REAL DEFAULT 0,
  ended_at REAL DEFAULT 0,
  retries INTEGER DEFAULT 0
);
CREATE INDEX IF NOT EXISTS idx_tasks_stub ON tasks(stub_id, created_at);
CREATE TABLE IF NOT EXISTS objects (
  id TEXT PRIMARY KEY,
  workspace_id TEXT NOT NULL,
  hash TEXT NOT NULL,
  size INTEGER NOT NULL,
  path TEXT NOT NULL,
  created_at REAL NOT NULL
);
CREATE TABLE IF NOT EXISTS secrets (
  workspace_id TEXT NOT NULL,
  name TEXT NOT NULL,
  value TEXT NOT NULL,
  created_at REAL NOT NULL,
  updated_at REAL NOT NULL,
  PRIMARY KEY (workspace_id, name)
);
CREATE TABLE IF NOT EXISTS checkpoints (
  id TEXT PRIMARY KEY,
  workspace_id TEXT NOT NULL,
  stub_id TEXT NOT NULL,
  source_container_id TEXT DEFAULT '',
  status TEXT NOT NULL,
  remote_key TEXT DEFAULT '',
  stub_type TEXT DEFAULT '',
  accelerator TEXT DEFAULT '',
  gpu_count INTEGER DEFAULT 0,
  size_bytes INTEGER DEFAULT 0,
  created_at REAL NOT NULL,
  cache_digest TEXT DEFAULT '',
  obj_digest TEXT DEFAULT ''
);
CREATE TABLE IF NOT EXISTS images (
  id TEXT PRIMARY KEY,
  workspace_id TEXT DEFAULT '',
  status TEXT NOT NULL,
  manifest TEXT DEFAULT '',
  created_at REAL NOT NULL
);
CREATE TABLE IF NOT EXISTS disks (
  workspace_id TEXT NOT NULL,
  name TEXT NOT NULL,
  size_gb INTEGER NOT NULL,
  filesystem TEXT DEFAULT 'ext4',
  path TEXT DEFAULT '',
  created_at REAL NOT NULL,
  PRIMARY KEY (workspace_id, name)
);
"""


class BackendDB:
    def __init__(self, path: str = ":memory:"):
        if path != ":memory:":
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        self._conn = sqlite3.connect(path, check_same_thread=False)
        self._conn.row_factory = sqlite3.Row
        self._lock = threading.Lock()
        self._stub_cache: Dict[str, Dict[str, Any]] = {}
        with self._lock:
            self._conn.execute("PRAGMA journal_mode=WAL")
            self._conn.execute("PRAGMA synchronous=NORMAL")
            self._conn.executescript(_SCHEMA)
            for mig in (
                "ALTER TABLE tokens ADD COLUMN token_type TEXT DEFAULT 'workspace'",
                "ALTER TABLE checkpoints ADD COLUMN cache_digest TEXT DEFAULT ''",
                "ALTER TABLE checkpoints ADD COLUMN obj_digest TEXT DEFAULT ''",
            ):
                try:  # migrate older DB files in place
                    self._conn.execute(mig)
                except sqlite3.OperationalError:
                    pass
            self._conn.commit()
        # write-behind queue for hot-path task rows (authoritative task
        # state lives in the state store; rows are the durable audit log)
        import queue as _queue

        self._wq: "_queue.Queue" = _queue.Queue()
        self._writer = threading.Thread(target=self._write_loop, daemon=True)
        self._writer_stop = threading.Event()
        self._writer.start()

    def _write_loop(self) -> None:
        import queue as _queue

        while not self._writer_stop.is_set():
            try:
                item = self._wq.get(timeout=0.2)
            except _queue.Empty:
                continue
            batch = [item]
            try:
                while len(batch) < 256:
                    batch.append(self._wq.get_nowait())
            except _queue.Empty:
                pass
            with self._lock:
                for sql, args in batch:
                    try:
                        self._conn.execute(sql, args)
                    except sqlite3.Error:
                        pass
                try:
                    self._conn.commit()
                except sqlite3.Error:
                    pass

    def flush(self, timeout: float = 5.0) -> None:
        """Drain the write-behind queue (tests / shutdown)."""
        deadline = time.time() + timeout
        while not self._wq.empty() and time.time() < deadline:
            time.sleep(0.005)

    def close(self) -> None:
        self.flush()
        self._writer_stop.set()
        self._writer.join(timeout=2.0)
        self._conn.close()

    def _exec(self, sql: str, args: tuple = ()) -> sqlite3.Cursor:
        with self._lock:
            cur = self._conn.execute(sql, args)
            self._conn.commit()
            return cur

    def _query(self, sql: str, args: tuple = ()) -> List[sqlite3.Row]:
        with self._lock:
            return self._conn.execute(sql, args).fetchall()

    # -- workspaces / tokens ------------------------------------------------
    def create_workspace(self, name: str) -> Dict[str, Any]:
        ws_id = new_id("ws-")
        self._exec(
            "INSERT INTO workspaces (id, name, created_at) VALUES (?,?,?)",
            (ws_id, name, time.time()),
        )
        return {"id": ws_id, "name": name}

    def get_workspace(self, ws_id: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM workspaces WHERE id=?", (ws_id,))
        return dict(rows[0]) if rows else None

    def get_workspace_by_name(self, name: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM workspaces WHERE name=?", (name,))
        return dict(rows[0]) if rows else None

    def get_or_create_signing_key(self, workspace_id: str) -> str:
        """Per-workspace HMAC key for signed public-endpoint payloads
        (reference: auth SignPayload surface)."""
        self._exec(
            "CREATE TABLE IF NOT EXISTS workspace_keys "
            "(workspace_id TEXT PRIMARY KEY, key TEXT)", ()
        )
        rows = self._query(
            "SELECT key FROM workspace_keys WHERE workspace_id=?", (workspace_id,)
        )
        if rows:
            return rows[0]["key"]
        import secrets as _secrets

        key = _secrets.token_hex(32)
        self._exec(
            "INSERT OR IGNORE INTO workspace_keys (workspace_id, key) VALUES (?,?)",
            (workspace_id, key),
        )
        rows = self._query(
            "SELECT key FROM workspace_keys WHERE workspace_id=?", (workspace_id,)
        )
        return rows[0]["key"]

    def create_token(self, workspace_id: str, token_type: str = "workspace") -> str:
        """token_type 'workspace' (tenant-scoped) or 'cluster_admin'
        (reference: pkg/auth token types gate cross-workspace ops,
        pkg/auth/http.go:106)."""
        key = uuid.uuid4().hex + uuid.uuid4().hex[:8]
        self._exec(
            "INSERT INTO tokens (key, workspace_id, created_at, token_type) VALUES (?,?,?,?)",
            (key, workspace_id, time.time(), token_type),
        )
        return key

    def authorize_token(self, key: str) -> Optional[Dict[str, Any]]:
        rows = self._query(
            "SELECT t.key, t.workspace_id, t.token_type, w.name as workspace_name FROM tokens t "
            "JOIN workspaces w ON w.id = t.workspace_id WHERE t.key=? AND t.active=1",
            (key,),
        )
        return dict(rows[0]) if rows else None

    def get_token(self, key: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM tokens WHERE key=?", (key,))
        return dict(rows[0]) if rows else None

    def revoke_token(self, key: str) -> None:
        self._exec("UPDATE tokens SET active=0 WHERE key=?", (key,))

    # -- apps ----------------------------------------------------------------
    def get_or_create_app(self, workspace_id: str, name: str) -> Dict[str, Any]:
        rows = self._query(
            "SELECT * FROM apps WHERE workspace_id=? AND name=?", (workspace_id, name)
        )
        if rows:
            return dict(rows[0])
        app_id = new_id("app-")
        self._exec(
            "INSERT INTO apps (id, workspace_id, name, created_at) VALUES (?,?,?,?)",
            (app_id, workspace_id, name, time.time()),
        )
        return {"id": app_id, "workspace_id": workspace_id, "name": name}

    def list_apps(self, workspace_id: str) -> List[Dict[str, Any]]:
        return [dict(r) for r in self._query("SELECT * FROM apps WHERE workspace_id=?", (workspace_id,))]

    # -- stubs ----------------------------------------------------------------
    def get_or_create_stub(
        self,
        workspace_id: str,
        name: str,
        stub_type: str,
        config: StubConfig,
        app_id: str = "",
        force_create: bool = False,
    ) -> Dict[str, Any]:
        cfg_json = config.model_dump_json()
        if not force_create:
            rows = self._query(
                "SELECT * FROM stubs WHERE workspace_id=? AND name=? AND type=? AND config=?",
                (workspace_id, name, stub_type, cfg_json),
            )
            if rows:
                return dict(rows[0])
        stub_id = new_id("stub-")
        self._exec(
            "INSERT INTO stubs (id, workspace_id, app_id, name, type, config, created_at) VALUES (?,?,?,?,?,?,?)",
            (stub_id, workspace_id, app_id, name, stub_type, cfg_json, time.time()),
        )
        return {
            "id": stub_id,
            "workspace_id": workspace_id,
            "app_id": app_id,
            "name": name,
            "type": stub_type,
            "config": cfg_json,
        }

    def get_stub(self, stub_id: str) -> Optional[Dict[str, Any]]:
        # stub rows are immutable after creation (a new config hashes to a
        # new stub) and this runs on every endpoint invoke — cache hits
        # skip the blocking sqlite query on the gateway event loop
        hit = self._stub_cache.get(stub_id)
        if hit is not None:
            return dict(hit)
        rows = self._query("SELECT * FROM stubs WHERE id=?", (stub_id,))
        if not rows:
            return None
        row = dict(rows[0])
        if len(self._stub_cache) > 4096:
            self._stub_cache.clear()
        self._stub_cache[stub_id] = dict(row)
        return row

    def get_stub_config(self, stub_id: str) -> Optional[StubConfig]:
        row = self.get_stub(stub_id)
        if row is None:
            return None
        return StubConfig.model_validate_json(row["config"])

    def update_stub_config(self, stub_id: str, config: StubConfig) -> None:
        self._exec("UPDATE stubs SET config=? WHERE id=?", (config.model_dump_json(), stub_id))
        self._stub_cache.pop(stub_id, None)

    # -- deployments -----------------------------------------------------------
    def create_deployment(
        self, workspace_id: str, name: str, stub_id: str, stub_type: str, app_id: str = ""
    ) -> Dict[str, Any]:
        rows = self._query(
            "SELECT MAX(version) as v FROM deployments WHERE workspace_id=? AND name=?",
            (workspace_id, name),
        )
        version = (rows[0]["v"] or 0) + 1
        dep_id = new_id("dep-")
        self._exec(
            "UPDATE deployments SET active=0 WHERE workspace_id=? AND name=?",
            (workspace_id, name),
        )
        subdomain = f"{name}-{version}"
        self._exec(
            "INSERT INTO deployments (id, workspace_id, app_id, name, stub_id, stub_type, version, active, subdomain, created_at)"
            " VALUES (?,?,?,?,?,?,?,1,?,?)",
            (dep_id, workspace_id, app_id, name, stub_id, stub_type, version, subdomain, time.time()),
        )
        return {
            "id": dep_id,
            "workspace_id": workspace_id,
            "name": name,
            "stub_id": stub_id,
            "stub_type": stub_type,
            "version": version,
            "subdomain": subdomain,
            "active": 1,
        }

    def get_deployment(self, workspace_id: str, name: str, version: Optional[int] = None) -> Optional[Dict[str, Any]]:
        if version is None:
            rows = self._query(
                "SELECT * FROM deployments WHERE workspace_id=? AND name=? AND active=1 ORDER BY version DESC LIMIT 1",
                (workspace_id, name),
            )
        else:
            rows = self._query(
                "SELECT * FROM deployments WHERE workspace_id=? AND name=? AND version=?",
                (workspace_id, name, version),
            )
        return dict(rows[0]) if rows else None

    def list_deployments(self, workspace_id: str) -> List[Dict[str, Any]]:
        return [
            dict(r)
            for r in self._query(
                "SELECT * FROM deployments WHERE workspace_id=? ORDER BY created_at DESC", (workspace_id,)
            )
        ]

    def list_active_deployments(self) -> List[Dict[str, Any]]:
        return [dict(r) for r in self._query("SELECT * FROM deployments WHERE active=1")]

    def stop_deployment(self, dep_id: str) -> None:
        self._exec("UPDATE deployments SET active=0 WHERE id=?", (dep_id,))

    # -- tasks -----------------------------------------------------------------
    def create_task(self, task_id: str, workspace_id: str, stub_id: str, status: str) -> None:
        self._exec(
            "INSERT INTO tasks (id, workspace_id, stub_id, status, created_at) VALUES (?,?,?,?,?)",
            (task_id, workspace_id, stub_id, status, time.time()),
        )

    def create_task_async(self, task_id: str, workspace_id: str, stub_id: str, status: str) -> None:
        self._wq.put(
            (
                "INSERT INTO tasks (id, workspace_id, stub_id, status, created_at) VALUES (?,?,?,?,?)",
                (task_id, workspace_id, stub_id, status, time.time()),
            )
        )

    def update_task(self, task_id: str, **fields: Any) -> None:
        sets = ", ".join(f"{k}=?" for k in fields)
        self._exec(f"UPDATE tasks SET {sets} WHERE id=?", (*fields.values(), task_id))

    def update_task_async(self, task_id: str, **fields: Any) -> None:
        sets = ", ".join(f"{k}=?" for k in fields)
        self._wq.put((f"UPDATE tasks SET {sets} WHERE id=?", (*fields.values(), task_id)))

    def get_task(self, task_id: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM tasks WHERE id=?", (task_id,))
        return dict(rows[0]) if rows else None

    def list_tasks(self, workspace_id: str, stub_id: str = "", limit: int = 100) -> List[Dict[str, Any]]:
        if stub_id:
            rows = self._query(
                "SELECT * FROM tasks WHERE workspace_id=? AND stub_id=? ORDER BY created_at DESC LIMIT ?",
                (workspace_id, stub_id, limit),
            )
        else:
            rows = self._query(
                "SELECT * FROM tasks WHERE workspace_id=? ORDER BY created_at DESC LIMIT ?",
                (workspace_id, limit),
            )
        return [dict(r) for r in rows]

    # -- objects ----------------------------------------------------------------
    def create_object(self, workspace_id: str, obj_hash: str, size: int, path: str) -> str:
        rows = self._query(
            "SELECT id FROM objects WHERE workspace_id=? AND hash=?", (workspace_id, obj_hash)
        )
        if rows:
            return rows[0]["id"]
        obj_id = new_id("obj-")
        self._exec(
            "INSERT INTO objects (id, workspace_id, hash, size, path, created_at) VALUES (?,?,?,?,?,?)",
            (obj_id, workspace_id, obj_hash, size, path, time.time()),
        )
        return obj_id

    def get_object(self, obj_id: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM objects WHERE id=?", (obj_id,))
        return dict(rows[0]) if rows else None

    # -- secrets ----------------------------------------------------------------
    def set_secret(self, workspace_id: str, name: str, value: str) -> None:
        now = time.time()
        self._exec(
            "INSERT INTO secrets (workspace_id, name, value, created_at, updated_at) VALUES (?,?,?,?,?) "
            "ON CONFLICT(workspace_id, name) DO UPDATE SET value=excluded.value, updated_at=excluded.updated_at",
            (workspace_id, name, value, now, now),
        )

    def get_secret(self, workspace_id: str, name: str) -> Optional[str]:
        rows = self._query(
            "SELECT value FROM secrets WHERE workspace_id=? AND name=?", (workspace_id, name)
        )
        return rows[0]["value"] if rows else None

    def list_secrets(self, workspace_id: str) -> List[str]:
        return [r["name"] for r in self._query("SELECT name FROM secrets WHERE workspace_id=?", (workspace_id,))]

    def delete_secret(self, workspace_id: str, name: str) -> None:
        self._exec("DELETE FROM secrets WHERE workspace_id=? AND name=?", (workspace_id, name))

    # -- checkpoints ---------------------------------------------------------------
    def upsert_checkpoint(self, cp: Dict[str, Any]) -> None:
        self._exec(
            "INSERT INTO checkpoints (id, workspace_id, stub_id, source_container_id, status, remote_key, stub_type, accelerator, gpu_count, size_bytes, created_at, cache_digest, obj_digest)"
            " VALUES (?,?,?,?,?,?,?,?,?,?,?,?,?) ON CONFLICT(id) DO UPDATE SET status=excluded.status, remote_key=excluded.remote_key, size_bytes=excluded.size_bytes, cache_digest=excluded.cache_digest, obj_digest=excluded.obj_digest",
            (
                cp["id"], cp["workspace_id"], cp["stub_id"], cp.get("source_container_id", ""),
                cp["status"], cp.get("remote_key", ""), cp.get("stub_type", ""),
                cp.get("accelerator", ""), cp.get("gpu_count", 0), cp.get("size_bytes", 0),
                cp.get("created_at", time.time()),
                cp.get("cache_digest", ""), cp.get("obj_digest", ""),
            ),
        )

    def latest_checkpoint(self, workspace_id: str, stub_id: str, status: str = "available") -> Optional[Dict[str, Any]]:
        rows = self._query(
            "SELECT * FROM checkpoints WHERE workspace_id=? AND stub_id=? AND status=? ORDER BY created_at DESC LIMIT 1",
            (workspace_id, stub_id, status),
        )
        return dict(rows[0]) if rows else None

    # -- images ---------------------------------------------------------------
    def upsert_image(self, image_id: str, status: str, manifest: str = "", workspace_id: str = "") -> None:
        self._exec(
            "INSERT INTO images (id, workspace_id, status, manifest, created_at) VALUES (?,?,?,?,?)"
            " ON CONFLICT(id) DO UPDATE SET status=excluded.status, manifest=excluded.manifest",
            (image_id, workspace_id, status, manifest, time.time()),
        )

    def get_image(self, image_id: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM images WHERE id=?", (image_id,))
        return dict(rows[0]) if rows else None

    # -- disks ---------------------------------------------------------------
    def create_disk(self, workspace_id: str, name: str, size_gb: int, path: str) -> None:
        self._exec(
            "INSERT INTO disks (workspace_id, name, size_gb, path, created_at) VALUES (?,?,?,?,?)",
            (workspace_id, name, size_gb, path, time.time()),
        )

    def get_disk(self, workspace_id: str, name: str) -> Optional[Dict[str, Any]]:
        rows = self._query("SELECT * FROM disks WHERE workspace_id=? AND name=?", (workspace_id, name))
        return dict(rows[0]) if rows else None

    def list_disks(self, workspace_id: str) -> List[Dict[str, Any]]:
        return [dict(r) for r in self._query("SELECT * FROM disks WHERE workspace_id=?", (workspace_id,))]

    def delete_disk(self, workspace_id: str, name: str) -> None:
        self._exec("DELETE FROM disks WHERE workspace_id=? AND name=?", (workspace_id, name))
