"""Isolation is the DEFAULT for user code (VERDICT round-1 item #1/#2):
every container — including prewarm-pool hits — runs under b9launch
namespaces; the gateway DB and other tenants' state are masked out.

Reference parity target: runc is the reference's default runtime
(pkg/runtime/runc.go); bare host processes were round-1's shortcut.
"""

import asyncio
import os

import pytest

import beta9_amd
from beta9_amd.worker.runtime.native import namespaces_available

from .helpers import cluster

needs_ns = pytest.mark.skipif(not namespaces_available(), reason="no namespace privileges")


@needs_ns
@pytest.mark.asyncio
async def test_masks_hide_gateway_state():
    async with cluster(n_workers=1) as c:
        import handlers

        # tests run the DB in-memory; plant a sentinel where the on-disk DB
        # (and its WAL) would live — the db/ dir is tmpfs-masked
        os.makedirs(os.path.join(c.tmpdir, "db"), exist_ok=True)
        db_path = os.path.join(c.tmpdir, "db", "backend.db")
        with open(db_path, "w") as f:
            f.write("sqlite-tokens")
        assert os.path.exists(db_path)  # host side: DB exists
        ws_root = c.cfg.storage.root
        os.makedirs(ws_root, exist_ok=True)
        secret_file = os.path.join(ws_root, "tenant-secret.txt")
        with open(secret_file, "w") as f:
            f.write("other-tenant-data")

        fn = beta9_amd.Function(handlers.probe_paths, cpu=100, memory=128)
        out = await asyncio.to_thread(
            fn.remote, paths=[db_path, secret_file, "/proc/1/cmdline"]
        )
        assert out[db_path] is False, "gateway DB visible inside container"
        assert out[secret_file] is False, "workspace storage visible inside container"
        # /proc is the container's own (fresh procfs in the PID namespace)
        assert out["/proc/1/cmdline"] is True


@needs_ns
@pytest.mark.asyncio
async def test_prewarm_hit_is_namespaced():
    async with cluster(n_workers=1) as c:
        import handlers

        w = c.workers[0]
        assert w.prewarm_pool is not None and w.prewarm_pool.isolate
        await w.wait_prewarmed(timeout=60)

        fn = beta9_amd.Function(handlers.pidns_probe, cpu=100, memory=128)
        out = await asyncio.to_thread(fn.remote)
        assert out["canary"] is True  # wrote inside its overlay
        assert not os.path.exists("/isolation-canary")  # host untouched
        assert out["pid"] < 100, f"pid {out['pid']}: not in a fresh PID namespace"
