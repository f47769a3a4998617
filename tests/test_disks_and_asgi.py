"""Durable disks + true-ASGI apps + shell exec surface."""

import asyncio
import os

import httpx
import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_disk_create_attach_snapshot_delete():
    async with cluster(n_workers=0) as c:
        client = beta9_amd.Client()
        info = await asyncio.to_thread(
            client.call, "DiskService", "CreateDisk", {"name": "d1", "size_gb": 1}
        )
        assert info["backend"] in ("loop-ext4", "dir")
        disks = await asyncio.to_thread(client.call, "DiskService", "ListDisks", {})
        assert disks[0]["name"] == "d1"
        att = await asyncio.to_thread(client.call, "DiskService", "AttachDisk", {"name": "d1"})
        # write into the attached disk
        with open(os.path.join(att["path"], "hello.txt"), "w") as f:
            f.write("persist")
        snap = await asyncio.to_thread(
            client.call, "DiskService", "SnapshotDisk", {"name": "d1", "snapshot_name": "s1"}
        )
        assert os.path.exists(snap["path"])
        await asyncio.to_thread(client.call, "DiskService", "DeleteDisk", {"name": "d1"})
        assert await asyncio.to_thread(client.call, "DiskService", "ListDisks", {}) == []


@pytest.mark.asyncio
async def test_asgi_user_app():
    async with cluster(n_workers=1) as c:
        import handlers_asgi  # fixture

        app_def = beta9_amd.asgi(cpu=100, memory=256)(handlers_asgi.make_app)
        stub_id = await asyncio.to_thread(app_def.prepare_runtime)
        # invoke through the gateway HTTP asgi route
        async with httpx.AsyncClient() as client:
            r = await client.post(
                f"http://127.0.0.1:{c.gateway.http_port}/asgi/id/{stub_id}/double",
                json={"x": 21},
                headers={"authorization": f"Bearer {c.token}"},
                timeout=60.0,
            )
            assert r.status_code == 200, r.text
            assert r.json()["doubled"] == 42
