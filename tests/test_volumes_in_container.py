"""Volumes mounted into containers (native runtime bind; skipped where
namespaces are unavailable)."""

import asyncio

import pytest

import beta9_amd
from beta9_amd.worker.runtime.native import namespaces_available
from tests.helpers import cluster

needs_ns = pytest.mark.skipif(not namespaces_available(), reason="no namespace privileges")


@needs_ns
@pytest.mark.asyncio
async def test_volume_visible_at_mount_path():
    async with cluster(n_workers=1) as c:
        client = beta9_amd.Client()
        # put a file into the volume through the volume API
        await asyncio.to_thread(
            client.client_stream,
            "VolumeService",
            "UploadFile",
            [{"volume_name": "data", "path": "models/weights.txt", "data": b"W" * 16}],
        )
        import handlers_volume  # noqa: fixture

        fn = beta9_amd.Function(
            handlers_volume.read_mounted,
            cpu=100,
            memory=128,
            volumes=[beta9_amd.Volume(name="data", mount_path="/vol/data")],
        )
        result = await asyncio.to_thread(fn.remote)
        assert result == {"content": "W" * 16, "exists": True}

        # writes from the container persist to the host volume
        fn2 = beta9_amd.Function(
            handlers_volume.write_mounted,
            cpu=100,
            memory=128,
            volumes=[beta9_amd.Volume(name="data", mount_path="/vol/data")],
        )
        assert await asyncio.to_thread(fn2.remote) is True
        ls = await asyncio.to_thread(
            client.call, "VolumeService", "ListPath", {"volume_name": "data", "path": ""}
        )
        names = {e["name"] for e in ls}
        assert "from-container.txt" in names
