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


@pytest.mark.gpu
@pytest.mark.asyncio
async def test_gpu_function_in_native_runtime():
    """GPU container under b9launch: overlay rootfs, minimal /dev with only
    /dev/kfd + the assigned renderD node, sysfs mounted, volume bound."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    if not namespaces_available():
        pytest.skip("no namespace privileges")
    async with cluster(n_workers=0) as c:
        from beta9_amd.worker.gpu import GPUManager
        from beta9_amd.worker.repo_client import DirectRepoClient
        from beta9_amd.worker.worker import Worker

        gm = GPUManager()
        assert gm.total >= 1
        repo = DirectRepoClient(c.gateway.worker_repo, c.gateway.container_repo)
        w = Worker(
            worker_id="gpunat", repo=repo, cpu_limit=32000, memory_limit_mb=131072,
            gpu_type=gm.gpu_type, gpu_manager=gm,
            gateway_host="127.0.0.1", gateway_grpc_port=c.gateway.grpc_port,
            prewarm=False,
        )
        await w.start()
        try:
            client = beta9_amd.Client()
            await asyncio.to_thread(
                client.client_stream, "VolumeService", "UploadFile",
                [{"volume_name": "data", "path": "models/weights.txt", "data": b"W" * 16}],
            )
            import handlers_volume  # noqa

            fn = beta9_amd.Function(
                handlers_volume.gpu_in_native,
                cpu=2000, memory=16384, gpu=gm.gpu_type,
                volumes=[beta9_amd.Volume(name="data", mount_path="/vol/data")],
            )
            r = await asyncio.to_thread(fn.remote)
            assert r["cuda"] is True, r
            assert r["kfd"] is True and r["dri"] is True
            assert r["sum"] == 128.0
            assert r["vol"] is True
        finally:
            await w.shutdown()


@pytest.mark.asyncio
async def test_sandbox_pty_terminal():
    """PTY terminal in the sandbox PM: real terminal semantics (prompt,
    echo, interactive state) — the dropbear/shell parity path."""
    import beta9_amd

    from .helpers import cluster

    async with cluster(n_workers=1) as c:
        sb = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=128).create)
        try:
            term = await asyncio.to_thread(sb.terminal, 24, 80)
            await asyncio.to_thread(term.write, "echo T$((40+2))T\n")
            out = b""
            for _ in range(20):
                data, closed = await asyncio.to_thread(term.read, 2.0)
                out += data
                if b"T42T" in out or closed:
                    break
            assert b"T42T" in out, out[-500:]
            # interactive state persists across writes (shell variable)
            await asyncio.to_thread(term.write, "X=hello\n")
            await asyncio.to_thread(term.write, "echo GOT:$X\n")
            out2 = b""
            for _ in range(20):
                data, closed = await asyncio.to_thread(term.read, 2.0)
                out2 += data
                if b"GOT:hello" in out2 or closed:
                    break
            assert b"GOT:hello" in out2, out2[-500:]
            await asyncio.to_thread(term.close)
        finally:
            await asyncio.to_thread(sb.terminate)


@pytest.mark.asyncio
async def test_async_sandbox_api():
    """AsyncSandbox: the asyncio twin of the Sandbox surface (reference:
    sdk sandbox.py sync+asyncio APIs)."""
    import beta9_amd

    from .helpers import cluster

    async with cluster(n_workers=1) as c:
        inst = await beta9_amd.AsyncSandbox(cpu=100, memory=128).create()
        try:
            r = await inst.run_code("print(6*7)")
            assert "42" in r["stdout"]
            await inst.upload("/tmp/workspace/a.txt", b"async-bytes")
            assert await inst.download("/tmp/workspace/a.txt") == b"async-bytes"
            names = [f["name"] for f in await inst.ls("/tmp/workspace")]
            assert "a.txt" in names
            st = await inst.status()
            assert st.get("ok") or st.get("running", True)
        finally:
            await inst.terminate()
            await inst.close()
