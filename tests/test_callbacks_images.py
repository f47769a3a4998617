"""Task callbacks, image env injection into containers, log streaming."""

import asyncio

import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_callback_url_fired():
    received = []

    # tiny raw asyncio HTTP receiver (uvicorn-in-test stalls the shared loop)
    import json as _json

    async def handle(reader, writer):
        data = b""
        while b"\r\n\r\n" not in data:
            data += await reader.read(65536)
        head, _, body = data.partition(b"\r\n\r\n")
        length = 0
        for line in head.split(b"\r\n"):
            if line.lower().startswith(b"content-length:"):
                length = int(line.split(b":")[1])
        while len(body) < length:
            body += await reader.read(65536)
        received.append(_json.loads(body.decode()))
        writer.write(b"HTTP/1.1 200 OK\r\ncontent-length: 2\r\n\r\nok")
        await writer.drain()
        writer.close()

    server = await asyncio.start_server(handle, "127.0.0.1", 0)
    port = server.sockets[0].getsockname()[1]

    try:
        async with cluster(n_workers=1):
            import handlers

            fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
            import beta9_amd as b9

            client = b9.Client()
            # recreate stub with callback
            cfg = fn._stub_config()
            cfg.callback_url = f"http://127.0.0.1:{port}/cb"
            r = await asyncio.to_thread(
                client.call,
                "GatewayService",
                "GetOrCreateStub",
                {"name": "cbfn", "stub_type": "function", "config": cfg.model_dump()},
            )
            fn._stub_id = r["stub_id"]
            result = await asyncio.to_thread(fn.remote, 5)
            assert result == {"hello": 10}
            for _ in range(100):
                if received:
                    break
                await asyncio.sleep(0.05)
            assert received and received[0]["status"] == "COMPLETE"
    finally:
        server.close()
        await server.wait_closed()


@pytest.mark.asyncio
async def test_image_env_injected_into_container():
    async with cluster(n_workers=1) as c:
        img = beta9_amd.Image(
            commands=["mkdir -p $IMAGE_SITE && echo 'IMG_MARKER = 42' > $IMAGE_SITE/imgmod.py"]
        )
        image_id = await asyncio.to_thread(img.build)

        import handlers  # noqa

        fn = beta9_amd.Function.__new__(beta9_amd.Function)
        beta9_amd.Function.__init__(fn, None, cpu=100, memory=128, image=img)

        def handler():
            import imgmod  # only importable via the image site dir

            return imgmod.IMG_MARKER

        # use a spec-able handler from fixtures instead
        from beta9_amd.types import RuntimeConfig, StubConfig

        cfg = StubConfig(
            runtime=RuntimeConfig(cpu=100, memory_mb=128, image_id=image_id),
            handler="handlers_image:use_image",
        )
        ws = c.creds["workspace_id"]
        stub = c.gateway.db.get_or_create_stub(ws, "imgtest", "function", cfg)
        task = await c.gateway.dispatcher.send_and_execute("function", ws, stub["id"], None)
        done = await c.gateway.dispatcher.wait(task.task_id, timeout=60)
        result = await c.gateway.dispatcher.get_result(task.task_id)
        assert done.status.value == "COMPLETE", str(result)
        assert result == 42


@pytest.mark.asyncio
async def test_log_streaming():
    async with cluster(n_workers=1) as c:
        from beta9_amd.types import RuntimeConfig, StubConfig

        cfg = StubConfig(runtime=RuntimeConfig(cpu=100, memory_mb=64), handler="handlers:hello")
        ws = c.creds["workspace_id"]
        stub = c.gateway.db.get_or_create_stub(ws, "logs", "function", cfg)
        from beta9_amd.types import ContainerRequest

        req = ContainerRequest(
            container_id="logstream-1",
            stub_id=stub["id"],
            workspace_id=ws,
            stub_type="function",
            cpu=100,
            memory_mb=64,
            entry_point=["sh", "-c", "echo line-A; sleep 0.3; echo line-B"],
        )
        await c.gateway.scheduler.run(req)
        import beta9_amd as b9

        client = b9.Client()
        lines = []
        exit_code = None

        def consume():
            nonlocal exit_code
            for msg in client.stream(
                "GatewayService", "StreamContainerLogs", {"container_id": "logstream-1", "timeout": 30}
            ):
                if "line" in msg:
                    lines.append(msg["line"])
                if "exit_code" in msg:
                    exit_code = msg["exit_code"]

        await asyncio.to_thread(consume)
        assert "line-A" in lines and "line-B" in lines
        assert exit_code == 0


@pytest.mark.asyncio
async def test_sandbox_create_image_from_filesystem():
    """Live sandbox fs -> new image -> new container sees the files tree
    (reference: sandbox.py:527-549 create_image_from_filesystem)."""
    async with cluster(n_workers=1) as c:
        sb = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=128).create)
        try:
            await asyncio.to_thread(sb.fs.upload, "data/hello.txt", b"snapshot-payload")
            image_id = await asyncio.to_thread(sb.create_image_from_filesystem)
            assert image_id.startswith("snap-")
            # image registered with a files tree containing the sandbox file
            import os

            files = os.path.join(c.gateway.cfg.image_service.store_path, image_id, "files")
            found = []
            for dirpath, _, fns in os.walk(files):
                found += [os.path.join(dirpath, f) for f in fns]
            assert any(f.endswith("data/hello.txt") for f in found), found
            # a function using the snapshot image gets B9_IMAGE_FILES
            import handlers

            fn = beta9_amd.Function(
                handlers.image_files_env, cpu=100, memory=128,
                image=beta9_amd.Image.from_snapshot(image_id),
            )
            out = await asyncio.to_thread(fn.remote)
            assert out and image_id in out
        finally:
            await asyncio.to_thread(sb.terminate)


@pytest.mark.asyncio
async def test_sandbox_snapshot_memory_reports_unsupported():
    async with cluster(n_workers=1) as c:
        sb = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=128).create)
        try:
            r = await asyncio.to_thread(sb.snapshot_memory)
            assert r["supported"] is False and "hipsnap" in r["reason"]
        finally:
            await asyncio.to_thread(sb.terminate)


def test_sdk_fractional_gpu_spec():
    """gpu="mi355x:0.5" -> shared-device request (Thunder-analog syntax;
    fraction flows stub config -> ContainerRequest -> worker)."""
    import pytest as _pytest

    import beta9_amd

    fn = beta9_amd.function(gpu="MI355X:0.5")(lambda: 1)
    cfg = fn._stub_config()
    assert cfg.runtime.gpu == "MI355X"
    assert cfg.runtime.gpu_fraction == 0.5
    assert cfg.runtime.gpu_count == 1
    with _pytest.raises(ValueError):
        beta9_amd.function(gpu="MI355X:1.5")(lambda: 1)
    with _pytest.raises(ValueError):
        beta9_amd.function(gpu="MI355X:0.5", gpu_count=2)(lambda: 1)
    # whole-GPU spec unchanged
    assert beta9_amd.function(gpu="MI355X")(lambda: 1)._stub_config().runtime.gpu_fraction == 1.0
