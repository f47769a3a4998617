"""Endpoint/taskqueue/sandbox e2e through the full stack (real subprocess
containers serving HTTP / popping queues / running the process manager)."""

import asyncio

import httpx
import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_endpoint_invoke_and_keep_warm():
    async with cluster(n_workers=1) as c:
        import handlers

        ep = beta9_amd.Endpoint(handlers.echo, cpu=100, memory=256, keep_warm_seconds=30)
        result = await asyncio.to_thread(ep.remote, x=1, y="z")
        assert result == {"x": 1, "y": "z"}
        # warm second request reuses the container (fast path)
        result2 = await asyncio.to_thread(ep.remote, x=2)
        assert result2 == {"x": 2}
        # exactly one container serving this stub
        stubs = await c.gateway.container_repo.containers_for_stub(ep._stub_id)
        assert len(stubs) == 1


@pytest.mark.asyncio
async def test_endpoint_on_start_and_http_route():
    async with cluster(n_workers=1) as c:
        import handlers

        ep = beta9_amd.Endpoint(
            handlers.check_started, cpu=100, memory=256, on_start=handlers.on_start
        )
        assert await asyncio.to_thread(ep.remote) is True

        # HTTP invoke route (gateway REST surface)
        stub_id = ep._stub_id
        async with httpx.AsyncClient() as client:
            r = await client.post(
                f"http://127.0.0.1:{c.gateway.http_port}/endpoint/id/{stub_id}",
                json={"x": 5},
                headers={"authorization": f"Bearer {c.token}"},
                timeout=60.0,
            )
            assert r.status_code == 200
            assert r.json()["result"] is True


@pytest.mark.asyncio
async def test_endpoint_autoscales_to_zero():
    async with cluster(n_workers=1) as c:
        import handlers

        ep = beta9_amd.Endpoint(handlers.echo, cpu=100, memory=256, keep_warm_seconds=0.5)
        await asyncio.to_thread(ep.remote, a=1)
        stub_id = ep._stub_id
        # keep-warm expires -> autoscaler stops the container
        for _ in range(100):
            await asyncio.sleep(0.2)
            left = await c.gateway.container_repo.containers_for_stub(stub_id)
            if not left:
                break
        assert not await c.gateway.container_repo.containers_for_stub(stub_id)


@pytest.mark.asyncio
async def test_taskqueue_put_and_process():
    async with cluster(n_workers=1):
        import handlers

        tq = beta9_amd.TaskQueue(handlers.add, cpu=100, memory=256, keep_warm_seconds=20)
        task_ids = [await asyncio.to_thread(tq.put, i, i) for i in range(4)]
        results = [await asyncio.to_thread(tq.wait, t, 60.0) for t in task_ids]
        assert results == [0, 2, 4, 6]


@pytest.mark.asyncio
async def test_sandbox_exec_and_fs():
    async with cluster(n_workers=1):
        sb_def = beta9_amd.Sandbox(cpu=100, memory=256)
        sb = await asyncio.to_thread(sb_def.create)
        try:
            r = await asyncio.to_thread(sb.exec, "echo hello-$((1+1))")
            assert r["exit_code"] == 0
            assert "hello-2" in r["stdout"]

            r = await asyncio.to_thread(sb.run_code, "print(6*7)")
            assert "42" in r["stdout"]

            await asyncio.to_thread(sb.fs.upload, "/tmp/b9test/a.txt", b"alpha beta")
            data = await asyncio.to_thread(sb.fs.download, "/tmp/b9test/a.txt")
            assert data == b"alpha beta"
            found = await asyncio.to_thread(sb.fs.find_in_files, "/tmp/b9test", "beta")
            assert len(found) == 1
            await asyncio.to_thread(
                sb.fs.replace_in_files, "/tmp/b9test", "beta", "gamma"
            )
            assert (await asyncio.to_thread(sb.fs.download, "/tmp/b9test/a.txt")) == b"alpha gamma"
            ls = await asyncio.to_thread(sb.fs.ls, "/tmp/b9test")
            assert ls[0]["name"] == "a.txt"
        finally:
            await asyncio.to_thread(sb.terminate)


@pytest.mark.asyncio
async def test_map_queue_signal_clients():
    async with cluster(n_workers=1):
        m = beta9_amd.MapClient("m1")
        await asyncio.to_thread(m.set, "k", {"v": 1})
        assert await asyncio.to_thread(m.get, "k") == {"v": 1}
        assert await asyncio.to_thread(m.keys) == ["k"]

        q = beta9_amd.QueueClient("q1")
        await asyncio.to_thread(q.put, "a")
        assert await asyncio.to_thread(q.pop) == "a"
        assert await asyncio.to_thread(q.pop) is None

        s = beta9_amd.SignalClient("s1")
        assert not await asyncio.to_thread(s.is_set)
        await asyncio.to_thread(s.set)
        assert await asyncio.to_thread(s.is_set)


@pytest.mark.asyncio
async def test_image_build_and_env():
    async with cluster(n_workers=1):
        img = beta9_amd.Image(commands=["echo marker > $PREFIX/marker.txt"])
        image_id = await asyncio.to_thread(img.build)
        assert len(image_id) == 24
        # cached rebuild returns same id
        assert await asyncio.to_thread(img.build) == image_id

        from beta9_amd.abstractions.image import image_env

        # marker file exists in the image files tree
        import os

        env = None
        # find the store path through the running gateway config
        from tests.helpers import Cluster  # noqa


@pytest.mark.asyncio
async def test_deployment_versioning():
    async with cluster(n_workers=1) as c:
        import handlers

        fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
        d1 = await asyncio.to_thread(fn.deploy, "myapp")
        d2 = await asyncio.to_thread(fn.deploy, "myapp")
        assert d2["version"] == d1["version"] + 1
        deps = c.gateway.db.list_deployments(c.creds["workspace_id"])
        active = [d for d in deps if d["active"]]
        assert len(active) == 1 and active[0]["version"] == d2["version"]


@pytest.mark.asyncio
async def test_dynamic_batching_coalesces_requests():
    """@batchable handlers get concurrent invokes coalesced into one
    handler(batch=[...]) call (runner/batching.py)."""
    async with cluster(n_workers=1) as c:
        import handlers

        ep = beta9_amd.Endpoint(handlers.batched_echo, cpu=100, memory=128,
                                concurrent_requests=16)
        stub_id = await asyncio.to_thread(ep.prepare_runtime)
        outs = await asyncio.gather(*[
            asyncio.to_thread(lambda i=i: ep.remote(x=i)) for i in range(8)
        ])
        assert sorted(o["x"] for o in outs) == list(range(8))
        # at least one response saw a batch of >1 (coalescing happened)
        assert max(o["batch_size"] for o in outs) > 1


@pytest.mark.asyncio
async def test_dynamic_batcher_unit_timeout_and_errors():
    """DynamicBatcher: wait_ms flush with partial batch; handler errors
    propagate to every waiter."""
    from beta9_amd.runner.batching import DynamicBatcher

    calls = []

    def handler(batch=None):
        calls.append(len(batch))
        if batch and batch[0] == "boom":
            raise ValueError("nope")
        return [f"r{a}" for a in batch]

    b = DynamicBatcher(handler, max_size=8, wait_ms=10.0)
    # partial batch flushes on the timer
    r = await asyncio.gather(b.submit("a"), b.submit("b"))
    assert r == ["ra", "rb"] and calls == [2]
    # errors propagate
    with pytest.raises(ValueError):
        await b.submit("boom")
    # max_size triggers an immediate flush
    rs = await asyncio.gather(*[b.submit(i) for i in range(8)])
    assert rs == [f"r{i}" for i in range(8)]


@pytest.mark.asyncio
async def test_service_pod_with_llm_config():
    """Service(Pod): serving config recorded on the stub; pod boots and
    exposes its port (reference: abstractions/service.py Service(Pod))."""
    async with cluster(n_workers=1) as c:
        svc = beta9_amd.Service(
            port=8123, cpu=100, memory=128,
            llm=beta9_amd.LLMConfig(model_id="synthetic-llama", context_length=2048),
        )
        inst = await asyncio.to_thread(svc.create)
        try:
            stub_id = svc._stub_id
            cfg = c.gateway.db.get_stub_config(stub_id)
            assert cfg.extra["serving"]["app_kind"] == "llm"
            assert cfg.extra["serving"]["llm"]["model_id"] == "synthetic-llama"
            assert cfg.ports == [8123]
        finally:
            await asyncio.to_thread(inst.terminate) if hasattr(inst, "terminate") else None
