"""Minimum end-to-end slice: @function hello-world through the full stack
(SDK -> gateway gRPC -> dispatcher -> scheduler backlog -> worker -> runner
subprocess -> result). BASELINE.json config 1."""

import asyncio

import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_function_remote_roundtrip():
    async with cluster(n_workers=1):
        import handlers  # from tests/fixtures (PYTHONPATH)

        fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)

        # .remote() blocks; run it in a thread so our loop keeps serving
        result = await asyncio.to_thread(fn.remote, 21)
        assert result == {"hello": 42}


@pytest.mark.asyncio
async def test_function_with_args_kwargs_and_error():
    async with cluster(n_workers=1):
        import handlers

        add = beta9_amd.Function(handlers.add, cpu=100, memory=128)
        assert await asyncio.to_thread(add.remote, 2, 3) == 5

        boom = beta9_amd.Function(handlers.boom, cpu=100, memory=128)
        with pytest.raises(RuntimeError):
            await asyncio.to_thread(boom.remote)


@pytest.mark.asyncio
async def test_function_map_parallel():
    async with cluster(n_workers=1):
        import handlers

        add1 = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
        results = await asyncio.to_thread(add1.map, [1, 2, 3])
        assert results == [{"hello": 2}, {"hello": 4}, {"hello": 6}]


@pytest.mark.asyncio
async def test_secret_injection():
    async with cluster(n_workers=1) as c:
        import handlers

        client = beta9_amd.Client()
        await asyncio.to_thread(
            client.call, "SecretService", "SecretCreate", {"name": "TEST_SECRET", "value": "s3cr3t"}
        )
        fn = beta9_amd.Function(handlers.envdump, cpu=100, memory=128, secrets=["TEST_SECRET"])
        result = await asyncio.to_thread(fn.remote)
        assert result["secret"] == "s3cr3t"
        assert result["stub_type"] == "function"
        assert result["container_id"].startswith("function-")


@pytest.mark.asyncio
async def test_fractional_gpu_sharing_e2e():
    """Two half-GPU functions run CONCURRENTLY on a 1-GPU worker, packed
    onto the same device with the HBM-budget env set; a whole-GPU request
    gets 'no capacity' while the shares hold the device (Thunder-analog
    sharing, pkg/worker/thunder.go — fraction flows SDK -> scheduler ->
    worker -> runner env)."""
    async with cluster(n_workers=1, fake_gpus=1):
        import handlers

        fn = beta9_amd.Function(
            handlers.gpu_envdump, cpu=100, memory=128, gpu="MI355X:0.5"
        )
        r1, r2 = await asyncio.gather(
            asyncio.to_thread(fn.remote), asyncio.to_thread(fn.remote)
        )
        assert r1["fraction"] == "0.5" and r2["fraction"] == "0.5"
        # both shares landed on the single device
        assert r1["visible"] == r2["visible"] != ""
