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
