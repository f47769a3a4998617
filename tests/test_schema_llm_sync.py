"""Typed IO schemas, LLM token-pressure autoscaling, serve-mode hot reload."""

import asyncio

import pytest
from pydantic import BaseModel

import beta9_amd
from beta9_amd.schema import Schema
from tests.helpers import cluster


class InferIn(BaseModel):
    tokens: int
    prompt: str = "hi"


def test_schema_spec_and_validate():
    s = Schema.from_pydantic(InferIn)
    spec = s.to_spec()
    assert spec["kind"] == "pydantic"
    Schema.validate_payload(spec, {"tokens": 5})
    with pytest.raises(ValueError):
        Schema.validate_payload(spec, {"prompt": "x"})  # missing tokens
    with pytest.raises(ValueError):
        Schema.validate_payload(spec, {"tokens": "five"})  # wrong type


@pytest.mark.asyncio
async def test_endpoint_input_validation_rejects():
    async with cluster(n_workers=1):
        import handlers

        ep = beta9_amd.Endpoint(
            handlers.echo, cpu=100, memory=256, inputs=Schema.from_pydantic(InferIn)
        )
        # valid payload passes
        r = await asyncio.to_thread(ep.remote, tokens=3)
        assert r == {"tokens": 3}
        # invalid payload -> 422 from the runner, surfaced as failure
        stub_id = ep._stub_id
        task = await cluster_invoke(ep, {"bad": 1})
        assert task["status_code"] == 422


async def cluster_invoke(ep, payload):
    # go through the executor path to read the raw status code
    import beta9_amd as b9

    client = b9.Client()
    r = await asyncio.to_thread(
        client.call,
        "EndpointService",
        "EndpointInvoke",
        {"stub_id": ep._stub_id, "args": payload, "timeout": 60.0},
        70.0,
    )
    return r["result"]


@pytest.mark.asyncio
async def test_llm_pressure_autoscaler_scales():
    async with cluster(n_workers=1) as c:
        import handlers

        ep = beta9_amd.Endpoint(
            handlers.echo,
            cpu=100,
            memory=256,
            keep_warm_seconds=30,
            autoscaler=beta9_amd.sdk.LLMTokenPressureAutoscaler(
                max_containers=3, pressure_per_container=1
            ),
        )
        await asyncio.to_thread(ep.remote, x=1)  # one container up
        stub_id = ep._stub_id
        cids = list((await c.gateway.container_repo.containers_for_stub(stub_id)).keys())
        assert len(cids) == 1
        # container reports pressure 3 -> autoscaler should go to 3
        await c.gateway.store.set(f"llm:pressure:{stub_id}:{cids[0]}", 3.0, ttl=15.0)
        inst = c.gateway._instances[stub_id]
        inst.instance.autoscaler.trigger()
        for _ in range(100):
            n = len(await c.gateway.container_repo.containers_for_stub(stub_id))
            if n >= 3:
                break
            await asyncio.sleep(0.1)
        assert n >= 3


@pytest.mark.asyncio
async def test_serve_hot_reload(tmp_path):
    async with cluster(n_workers=1) as c:
        # deploy a handler from a synced workspace dir, then hot-swap it
        ws_dir = tmp_path / "app"
        ws_dir.mkdir()
        (ws_dir / "liveapp.py").write_text("def handler(**kw):\n    return 'v1'\n")
        from beta9_amd.types import RuntimeConfig, StubConfig

        cfg = StubConfig(
            runtime=RuntimeConfig(cpu=100, memory_mb=256),
            handler="liveapp:handler",
            keep_warm_seconds=30,
            env={"B9_WORKSPACE_DIR": str(ws_dir)},
        )
        ws = c.creds["workspace_id"]
        stub = c.gateway.db.get_or_create_stub(ws, "live", "endpoint/serve", cfg)
        import beta9_amd as b9

        client = b9.Client()
        # serve mode: the serve lock keeps exactly one container up
        await asyncio.to_thread(
            client.call, "EndpointService", "StartEndpointServe",
            {"stub_id": stub["id"], "timeout": 120.0},
        )
        r1 = await asyncio.to_thread(
            client.call, "EndpointService", "EndpointInvoke",
            {"stub_id": stub["id"], "args": {}, "timeout": 60.0}, 70.0,
        )
        assert r1["result"]["body"]["result"] == "v1"
        # push v2
        sync = await asyncio.to_thread(
            client.call,
            "GatewayService",
            "SyncServeWorkspace",
            {
                "stub_id": stub["id"],
                "files": [{"path": "liveapp.py", "data": b"def handler(**kw):\n    return 'v2'\n"}],
            },
        )
        assert any(v.get("reloaded") for v in sync.values()), sync
        r2 = await asyncio.to_thread(
            client.call, "EndpointService", "EndpointInvoke",
            {"stub_id": stub["id"], "args": {}, "timeout": 60.0}, 70.0,
        )
        assert r2["result"]["body"]["result"] == "v2"
