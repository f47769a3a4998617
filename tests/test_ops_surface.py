"""Ops surface: workspaces/tokens, apps, rollout, worker cordon."""

import asyncio

import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_workspace_token_lifecycle():
    async with cluster(n_workers=0) as c:
        client = beta9_amd.Client()
        ws = await asyncio.to_thread(client.call, "GatewayService", "CreateWorkspace", {"name": "team-b"})
        tok = await asyncio.to_thread(
            client.call, "GatewayService", "CreateToken", {"workspace_id": ws["workspace_id"]}
        )
        # new token authorizes
        c2 = beta9_amd.Client(beta9_amd.ClientConfig(c.grpc_addr, tok["token"]))
        r = await asyncio.to_thread(c2.call, "GatewayService", "Authorize", {"token": tok["token"]})
        assert r["ok"] and r["workspace_id"] == ws["workspace_id"]
        # revoke -> rejected
        await asyncio.to_thread(client.call, "GatewayService", "RevokeToken", {"token": tok["token"]})
        import grpc

        with pytest.raises(grpc.RpcError):
            await asyncio.to_thread(c2.call, "GatewayService", "ListContainers", {})
        c2.close()


@pytest.mark.asyncio
async def test_rollout_switches_active_version():
    async with cluster(n_workers=1) as c:
        import handlers

        fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
        d1 = await asyncio.to_thread(fn.deploy, "app-r")
        d2 = await asyncio.to_thread(fn.deploy, "app-r")
        client = beta9_amd.Client()
        r = await asyncio.to_thread(
            client.call, "GatewayService", "RolloutDeployment", {"name": "app-r", "version": d1["version"]}
        )
        assert r["version"] == d1["version"]
        dep = c.gateway.db.get_deployment(c.creds["workspace_id"], "app-r")
        assert dep["version"] == d1["version"]

        apps = await asyncio.to_thread(client.call, "GatewayService", "ListApps", {})
        assert any(a["name"] == "hello" for a in apps) or len(apps) >= 1


@pytest.mark.asyncio
async def test_cordoned_worker_excluded():
    async with cluster(n_workers=1) as c:
        client = beta9_amd.Client()
        workers = await asyncio.to_thread(client.call, "GatewayService", "ListWorkers", {})
        wid = workers[0]["id"]
        await asyncio.to_thread(client.call, "GatewayService", "CordonWorker", {"worker_id": wid})
        live = await c.gateway.scheduler._live_workers()
        assert all(w.id != wid for w in live)
        await asyncio.to_thread(
            client.call, "GatewayService", "CordonWorker", {"worker_id": wid, "cordoned": False}
        )
        live = await c.gateway.scheduler._live_workers()
        assert any(w.id == wid for w in live)


@pytest.mark.asyncio
async def test_output_sdk_roundtrip():
    async with cluster(n_workers=0) as c:
        out = beta9_amd.Output(task_id="task-x")
        r = await asyncio.to_thread(out.save, b"result-bytes", "result.bin")
        assert r["path"].endswith("result.bin")
        st = await asyncio.to_thread(out.stat, r["path"])
        assert st["exists"] and st["size"] == len(b"result-bytes")
        url = await asyncio.to_thread(out.public_url, r["path"])
        assert url.startswith("/api/v1/outputs/")
        import httpx

        async with httpx.AsyncClient() as hc:
            resp = await hc.get(
                f"http://127.0.0.1:{c.gateway.http_port}{url}",
                headers={"authorization": f"Bearer {c.token}"},
            )
            assert resp.status_code == 200 and resp.content == b"result-bytes"


@pytest.mark.asyncio
async def test_gateway_drain():
    async with cluster(n_workers=1) as c:
        import httpx

        c.gateway.draining = True
        async with httpx.AsyncClient() as hc:
            r = await hc.post(
                f"http://127.0.0.1:{c.gateway.http_port}/endpoint/id/whatever",
                json={},
                headers={"authorization": f"Bearer {c.token}"},
            )
            assert r.status_code == 503
        c.gateway.draining = False
        await c.gateway.drain(timeout=1.0)
        assert c.gateway.draining is True


@pytest.mark.asyncio
async def test_drain_worker():
    async with cluster(n_workers=1) as c:
        client = beta9_amd.Client()
        workers = await asyncio.to_thread(client.call, "GatewayService", "ListWorkers", {})
        wid = workers[0]["id"]
        r = await asyncio.to_thread(
            client.call, "GatewayService", "DrainWorker", {"worker_id": wid, "timeout": 5.0}
        )
        assert r["drained"] is True
        w = await c.gateway.worker_repo.get(wid)
        assert w.cordoned is True


@pytest.mark.asyncio
async def test_pricing_policy_meters_usage():
    """PricingPolicy on a stub meters task count + cost into workspace usage
    (reference: pkg/abstractions/common/usage.go TrackTaskCost)."""
    async with cluster(n_workers=1) as c:
        import handlers

        fn = beta9_amd.Function(
            handlers.hello, cpu=100, memory=128,
            pricing=beta9_amd.PricingPolicy(cost_model="task", cost_per_task=0.02),
        )
        for _ in range(3):
            assert await asyncio.to_thread(fn.remote, 2) == {"hello": 4}
        usage_hashes = {k: v for k, v in c.gateway.store._hashes.items() if k.startswith("usage:")}
        assert usage_hashes, "no usage counters recorded"
        ws = next(iter(usage_hashes.values()))
        assert int(ws.get("task_count", 0)) >= 3
        assert abs(float(ws.get("cost_cents", 0)) - 6.0) < 1e-6  # 3 * $0.02 = 6 cents
        assert float(ws.get("task_duration_ms", 0)) > 0
