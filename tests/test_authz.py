"""Cross-tenant authorization: token types + workspace-scoped access.

Reference behavior: cross-workspace token ops are ClusterAdmin-only
(pkg/auth/http.go:106) and container/task RPCs are scoped to the
authenticated workspace. ADVICE.md round-1 high-severity findings.
"""

import asyncio

import httpx
import pytest

import beta9_amd
from beta9_amd.rpc.channel import RpcClient

from .helpers import cluster


def _client(addr, token):
    from beta9_amd.sdk import ClientConfig

    return beta9_amd.Client(ClientConfig(gateway_addr=addr, token=token))


@pytest.mark.asyncio
async def test_token_type_gating():
    async with cluster(n_workers=1) as c:
        admin = beta9_amd.Client()  # configured with the bootstrap (admin) token

        # admin can create a second workspace + tenant token
        ws_b = await asyncio.to_thread(
            admin.call, "GatewayService", "CreateWorkspace", {"name": "tenant-b"}
        )
        tok_b = await asyncio.to_thread(
            admin.call, "GatewayService", "CreateToken", {"workspace_id": ws_b["workspace_id"]}
        )
        tenant = _client(c.grpc_addr, tok_b["token"])

        # tenant cannot create workspaces
        with pytest.raises(Exception):
            await asyncio.to_thread(
                tenant.call, "GatewayService", "CreateWorkspace", {"name": "tenant-c"}
            )

        # tenant cannot mint tokens for another workspace
        ws_a = c.creds["workspace_id"]
        with pytest.raises(Exception):
            await asyncio.to_thread(
                tenant.call, "GatewayService", "CreateToken", {"workspace_id": ws_a}
            )

        # tenant CAN mint a token for its own workspace
        own = await asyncio.to_thread(tenant.call, "GatewayService", "CreateToken", {})
        assert own["workspace_id"] == ws_b["workspace_id"]

        # tenant cannot revoke the admin's token
        with pytest.raises(Exception):
            await asyncio.to_thread(
                tenant.call, "GatewayService", "RevokeToken", {"token": c.token}
            )
        # admin token still works
        apps = await asyncio.to_thread(admin.call, "GatewayService", "ListApps", {})
        assert isinstance(apps, list)

        # tenant can revoke its own token
        ok = await asyncio.to_thread(
            tenant.call, "GatewayService", "RevokeToken", {"token": own["token"]}
        )
        assert ok is True


@pytest.mark.asyncio
async def test_cross_tenant_container_and_task_isolation():
    async with cluster(n_workers=1) as c:
        import handlers

        admin = beta9_amd.Client()
        # run a function in workspace A (bootstrap workspace) so a task +
        # container exist there
        fn = beta9_amd.Function(handlers.add, cpu=100, memory=128)
        result = await asyncio.to_thread(fn.remote, 2, 3)
        assert result == 5

        tasks_a = await asyncio.to_thread(admin.call, "GatewayService", "ListTasks", {})
        assert tasks_a, "workspace A should have at least one task"
        task_id = tasks_a[0]["id"]
        container_id = tasks_a[0].get("container_id", "")

        # make tenant B
        ws_b = await asyncio.to_thread(
            admin.call, "GatewayService", "CreateWorkspace", {"name": "tenant-b2"}
        )
        tok_b = await asyncio.to_thread(
            admin.call, "GatewayService", "CreateToken", {"workspace_id": ws_b["workspace_id"]}
        )
        tenant = _client(c.grpc_addr, tok_b["token"])

        # tenant B cannot read workspace A's task
        with pytest.raises(Exception):
            await asyncio.to_thread(tenant.call, "GatewayService", "GetTask", {"task_id": task_id})
        with pytest.raises(Exception):
            await asyncio.to_thread(
                tenant.call, "GatewayService", "GetTaskResult", {"task_id": task_id}
            )
        # tenant B cannot read A's container logs / stop A's container
        if container_id:
            with pytest.raises(Exception):
                await asyncio.to_thread(
                    tenant.call, "GatewayService", "GetContainerLogs",
                    {"container_id": container_id},
                )
            with pytest.raises(Exception):
                await asyncio.to_thread(
                    tenant.call, "GatewayService", "StopContainer",
                    {"container_id": container_id},
                )

        # the same over REST
        base = f"http://127.0.0.1:{c.gateway.http_port}"
        async with httpx.AsyncClient() as hc:
            hdr = {"authorization": f"Bearer {tok_b['token']}"}
            r = await hc.get(f"{base}/api/v1/tasks/{task_id}", headers=hdr)
            assert r.status_code == 404
            if container_id:
                r = await hc.get(f"{base}/api/v1/logs/{container_id}", headers=hdr)
                assert r.status_code == 404
                r = await hc.post(f"{base}/api/v1/containers/{container_id}/stop", headers=hdr)
                assert r.status_code == 404
            # admin still sees the task
            hdr_a = {"authorization": f"Bearer {c.token}"}
            r = await hc.get(f"{base}/api/v1/tasks/{task_id}", headers=hdr_a)
            assert r.status_code == 200
