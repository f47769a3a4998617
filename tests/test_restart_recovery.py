"""Gateway restart recovery: a new gateway over the same sqlite backend
rehydrates deployments and serves traffic again (reference:
instance.go:444-640 InstanceController boot rehydration)."""

import asyncio
import os

import pytest

import beta9_amd
from beta9_amd.common.config import AppConfig
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_gateway_restart_rehydrates_and_serves(tmp_path):
    from tests.helpers import Cluster

    db_path = str(tmp_path / "backend.db")

    async with cluster(n_workers=1, db_path=db_path) as c:
        import handlers

        ep = beta9_amd.Endpoint(handlers.hello, cpu=100, memory=128)
        d = await asyncio.to_thread(ep.deploy, "survivor")
        assert d["version"] == 1

    # new gateway process-equivalent over the same backend DB
    async with cluster(n_workers=1, db_path=db_path) as c2:
        # the deployment row survived and traffic routes to it by name
        import httpx

        base = f"http://127.0.0.1:{c2.gateway.http_port}"
        async with httpx.AsyncClient() as hc:
            r = await hc.post(
                f"{base}/deployment/survivor/",
                headers={"authorization": f"Bearer {c2.token}"},
                json={"x": 5},
            )
            assert r.status_code == 200, r.text
            body = r.json();        assert body.get("hello") == 10 or body.get("result", {}).get("hello") == 10
