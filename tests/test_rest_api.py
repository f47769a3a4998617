"""REST /api/v1 surface + prometheus endpoint."""

import asyncio

import httpx
import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_rest_surface():
    async with cluster(n_workers=1) as c:
        base = f"http://127.0.0.1:{c.gateway.http_port}"
        headers = {"authorization": f"Bearer {c.token}"}
        async with httpx.AsyncClient() as hc:
            r = await hc.get(f"{base}/api/v1/health")
            assert r.status_code == 200 and r.json()["status"] == "ok"

            # unauthorized requests rejected
            r = await hc.get(f"{base}/api/v1/workers")
            assert r.status_code == 401

            r = await hc.get(f"{base}/api/v1/workers", headers=headers)
            assert r.status_code == 200 and len(r.json()) == 1

            # run a function so tasks/containers/events/metrics have content
            import handlers

            fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
            await asyncio.to_thread(fn.remote, 2)

            r = await hc.get(f"{base}/api/v1/tasks", headers=headers)
            assert r.status_code == 200
            c.gateway.db.flush()
            r = await hc.get(f"{base}/api/v1/tasks", headers=headers)
            assert len(r.json()) >= 1

            r = await hc.get(f"{base}/api/v1/metrics", headers=headers)
            body = r.json()
            assert "scheduler" in body and body["scheduler"]["scheduled"] >= 1

            r = await hc.get(f"{base}/metrics")
            assert r.status_code == 200
            assert "scheduler_containers_scheduled" in r.text

            r = await hc.get(f"{base}/api/v1/events", headers=headers)
            names = {e["name"] for e in r.json()}
            assert "container.scheduled" in names
            assert "container.started" in names

            r = await hc.get(f"{base}/api/v1/deployments", headers=headers)
            assert r.status_code == 200


@pytest.mark.asyncio
async def test_host_style_subdomain_routing():
    """{name}-v{N}.app.{base} and {name}-latest.app.{base} hosts resolve to
    deployment invoke routes (reference: subdomain.go:19-36)."""
    async with cluster(n_workers=1) as c:
        import handlers
        import httpx

        fn = beta9_amd.Endpoint(handlers.hello, cpu=100, memory=128)
        d = await asyncio.to_thread(fn.deploy, "subdom")
        base = f"http://127.0.0.1:{c.gateway.http_port}"
        tok = c.token
        async with httpx.AsyncClient() as client:
            for host in ("subdom-latest.app.b9.local", f"subdom-v{d['version']}.app.b9.local"):
                r = await client.post(
                    base + "/",
                    headers={"host": host, "authorization": f"Bearer {tok}"},
                    json={"x": 3},
                )
                assert r.status_code == 200, (host, r.text)
                assert r.json().get("hello") == 6 or r.json().get("result", {}).get("hello") == 6, r.text


@pytest.mark.asyncio
async def test_public_stub_invokable_without_token():
    """authorized=False endpoints accept anonymous invokes, accounted to
    the stub's workspace; authorized stubs still 401."""
    async with cluster(n_workers=1) as c:
        import handlers

        pub = beta9_amd.Endpoint(handlers.hello, cpu=100, memory=128, authorized=False)
        pub_id = await asyncio.to_thread(pub.prepare_runtime)
        priv = beta9_amd.Endpoint(handlers.hello, cpu=100, memory=128, name="priv")
        priv_id = await asyncio.to_thread(priv.prepare_runtime)
        base = f"http://127.0.0.1:{c.gateway.http_port}"
        async with httpx.AsyncClient() as hc:
            r = await hc.post(f"{base}/endpoint/id/{pub_id}", json={"x": 4}, timeout=60.0)
            assert r.status_code == 200, r.text
            body = r.json()
            assert body.get("hello") == 8 or body.get("result", {}).get("hello") == 8
            r = await hc.post(f"{base}/endpoint/id/{priv_id}", json={"x": 4})
            assert r.status_code == 401


@pytest.mark.asyncio
async def test_signed_public_payloads():
    """SignPayload RPC + require_signature public stubs: a signed request
    passes, unsigned/garbage is 401."""
    import json as _json

    async with cluster(n_workers=1) as c:
        import handlers

        ep = beta9_amd.Endpoint(handlers.hello, cpu=100, memory=128, authorized=False)
        stub_id = await asyncio.to_thread(ep.prepare_runtime)
        cfg = c.gateway.db.get_stub_config(stub_id)
        cfg.extra["require_signature"] = True
        c.gateway.db.update_stub_config(stub_id, cfg) if hasattr(c.gateway.db, "update_stub_config") else None
        # fall back: mutate the row directly if no update helper
        if not hasattr(c.gateway.db, "update_stub_config"):
            c.gateway.db._exec(
                "UPDATE stubs SET config=? WHERE id=?",
                (cfg.model_dump_json(), stub_id),
            )
        # bust caches
        c.gateway.dispatcher._cfg_cache = {}

        body = _json.dumps({"x": 7}).encode()
        client = beta9_amd.Client()
        signed = await asyncio.to_thread(
            client.call, "GatewayService", "SignPayload",
            {"payload": body, "method": "POST", "stub_id": stub_id},
        )
        base = f"http://127.0.0.1:{c.gateway.http_port}"
        async with httpx.AsyncClient() as hc:
            r = await hc.post(f"{base}/endpoint/id/{stub_id}", content=body,
                              headers={"x-b9-signature": signed["signature"],
                                       "content-type": "application/json"}, timeout=60.0)
            assert r.status_code == 200, r.text
            r = await hc.post(f"{base}/endpoint/id/{stub_id}", content=body,
                              headers={"content-type": "application/json"})
            assert r.status_code == 401
            r = await hc.post(f"{base}/endpoint/id/{stub_id}", content=body,
                              headers={"x-b9-signature": "v1=1:deadbeef",
                                       "content-type": "application/json"})
            assert r.status_code == 401
