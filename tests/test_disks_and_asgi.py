"""Durable disks + true-ASGI apps + shell exec surface."""

import asyncio
import os

import httpx
import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_disk_create_attach_snapshot_delete():
    async with cluster(n_workers=0) as c:
        client = beta9_amd.Client()
        info = await asyncio.to_thread(
            client.call, "DiskService", "CreateDisk", {"name": "d1", "size_gb": 1}
        )
        assert info["backend"] in ("loop-ext4", "dir")
        disks = await asyncio.to_thread(client.call, "DiskService", "ListDisks", {})
        assert disks[0]["name"] == "d1"
        att = await asyncio.to_thread(client.call, "DiskService", "AttachDisk", {"name": "d1"})
        # write into the attached disk
        with open(os.path.join(att["path"], "hello.txt"), "w") as f:
            f.write("persist")
        snap = await asyncio.to_thread(
            client.call, "DiskService", "SnapshotDisk", {"name": "d1", "snapshot_name": "s1"}
        )
        assert os.path.exists(snap["path"])
        await asyncio.to_thread(client.call, "DiskService", "DeleteDisk", {"name": "d1"})
        assert await asyncio.to_thread(client.call, "DiskService", "ListDisks", {}) == []


@pytest.mark.asyncio
async def test_asgi_user_app():
    async with cluster(n_workers=1) as c:
        import handlers_asgi  # fixture

        app_def = beta9_amd.asgi(cpu=100, memory=256)(handlers_asgi.make_app)
        stub_id = await asyncio.to_thread(app_def.prepare_runtime)
        # invoke through the gateway HTTP asgi route
        async with httpx.AsyncClient() as client:
            r = await client.post(
                f"http://127.0.0.1:{c.gateway.http_port}/asgi/id/{stub_id}/double",
                json={"x": 21},
                headers={"authorization": f"Bearer {c.token}"},
                timeout=60.0,
            )
            assert r.status_code == 200, r.text
            assert r.json()["doubled"] == 42


@pytest.mark.asyncio
async def test_realtime_websocket_through_ingress():
    """@realtime: a WebSocket ASGI stub is reachable end-to-end — direct to
    the container AND through the protocol-agnostic TCP ingress (reference:
    proxyWebsocketConnection buffer.go:958; our ingress splices bytes, so
    WS rides through verbatim). Uses the first-party RFC6455 client."""
    import beta9_amd
    from beta9_amd.common.ws import WSClient
    from beta9_amd.gateway.tcp_ingress import TcpForwarder

    from .helpers import cluster

    async with cluster(n_workers=1) as c:
        import handlers_asgi  # fixture

        app_def = beta9_amd.realtime(cpu=200, memory=512)(handlers_asgi.make_ws_app)
        stub_id = await asyncio.to_thread(app_def.prepare_runtime)
        # force a container up via one HTTP invoke (scale 0 -> 1)
        async with httpx.AsyncClient() as hc:
            await hc.post(
                f"http://127.0.0.1:{c.gateway.http_port}/asgi/id/{stub_id}/",
                json={},
                headers={"authorization": f"Bearer {c.token}"},
                timeout=60.0,
            )
        addrs = await c.gateway.container_repo.addresses_for_stub(stub_id)
        assert addrs, "no asgi container up"
        host, _, port = list(addrs.values())[0].partition(":")

        # direct container websocket
        ws = await WSClient.connect(host, int(port), "/ws")
        await ws.send("hello")
        op, payload = await ws.recv()
        assert payload == b"echo:hello", payload
        await ws.close()

        # through the TCP ingress (the realtime routing path)
        fwd = TcpForwarder(host, int(port))
        ingress_port = await fwd.start()
        try:
            ws2 = await WSClient.connect("127.0.0.1", ingress_port, "/ws")
            await ws2.send("via-ingress")
            op, payload = await ws2.recv()
            assert payload == b"echo:via-ingress", payload
            await ws2.close()
        finally:
            await fwd.stop()
