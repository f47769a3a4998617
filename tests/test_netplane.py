"""Per-container network namespaces (VERDICT round-1 item #5): netns slot
pool + veth/bridge + IP allocation + userspace exposed-port proxies.

Reference parity: pkg/worker/network.go:1414-1520 Setup, :542-624 slot
pool, container_port_proxy.go:95-174 userspace proxy fallback (this image
ships no iptables, so the proxy path IS the DNAT path)."""

import asyncio

import pytest

from beta9_amd.worker.netplane import NetworkPlane, netplane_available
from beta9_amd.worker.runtime.native import namespaces_available

from .helpers import cluster

needs_net = pytest.mark.skipif(
    not (namespaces_available() and netplane_available()),
    reason="no netns/netlink privileges",
)


@needs_net
@pytest.mark.asyncio
async def test_two_sandboxes_bind_same_port():
    """Two sandbox containers run concurrently, each in its own netns with
    its own IP; both PM servers bind the SAME configured port."""
    import beta9_amd

    async with cluster(n_workers=1) as c:
        w = c.workers[0]
        assert w.netplane is not None, "network plane should be active"

        sb1 = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=128).create)
        sb2 = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=128).create)
        try:
            ip1 = sb1.addr.split(":")[0]
            ip2 = sb2.addr.split(":")[0]
            assert ip1 != ip2, f"sandboxes share an IP: {sb1.addr} vs {sb2.addr}"
            # both are bridge-subnet IPs, not the host
            assert ip1.startswith("10.") and ip2.startswith("10.")
            r1 = await asyncio.to_thread(sb1.run_code, "print(40 + 2)")
            r2 = await asyncio.to_thread(sb2.run_code, "print(50 + 5)")
            assert "42" in r1["stdout"]
            assert "55" in r2["stdout"]
        finally:
            await asyncio.to_thread(sb1.terminate)
            await asyncio.to_thread(sb2.terminate)


@needs_net
@pytest.mark.asyncio
async def test_slot_pool_hit_span_and_release():
    import beta9_amd

    async with cluster(n_workers=1) as c:
        w = c.workers[0]
        # wait for prealloc to land
        deadline = asyncio.get_event_loop().time() + 30
        while not w.netplane._pool and asyncio.get_event_loop().time() < deadline:
            await asyncio.sleep(0.1)
        assert w.netplane._pool, "slot pool should prealloc"

        sb = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=128).create)
        try:
            spans = await c.gateway.container_repo.get_spans(sb.container_id)
            names = {s["name"] for s in spans}
            assert "worker.network_setup" in names
            assert "worker.network_slot_hit" in names, names
        finally:
            await asyncio.to_thread(sb.terminate)
        # slot returned: the IP is eventually reusable
        await asyncio.sleep(1.0)
        assert len(w.netplane._active) == 0


@needs_net
@pytest.mark.asyncio
async def test_exposed_port_proxy_reaches_container():
    """A pod exposing a port is reachable through the worker's userspace
    proxy (host side), not only via the bridge IP."""
    import beta9_amd

    async with cluster(n_workers=1) as c:
        import handlers  # noqa: F401

        from beta9_amd.types import RuntimeConfig, StubConfig

        cfg = StubConfig(
            runtime=RuntimeConfig(cpu=100, memory_mb=256),
            entry_point=[],
            ports=[8765],
        )
        ws = c.creds["workspace_id"]
        stub = c.gateway.db.get_or_create_stub(ws, "port-pod", "sandbox", cfg)
        import beta9_amd as b9

        client = b9.Client()
        r = await asyncio.to_thread(
            client.call, "PodService", "CreatePod", {"stub_id": stub["id"], "timeout": 60.0}
        )
        cid = r["container_id"]
        try:
            assert r["ok"], str(r)
            w = c.workers[0]
            inst = w.lifecycle.containers.get(cid)
            assert inst is not None and inst.exposed_ports
            hport, cport = inst.exposed_ports[0]
            assert cport == 8765
            # run a server on 8765 inside the sandbox via the PM, then hit
            # the host-side proxy port
            # serve exactly one connection in the foreground (run_code owns
            # the process lifetime), while we connect through the proxy
            code = (
                "import socket\n"
                "s=socket.socket(); s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)\n"
                "s.bind(('0.0.0.0',8765)); s.listen(1); s.settimeout(20)\n"
                "c,_=s.accept(); c.sendall(b'pong'); c.close()\n"
                "print('served')\n"
            )
            run_task = asyncio.create_task(asyncio.to_thread(
                client.call, "PodService", "SandboxRunCode",
                {"container_id": cid, "code": code, "timeout": 30.0}, 40.0,
            ))
            data = b""
            deadline = asyncio.get_event_loop().time() + 15
            while asyncio.get_event_loop().time() < deadline:
                try:
                    reader, writer = await asyncio.open_connection("127.0.0.1", hport)
                    data = await asyncio.wait_for(reader.read(100), timeout=5)
                    writer.close()
                    if data:
                        break
                except OSError:
                    pass
                await asyncio.sleep(0.3)
            rr = await run_task
            assert data == b"pong", (data, rr)
            assert "served" in rr.get("stdout", "")
        finally:
            await asyncio.to_thread(client.call, "PodService", "StopPod", {"container_id": cid})


@needs_net
@pytest.mark.asyncio
async def test_nested_container_in_sandbox():
    """Docker-in-sandbox analog: a sandbox launches a NESTED isolated
    container (namespaces-in-namespaces via b9launch) whose writes and
    hostname don't leak into the sandbox."""
    import beta9_amd

    async with cluster(n_workers=1) as c:
        sb = await asyncio.to_thread(beta9_amd.Sandbox(cpu=200, memory=256).create)
        try:
            r = await asyncio.to_thread(
                sb.run_container,
                "hostname; echo nested-says-$((6*7)); touch /nested-canary",
            )
            assert r.get("exit_code") == 0, r
            assert "nested" in r["stdout"] and "nested-says-42" in r["stdout"]
            # the nested container's write stayed in ITS overlay
            chk = await asyncio.to_thread(sb.run_code, "import os; print(os.path.exists('/nested-canary'))")
            assert "False" in chk["stdout"]
            # block-network nested container has no interfaces beyond lo
            r2 = await asyncio.to_thread(
                sb.run_container, "ls /sys/class/net", block_network=True,
            )
            assert r2.get("exit_code") == 0, r2
            assert r2["stdout"].split() == ["lo"], r2["stdout"]
        finally:
            await asyncio.to_thread(sb.terminate)
