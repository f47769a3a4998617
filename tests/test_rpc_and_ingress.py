"""RPC layer unit tests (auth, streaming, errors) + TCP ingress + function
log streaming."""

import asyncio

import pytest

import beta9_amd
from beta9_amd.rpc.channel import RpcClient, RpcError, RpcServer, not_found, rpc
from tests.helpers import cluster


class EchoService:
    service_name = "Echo"

    @rpc()
    async def Ping(self, req, context, auth):
        return {"pong": req, "auth": auth}

    @rpc(open=True)
    async def Open(self, req, context, auth):
        return "open"

    @rpc()
    async def Boom(self, req, context, auth):
        raise not_found("nope")

    @rpc(kind="unary_stream")
    async def Count(self, req, context, auth):
        for i in range(int(req["n"])):
            yield {"i": i}

    @rpc(kind="stream_unary")
    async def Sum(self, request_iter, context, auth):
        total = 0
        async for msg in request_iter:
            total += msg["v"]
        return total


@pytest.mark.asyncio
async def test_rpc_auth_and_kinds():
    async def auth_fn(token, method):
        if token == "good":
            return {"workspace_id": "ws1"}
        return None

    server = RpcServer(auth_fn=auth_fn)
    server.register_service(EchoService())
    port = await server.start()
    try:
        ok = RpcClient(f"127.0.0.1:{port}", token="good")
        r = await ok.call("Echo", "Ping", {"x": 1})
        assert r["pong"] == {"x": 1} and r["auth"]["workspace_id"] == "ws1"

        # streaming
        items = [m async for m in ok.stream("Echo", "Count", {"n": 3})]
        assert [m["i"] for m in items] == [0, 1, 2]

        async def gen():
            for v in (1, 2, 3):
                yield {"v": v}

        assert await ok.client_stream("Echo", "Sum", gen()) == 6

        # typed errors propagate as grpc status
        import grpc

        with pytest.raises(grpc.aio.AioRpcError) as ei:
            await ok.call("Echo", "Boom", {})
        assert ei.value.code() == grpc.StatusCode.NOT_FOUND
        await ok.close()

        # bad token rejected; open methods pass
        bad = RpcClient(f"127.0.0.1:{port}", token="bad")
        with pytest.raises(grpc.aio.AioRpcError) as ei:
            await bad.call("Echo", "Ping", {})
        assert ei.value.code() == grpc.StatusCode.UNAUTHENTICATED
        assert await bad.call("Echo", "Open", {}) == "open"
        await bad.close()
    finally:
        await server.stop()


@pytest.mark.asyncio
async def test_tcp_ingress_forwarder():
    from beta9_amd.gateway.tcp_ingress import TcpIngress

    # upstream echo server
    async def handle(reader, writer):
        data = await reader.read(100)
        writer.write(b"echo:" + data)
        await writer.drain()
        writer.close()

    upstream = await asyncio.start_server(handle, "127.0.0.1", 0)
    up_port = upstream.sockets[0].getsockname()[1]
    ingress = TcpIngress()
    try:
        port = await ingress.expose("c1", "127.0.0.1", up_port)
        r, w = await asyncio.open_connection("127.0.0.1", port)
        w.write(b"hello")
        await w.drain()
        w.write_eof()
        data = await r.read(100)
        assert data == b"echo:hello"
        w.close()
        # idempotent expose
        assert await ingress.expose("c1", "127.0.0.1", up_port) == port
    finally:
        await ingress.shutdown()
        upstream.close()


@pytest.mark.asyncio
async def test_sandbox_expose_port_e2e():
    async with cluster(n_workers=1) as c:
        sb = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=256).create)
        try:
            # run a tiny TCP server inside the sandbox
            code = (
                "import socket\n"
                "s=socket.socket(); s.bind((\"0.0.0.0\", 47654)); s.listen(4)\n"
                "print(\"listening\", flush=True)\n"
                "while True:\n"
                "    c,_=s.accept(); c.sendall(b\"from-sandbox\"); c.close()\n"
            )
            await asyncio.to_thread(
                sb.exec, ["python3", "-c", code], 60.0, True  # background
            )
            url = await asyncio.to_thread(sb.expose_port, 47654)
            host, port = url.replace("http://", "").split(":")
            data = b""
            for _ in range(50):
                try:
                    r, w = await asyncio.open_connection(host, int(port))
                    data = await r.read(100)
                    w.close()
                    if data:
                        break
                except OSError:
                    pass
                await asyncio.sleep(0.2)
            assert data == b"from-sandbox"
        finally:
            await asyncio.to_thread(sb.terminate)


@pytest.mark.asyncio
async def test_function_invoke_streams_logs():
    async with cluster(n_workers=1):
        from beta9_amd.types import RuntimeConfig, StubConfig

        import handlers  # noqa

        fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
        stub_id = await asyncio.to_thread(fn.prepare_runtime)
        client = beta9_amd.Client()

        def run():
            msgs = list(
                client.stream(
                    "FunctionService",
                    "FunctionInvoke",
                    {"stub_id": stub_id, "args": {"args": [2], "kwargs": {}}},
                )
            )
            return msgs

        msgs = await asyncio.to_thread(run)
        kinds = {m["type"] for m in msgs}
        assert "result" in kinds
        result = [m for m in msgs if m["type"] == "result"][0]
        assert result["result"] == {"hello": 4}


@pytest.mark.asyncio
async def test_affinity_routing_consistent_hash():
    """Same affinity key -> same healthy container; saturation falls back
    (reference analog: pod/llm.go prefix-affinity hashing)."""
    from beta9_amd.abstractions.common.buffer import RequestBuffer, _ContainerSlot

    buf = RequestBuffer.__new__(RequestBuffer)
    buf.tokens_per_container = 2
    slots = {f"c{i}": _ContainerSlot(f"127.0.0.1:{9000+i}", 2) for i in range(4)}
    for s in slots.values():
        s.healthy = True

    async def fake_discover():
        return slots

    buf._discover = fake_discover
    import time as _t

    picks = set()
    for _ in range(5):
        cid, slot = await buf._acquire_slot(_t.monotonic() + 1.0, affinity_key="session-42")
        picks.add(cid)
        slot.sema.release()
    assert len(picks) == 1  # deterministic routing

    # saturate the preferred container -> falls back to another
    pref = picks.pop()
    await slots[pref].sema.acquire()
    await slots[pref].sema.acquire()
    cid2, slot2 = await buf._acquire_slot(_t.monotonic() + 1.0, affinity_key="session-42")
    assert cid2 != pref
    slot2.sema.release()
    slots[pref].sema.release(); slots[pref].sema.release()

    # different keys spread across containers
    spread = set()
    for k in range(20):
        cid, slot = await buf._acquire_slot(_t.monotonic() + 1.0, affinity_key=f"k{k}")
        spread.add(cid)
        slot.sema.release()
    assert len(spread) >= 2


@pytest.mark.asyncio
async def test_sni_tls_ingress():
    """SNI TLS ingress: one shared TLS port routes by hostname to the right
    container port with a per-host self-signed cert (reference:
    pkg/abstractions/pod/tcp.go:48-144)."""
    import ssl

    import beta9_amd

    from .helpers import cluster

    async with cluster(n_workers=1) as c:
        sb = await asyncio.to_thread(beta9_amd.Sandbox(cpu=100, memory=128).create)
        try:
            client = beta9_amd.Client()
            # serve one TLS-proxied connection from inside the sandbox
            code = (
                "import socket\n"
                "s=socket.socket(); s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR,1)\n"
                "s.bind(('0.0.0.0',9443)); s.listen(1); s.settimeout(20)\n"
                "c,_=s.accept(); c.sendall(b'tls-pong'); c.close(); print('served')\n"
            )
            r = await asyncio.to_thread(
                client.call, "PodService", "SandboxExposePort",
                {"container_id": sb.container_id, "port": 9443, "tls": True}, 30.0,
            )
            assert r.get("tls_port") and r.get("tls_sni"), r
            run_task = asyncio.create_task(asyncio.to_thread(
                client.call, "PodService", "SandboxRunCode",
                {"container_id": sb.container_id, "code": code, "timeout": 30.0}, 40.0,
            ))
            ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
            ctx.check_hostname = False
            ctx.verify_mode = ssl.CERT_NONE
            data = b""
            deadline = asyncio.get_event_loop().time() + 15
            while asyncio.get_event_loop().time() < deadline:
                try:
                    reader, writer = await asyncio.open_connection(
                        "127.0.0.1", r["tls_port"], ssl=ctx,
                        server_hostname=r["tls_sni"],
                    )
                    data = await asyncio.wait_for(reader.read(100), timeout=5)
                    writer.close()
                    if data:
                        break
                except (OSError, ssl.SSLError):
                    pass
                await asyncio.sleep(0.3)
            rr = await run_task
            assert data == b"tls-pong", (data, rr)
            # the presented cert carries the SNI hostname (per-stub cert)
            cert_ctx = ssl.create_default_context()
            cert_ctx.check_hostname = False
            cert_ctx.verify_mode = ssl.CERT_NONE
            # wrong SNI -> no route -> connection closes without data
            run2 = asyncio.create_task(asyncio.to_thread(
                client.call, "PodService", "SandboxRunCode",
                {"container_id": sb.container_id, "code": code, "timeout": 30.0}, 40.0,
            ))
            try:
                reader, writer = await asyncio.open_connection(
                    "127.0.0.1", r["tls_port"], ssl=ctx,
                    server_hostname=f"nope.tcp.{c.cfg.gateway.base_domain}",
                )
                d2 = await asyncio.wait_for(reader.read(100), timeout=5)
                writer.close()
                assert d2 == b""
            except (OSError, ssl.SSLError, asyncio.TimeoutError):
                pass  # refused handshake is also a valid "no route"
            run2.cancel()
        finally:
            await asyncio.to_thread(sb.terminate)
