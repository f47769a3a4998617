"""Multi-GPU rendezvous runner on CPU: WORLD_SIZE=2 over gloo, full rank
spawn + broadcast dispatch + tensor-parallel all-reduce (the gpu_count>1
endpoint path of SURVEY §2.6 item 6, exercised without GPUs)."""

import os
import socket
import subprocess
import sys
import time

import httpx
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(180)
def test_mgpu_endpoint_gloo_world2():
    bind = free_port()
    master = free_port()
    env = dict(os.environ)
    env.update(
        {
            "WORLD_SIZE": "2",
            "B9_RDZV_BACKEND": "gloo",
            "B9_RDZV_ORDINALS": "0,1",
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(master),
            "GLOO_SOCKET_IFNAME": "lo",
            "BIND_PORT": str(bind),
            "HANDLER": "beta9_amd.models.bench_handler:infer",
            "ON_START": "beta9_amd.models.bench_handler:load",
            "STUB_ID": "tp-test",
            "STUB_TYPE": "endpoint",
            "CONTAINER_ID": "tp-c1",
            "B9_GATEWAY_HOST": "127.0.0.1",
            "B9_GATEWAY_PORT": "1",
            "B9_BENCH_HIDDEN": "64",
            "B9_BENCH_INTER": "128",
            "B9_BENCH_LAYERS": "2",
            "TIMEOUT": "60",
        }
    )
    env.pop("RANK", None)
    env.pop("HIP_VISIBLE_DEVICES", None)
    proc = subprocess.Popen(
        [sys.executable, "-u", "-m", "beta9_amd.runner.endpoint"],
        env=env,
        cwd=REPO,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        start_new_session=True,
    )
    import asyncio

    from beta9_amd.rpc.channel import RpcClient

    async def call(method, payload, timeout=60.0):
        c = RpcClient(f"127.0.0.1:{bind}")
        try:
            return await c.call("Runner", method, payload, timeout=timeout)
        finally:
            await c.close()

    try:
        deadline = time.monotonic() + 120
        up = False
        while time.monotonic() < deadline:
            if proc.poll() is not None:
                out = proc.stdout.read().decode(errors="replace")
                raise AssertionError(f"runner exited early:\n{out[-3000:]}")
            try:
                h = asyncio.run(call("Health", {}, timeout=2.0))
                if h.get("ok"):
                    up = True
                    break
            except Exception:
                pass
            time.sleep(0.2)
        assert up, "mgpu endpoint never became healthy"

        r = asyncio.run(call("Invoke", {"args": {"tokens": 8}}))
        assert r["status_code"] == 200, str(r)
        result = r["body"]["result"]
        assert result["world"] == 2
        assert result["load"]["world"] == 2
        assert result["load"]["backend"] == "gloo"
        first_checksum = result["checksum"]

        # second request exercises the broadcast loop again
        r2 = asyncio.run(call("Invoke", {"args": {"tokens": 8}}))
        assert r2["status_code"] == 200
        assert r2["body"]["result"]["checksum"] != 0.0
        assert first_checksum != 0.0
    finally:
        import signal

        try:
            os.killpg(proc.pid, signal.SIGKILL)
        except ProcessLookupError:
            pass
        proc.wait(timeout=10)
