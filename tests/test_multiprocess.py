"""Full multi-process path: gateway CLI process + provisioned worker
processes + SDK over real gRPC — the closest CPU analog of production
(RemoteRepoClient, LocalProcessPoolController, delivery-token acks)."""

import json
import os
import signal
import subprocess
import sys
import time

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(240)
def test_gateway_process_with_provisioned_workers(tmp_path):
    cfg = {
        "database": {"path": str(tmp_path / "b.db")},
        "gateway": {"grpc_port": 0, "http_port": 0},
        "storage": {"root": str(tmp_path / "ws")},
        "cache": {"root": str(tmp_path / "cache")},
        "checkpoint": {"store_path": str(tmp_path / "ckpt")},
        "image_service": {"store_path": str(tmp_path / "img")},
        "worker": {"images_path": str(tmp_path / "wimg"), "scratch_path": str(tmp_path / "wscr")},
    }
    env = dict(os.environ)
    env["CONFIG_JSON"] = json.dumps(cfg)
    env["HOME"] = str(tmp_path)  # keep ~/.b9 out of the real home
    proc = subprocess.Popen(
        [sys.executable, "-u", "-m", "beta9_amd.cli", "gateway"],
        env=env,
        cwd=REPO,
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
        start_new_session=True,
    )
    creds = {}
    try:
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline and len(creds) < 2:
            line = proc.stdout.readline()
            if not line:
                break
            if line.startswith("token:"):
                creds["token"] = line.split()[-1]
            elif line.startswith("grpc:"):
                creds["grpc"] = line.split()[-1]
        assert "token" in creds and "grpc" in creds, "gateway did not print creds"

        import beta9_amd

        beta9_amd.configure(creds["grpc"], creds["token"])
        import handlers

        fn = beta9_amd.Function(handlers.add, cpu=100, memory=128)
        # no workers exist: the scheduler must provision one through the
        # local process pool, which spawns `python -m beta9_amd.worker`
        result = fn.remote(20, 22)
        assert result == 42

        # second call reuses the provisioned worker (no second cold boot)
        assert fn.remote(1, 2) == 3

        client = beta9_amd.Client()
        workers = client.call("GatewayService", "ListWorkers", {})
        assert len(workers) >= 1
        client.close()

        # the CLI works against the same gateway (context written by
        # `b9 gateway` bootstrap into $HOME/.b9/config.yaml)
        cli = subprocess.run(
            [sys.executable, "-m", "beta9_amd.cli", "task", "list"],
            env=env,
            cwd=REPO,
            capture_output=True,
            text=True,
            timeout=60,
        )
        assert cli.returncode == 0, cli.stderr[-500:]
        assert "task-" in cli.stdout

        cli = subprocess.run(
            [sys.executable, "-m", "beta9_amd.cli", "worker", "list"],
            env=env, cwd=REPO, capture_output=True, text=True, timeout=60,
        )
        assert cli.returncode == 0 and '"id"' in cli.stdout
    finally:
        try:
            os.killpg(proc.pid, signal.SIGTERM)
            proc.wait(timeout=15)
        except (subprocess.TimeoutExpired, ProcessLookupError):
            try:
                os.killpg(proc.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass


@pytest.mark.timeout(180)
def test_agent_join(tmp_path):
    """`b9 agent join`: preflight + identity persist + worker registration
    (reference: pkg/agent/agent.go:17-88)."""
    import asyncio

    async def amain():
        from beta9_amd.common.config import AppConfig
        from beta9_amd.gateway.gateway import Gateway

        cfg = AppConfig().with_root(str(tmp_path / "gw"))
        cfg.gateway.grpc_port = 0
        cfg.gateway.http_port = 0
        gw = Gateway(cfg)
        await gw.start()
        creds = gw.bootstrap_workspace("agents")
        env = dict(os.environ)
        env["HOME"] = str(tmp_path)
        env["B9_DATA_ROOT"] = str(tmp_path / "agentdata")
        proc = subprocess.Popen(
            [sys.executable, "-m", "beta9_amd.cli", "agent", "join",
             f"127.0.0.1:{gw.grpc_port}", "--token", creds["token"],
             "--worker-id", "agent-test-w", "--once"],
            env=env, cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            text=True, start_new_session=True,
        )
        try:
            deadline = time.monotonic() + 60
            seen = False
            while time.monotonic() < deadline:
                ws = await gw.worker_repo.list()
                if any(w.id == "agent-test-w" for w in ws):
                    seen = True
                    break
                if proc.poll() is not None:
                    raise AssertionError(f"agent exited: {proc.stdout.read()[-2000:]}")
                await asyncio.sleep(0.3)
            assert seen, "agent worker never registered"
            assert os.path.exists(tmp_path / ".b9" / "agent.json")
        finally:
            try:
                os.killpg(proc.pid, signal.SIGTERM)
                proc.wait(timeout=15)
            except (subprocess.TimeoutExpired, ProcessLookupError):
                try:
                    os.killpg(proc.pid, signal.SIGKILL)
                except ProcessLookupError:
                    pass
            await gw.shutdown()

    asyncio.run(amain())


def test_bench_world2_cpu_contract():
    """The driver's SCALE run shape: torchrun world=2 over gloo on CPU —
    one gateway+worker on rank 0, per-rank out-of-process load generators,
    barriers + reduction, ONE JSON line from rank 0 (bench.py contract)."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29377", os.path.join(repo, "bench.py"),
         "--gpus", "2", "--steps", "3", "--warmup", "1", "--cold-iters", "1",
         "--cold-miss-iters", "0", "--hidden", "64", "--inter", "128",
         "--layers", "1", "--tokens", "8"],
        capture_output=True, text=True, timeout=420, env=env, cwd=repo,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    assert d["metric"] == "tasks_per_sec" and d["n_gpus"] == 2
    assert d["value"] > 0 and len(d["cold_start_concurrent_ms"]) == 2
