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
