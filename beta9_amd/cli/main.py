"""b9 CLI — operator and developer surface.

(reference: sdk/src/beta9/cli/main.py click multi-group: config/container/
deploy/deployment/run/secret/serve/task/token/volume/worker/pool). Run as
`python -m beta9_amd.cli` or the `b9` console script.
"""

from __future__ import annotations

import importlib
import json
import os
import sys
from typing import List, Optional

import typer
import yaml

app = typer.Typer(help="beta9_amd — MI355X-native serverless GPU runtime")
config_app = typer.Typer(help="client contexts")
task_app = typer.Typer(help="tasks")
container_app = typer.Typer(help="containers")
deployment_app = typer.Typer(help="deployments")
secret_app = typer.Typer(help="secrets")
volume_app = typer.Typer(help="volumes")
worker_app = typer.Typer(help="workers")
app.add_typer(config_app, name="config")
app.add_typer(task_app, name="task")
app.add_typer(container_app, name="container")
app.add_typer(deployment_app, name="deployment")
app.add_typer(secret_app, name="secret")
app.add_typer(volume_app, name="volume")
app.add_typer(worker_app, name="worker")

CONFIG_PATH = os.path.expanduser("~/.b9/config.yaml")


def _load_context() -> dict:
    if os.path.exists(CONFIG_PATH):
        with open(CONFIG_PATH) as f:
            return yaml.safe_load(f) or {}
    return {}


def _client():
    import beta9_amd

    ctx = _load_context()
    if ctx.get("gateway_addr"):
        beta9_amd.configure(ctx["gateway_addr"], ctx.get("token", ""))
    return beta9_amd.Client()


def _load_handler(spec: str):
    """Load module.py:fn or module:fn from cwd."""
    mod_name, _, fn_name = spec.partition(":")
    if mod_name.endswith(".py"):
        mod_name = mod_name[:-3].replace("/", ".")
    sys.path.insert(0, os.getcwd())
    module = importlib.import_module(mod_name)
    obj = module
    for part in fn_name.split("."):
        obj = getattr(obj, part)
    return obj


# -- servers ---------------------------------------------------------------


@app.command()
def gateway(
    config: Optional[str] = typer.Option(None, help="config yaml path"),
    bootstrap: bool = typer.Option(True, help="create default workspace+token on first boot"),
):
    """Run the gateway (control plane) on this node."""
    import asyncio

    from ..common.config import load_config
    from ..gateway.gateway import Gateway
    from ..scheduler.pool import LocalProcessPoolController

    async def amain():
        cfg = load_config(path=config)
        gw = Gateway(cfg)
        await gw.start()
        system = gw.bootstrap_workspace("system")
        for pool_cfg in cfg.scheduler.pools:
            gw.scheduler.register_pool(
                LocalProcessPoolController(
                    cfg,
                    pool_cfg,
                    gateway_addr=f"{cfg.gateway.host}:{gw.grpc_port}",
                    worker_token=system["token"],
                )
            )
        if bootstrap:
            creds = gw.bootstrap_workspace("default")
            typer.echo(f"workspace: {creds['workspace_id']}")
            typer.echo(f"token:     {creds['token']}")
            typer.echo(f"grpc:      {cfg.gateway.host}:{gw.grpc_port}")
            typer.echo(f"http:      {cfg.gateway.host}:{gw.http_port}")
            os.makedirs(os.path.dirname(CONFIG_PATH), exist_ok=True)
            with open(CONFIG_PATH, "w") as f:
                yaml.dump(
                    {"gateway_addr": f"{cfg.gateway.host}:{gw.grpc_port}", "token": creds["token"]},
                    f,
                )
        stop = asyncio.Event()
        import signal as _signal

        loop = asyncio.get_event_loop()
        for s in (_signal.SIGINT, _signal.SIGTERM):
            loop.add_signal_handler(s, stop.set)
        await stop.wait()
        await gw.shutdown()

    asyncio.run(amain())


@worker_app.command("start")
def worker_start():
    """Run a worker against GATEWAY_ADDR (env contract)."""
    import asyncio

    from ..worker.__main__ import amain

    asyncio.run(amain())


agent_app = typer.Typer(help="BYO-machine agent (reference: pkg/agent)")
app.add_typer(agent_app, name="agent")


@agent_app.command("join")
def agent_join(
    gateway_addr: str,
    token: str = typer.Option(..., help="workspace/worker token"),
    worker_id: str = typer.Option("", help="stable worker id (default: host-derived)"),
    cpu_limit: int = typer.Option(0, help="millicores (0 = all cores)"),
    memory_limit_mb: int = typer.Option(0, help="0 = autodetect"),
    supervise: bool = typer.Option(True, help="restart the worker on exit"),
    once: bool = typer.Option(False, help="preflight + single worker run (tests)"),
):
    """Join THIS machine to a gateway as a worker node (reference:
    pkg/agent/agent.go:17-88 — preflight, persist identity, run the worker
    under supervision). Preflight checks: gateway reachable+token valid,
    GPU inventory, namespace privileges, disk space."""
    import json as _json
    import shutil
    import socket
    import subprocess
    import time

    from ..rpc.channel import RpcClient
    from ..worker.gpu import discover_devices
    from ..worker.runtime.native import namespaces_available

    # -- preflight ----------------------------------------------------------
    import asyncio

    async def _auth_check():
        c = RpcClient(gateway_addr, token=token)
        try:
            return await c.call("GatewayService", "Authorize", {"token": token}, timeout=10.0)
        finally:
            await c.close()

    r = asyncio.run(_auth_check())
    if not r.get("ok"):
        typer.echo("preflight FAILED: token rejected by gateway")
        raise typer.Exit(1)
    typer.echo(f"gateway ok (workspace {r['workspace_name']})")
    devices = discover_devices()
    typer.echo(f"GPUs: {len(devices)}" + (f" ({devices[0].name})" if devices else ""))
    typer.echo(f"namespace isolation: {'available' if namespaces_available() else 'UNAVAILABLE (ProcRuntime fallback)'}")
    free_gb = shutil.disk_usage("/").free / 1e9
    typer.echo(f"free disk: {free_gb:.0f} GB")
    if free_gb < 5:
        typer.echo("preflight FAILED: <5 GB free disk")
        raise typer.Exit(1)

    wid = worker_id or f"agent-{socket.gethostname()}"
    if not cpu_limit:
        cpu_limit = (os.cpu_count() or 8) * 1000
    if not memory_limit_mb:
        try:
            with open("/proc/meminfo") as f:
                kb = int(f.readline().split()[1])
            memory_limit_mb = max(1024, int(kb / 1024 * 0.9))
        except OSError:
            memory_limit_mb = 16384
    state = {
        "gateway_addr": gateway_addr, "worker_id": wid,
        "cpu_limit": cpu_limit, "memory_limit_mb": memory_limit_mb,
        "joined_at": time.time(),
    }
    agent_path = os.path.join(os.path.dirname(CONFIG_PATH), "agent.json")
    os.makedirs(os.path.dirname(agent_path), exist_ok=True)
    with open(agent_path, "w") as f:
        _json.dump(state, f)
    typer.echo(f"agent identity persisted to {agent_path}")

    env = dict(os.environ)
    env.update({
        "WORKER_ID": wid, "GATEWAY_ADDR": gateway_addr, "WORKER_TOKEN": token,
        "CPU_LIMIT": str(cpu_limit), "MEMORY_LIMIT": str(memory_limit_mb),
        "GPU_TYPE": devices[0].name if devices else "",
    })
    while True:
        typer.echo(f"starting worker {wid} -> {gateway_addr}")
        p = subprocess.run([sys.executable, "-u", "-m", "beta9_amd.worker"], env=env)
        typer.echo(f"worker exited rc={p.returncode}")
        if once or not supervise:
            raise typer.Exit(p.returncode)
        time.sleep(2.0)


# -- config ----------------------------------------------------------------


@config_app.command("set")
def config_set(gateway_addr: str, token: str):
    os.makedirs(os.path.dirname(CONFIG_PATH), exist_ok=True)
    with open(CONFIG_PATH, "w") as f:
        yaml.dump({"gateway_addr": gateway_addr, "token": token}, f)
    typer.echo(f"context saved to {CONFIG_PATH}")


@config_app.command("show")
def config_show():
    typer.echo(json.dumps(_load_context(), indent=2))


# -- dev loop ---------------------------------------------------------------


@app.command()
def run(handler: str, args_json: str = typer.Option("{}", "--args")):
    """Invoke a @function once: b9 run app.py:hello --args '{"x": 2}'"""
    fn = _load_handler(handler)
    kwargs = json.loads(args_json)
    if hasattr(fn, "remote"):
        result = fn.remote(**kwargs)
    else:
        result = fn(**kwargs)
    typer.echo(json.dumps(result, default=str))


@app.command()
def deploy(handler: str, name: str = typer.Option("", help="deployment name")):
    """Deploy a decorated handler: b9 deploy app.py:api --name my-api"""
    fn = _load_handler(handler)
    if not hasattr(fn, "deploy"):
        typer.echo("handler is not a deployable abstraction", err=True)
        raise typer.Exit(1)
    r = fn.deploy(name or getattr(fn, "name", "") or "app")
    typer.echo(json.dumps(r, indent=2))


@app.command()
def serve(handler: str, timeout: float = typer.Option(600.0)):
    """Serve mode (hot dev loop): keeps one container up while active."""
    fn = _load_handler(handler)
    if not hasattr(fn, "serve"):
        typer.echo("handler does not support serve", err=True)
        raise typer.Exit(1)
    stub_id = fn.serve(timeout=timeout)
    typer.echo(f"serving stub {stub_id} (lock ttl {timeout}s)")


# -- tasks -------------------------------------------------------------------


@task_app.command("list")
def task_list(stub_id: str = typer.Option("", "--stub"), limit: int = 50):
    typer.echo(json.dumps(_client().call("GatewayService", "ListTasks", {"stub_id": stub_id, "limit": limit}), indent=2, default=str))


@task_app.command("get")
def task_get(task_id: str):
    typer.echo(json.dumps(_client().call("GatewayService", "GetTaskResult", {"task_id": task_id}), indent=2, default=str))


@task_app.command("stop")
def task_stop(task_id: str):
    _client().call("GatewayService", "StopTask", {"task_id": task_id})
    typer.echo("stopped")


# -- containers ---------------------------------------------------------------


@container_app.command("list")
def container_list():
    typer.echo(json.dumps(_client().call("GatewayService", "ListContainers", {}), indent=2, default=str))


@container_app.command("stop")
def container_stop(container_id: str):
    _client().call("GatewayService", "StopContainer", {"container_id": container_id})
    typer.echo("stop requested")


@container_app.command("logs")
def container_logs(
    container_id: str,
    tail: int = 200,
    follow: bool = typer.Option(False, "--follow", "-f"),
):
    client = _client()
    if follow:
        for msg in client.stream(
            "GatewayService", "StreamContainerLogs", {"container_id": container_id}
        ):
            if "line" in msg:
                typer.echo(msg["line"])
            if "exit_code" in msg:
                typer.echo(f"[container exited: {msg['exit_code']}]", err=True)
        return
    for line in client.call("GatewayService", "GetContainerLogs", {"container_id": container_id, "tail": tail}):
        typer.echo(line)


@container_app.command("spans")
def container_spans(container_id: str):
    spans = _client().call("GatewayService", "GetContainerSpans", {"container_id": container_id})
    for s in spans:
        typer.echo(f"{s['name']:40s} {s['duration_ms']:10.2f} ms")


# -- deployments ---------------------------------------------------------------


@deployment_app.command("list")
def deployment_list():
    typer.echo(json.dumps(_client().call("GatewayService", "ListDeployments", {}), indent=2, default=str))


@deployment_app.command("stop")
def deployment_stop(deployment_id: str):
    _client().call("GatewayService", "StopDeployment", {"deployment_id": deployment_id})
    typer.echo("stopped")


@deployment_app.command("rollout")
def deployment_rollout(name: str, version: int):
    """Point a deployment at a specific version (rollback/rollforward)."""
    r = _client().call("GatewayService", "RolloutDeployment", {"name": name, "version": version})
    typer.echo(json.dumps(r))


token_app = typer.Typer(help="workspace tokens")
app.add_typer(token_app, name="token")


@token_app.command("create")
def token_create(workspace_id: str = typer.Option("", help="defaults to the caller's workspace")):
    r = _client().call("GatewayService", "CreateToken", {"workspace_id": workspace_id} if workspace_id else {})
    typer.echo(r["token"])


@token_app.command("revoke")
def token_revoke(token: str):
    _client().call("GatewayService", "RevokeToken", {"token": token})
    typer.echo("revoked")


@app.command("workspace-create")
def workspace_create(name: str):
    r = _client().call("GatewayService", "CreateWorkspace", {"name": name})
    typer.echo(json.dumps(r))


image_app = typer.Typer(help="container images")
app.add_typer(image_app, name="image")


@image_app.command("import")
def image_import(path: str, image_id: str = typer.Option("", help="pin a specific id")):
    """Import an OCI image layout dir / docker-save tar into the layer
    store (the offline `from_registry`). Admin token required."""
    payload = {"path": path}
    if image_id:
        payload["image_id"] = image_id
    r = _client().call("ImageService", "ImportImage", payload)
    typer.echo(json.dumps(r))


@image_app.command("get")
def image_get(image_id: str):
    r = _client().call("ImageService", "GetImage", {"image_id": image_id})
    typer.echo(json.dumps(r))


# -- secrets --------------------------------------------------------------------


@secret_app.command("create")
def secret_create(name: str, value: str):
    _client().call("SecretService", "SecretCreate", {"name": name, "value": value})
    typer.echo("created")


@secret_app.command("list")
def secret_list():
    typer.echo(json.dumps(_client().call("SecretService", "SecretList", {}), indent=2))


@secret_app.command("delete")
def secret_delete(name: str):
    _client().call("SecretService", "SecretDelete", {"name": name})
    typer.echo("deleted")


# -- volumes ----------------------------------------------------------------------


@volume_app.command("ls")
def volume_ls(volume_name: str, path: str = ""):
    typer.echo(json.dumps(_client().call("VolumeService", "ListPath", {"volume_name": volume_name, "path": path}), indent=2))


@volume_app.command("cp")
def volume_cp(volume_name: str, src: str, dst: str):
    """Upload a local file into a volume."""
    with open(src, "rb") as f:
        data = f.read()
    chunks = [{"volume_name": volume_name, "path": dst, "data": data[i : i + (1 << 20)]} for i in range(0, max(len(data), 1), 1 << 20)]
    r = _client().client_stream("VolumeService", "UploadFile", chunks)
    typer.echo(json.dumps(r))


@volume_app.command("cat")
def volume_cat(volume_name: str, path: str):
    for msg in _client().stream("VolumeService", "DownloadFile", {"volume_name": volume_name, "path": path}):
        sys.stdout.buffer.write(msg["data"])


# -- workers / ops -----------------------------------------------------------------


@worker_app.command("list")
def worker_list():
    typer.echo(json.dumps(_client().call("GatewayService", "ListWorkers", {}), indent=2, default=str))


@worker_app.command("cordon")
def worker_cordon(worker_id: str, uncordon: bool = typer.Option(False)):
    _client().call("GatewayService", "CordonWorker", {"worker_id": worker_id, "cordoned": not uncordon})


@worker_app.command("drain")
def worker_drain(worker_id: str, timeout: float = typer.Option(120.0), remove: bool = typer.Option(False)):
    """Cordon, wait for active containers to finish, optionally remove."""
    r = _client().call(
        "GatewayService", "DrainWorker",
        {"worker_id": worker_id, "timeout": timeout, "remove": remove},
        timeout=timeout + 30.0,
    )
    typer.echo(json.dumps(r, indent=2, default=str))
    typer.echo("ok")


def _pty_shell(container_id: str) -> None:
    """Raw-mode terminal bridged to the container PM's pty over gRPC."""
    import select
    import shutil
    import termios
    import threading
    import tty

    client = _client()
    size = shutil.get_terminal_size()
    r = client.call(
        "PodService", "SandboxTerminalOpen",
        {"container_id": container_id, "rows": size.lines, "cols": size.columns},
    )
    term_id = r["term_id"]
    typer.echo(f"[b9 shell] pty attached to {container_id} (exit the shell to detach)")
    stop = threading.Event()

    def reader():
        while not stop.is_set():
            try:
                rr = client.call(
                    "PodService", "SandboxTerminalRead",
                    {"container_id": container_id, "term_id": term_id, "timeout": 5.0},
                    timeout=20.0,
                )
            except Exception:
                break
            data = rr.get("data") or b""
            if data:
                os.write(1, data if isinstance(data, bytes) else data.encode())
            if rr.get("closed"):
                break
        stop.set()

    old = termios.tcgetattr(0)
    t = threading.Thread(target=reader, daemon=True)
    try:
        tty.setraw(0)
        t.start()
        while not stop.is_set():
            ready, _, _ = select.select([0], [], [], 0.1)
            if not ready:
                continue
            data = os.read(0, 4096)
            if not data:
                break
            client.call(
                "PodService", "SandboxTerminalWrite",
                {"container_id": container_id, "term_id": term_id, "data": data},
            )
    finally:
        stop.set()
        termios.tcsetattr(0, termios.TCSADRAIN, old)
        try:
            client.call(
                "PodService", "SandboxTerminalClose",
                {"container_id": container_id, "term_id": term_id},
            )
        except Exception:
            pass
        typer.echo("\n[b9 shell] detached")


@app.command()
def shell(container_id: str, pty: bool = typer.Option(True, help="full PTY terminal (raw mode); --no-pty for the line shell")):
    """Interactive shell into a sandbox/pod container (reference: beta9
    shell via in-container dropbear; here: a PTY owned by the container's
    process manager — full terminal semantics, no SSH daemon)."""
    if pty and sys.stdin.isatty():
        return _pty_shell(container_id)
    client = _client()
    typer.echo(f"connected to {container_id} (line shell; 'exit' to quit)")
    cwd = "/"
    while True:
        try:
            cmd = input(f"{container_id}:{cwd}$ ")
        except (EOFError, KeyboardInterrupt):
            break
        if cmd.strip() in ("exit", "quit"):
            break
        if not cmd.strip():
            continue
        if cmd.strip().startswith("cd "):
            cwd = cmd.strip()[3:].strip() or "/"
            continue
        try:
            r = client.call(
                "PodService",
                "SandboxExec",
                {"container_id": container_id, "command": cmd, "cwd": cwd, "timeout": 120.0},
                timeout=130.0,
            )
            if r.get("stdout"):
                typer.echo(r["stdout"], nl=False)
            if r.get("stderr"):
                typer.echo(r["stderr"], nl=False, err=True)
        except Exception as e:
            typer.echo(f"error: {e}", err=True)


@app.command()
def usage():
    """Workspace usage counters (task count, duration, billed cost)."""
    import httpx

    ctx = _load_context()
    host = (ctx.get("gateway_addr") or "127.0.0.1:1993").split(":")[0]
    r = httpx.get(
        f"http://{host}:{ctx.get('http_port', 1994)}/api/v1/usage",
        headers={"authorization": f"Bearer {ctx.get('token', '')}"},
        timeout=10.0,
    )
    typer.echo(json.dumps(r.json(), indent=2))


@app.command()
def metrics():
    """Control-plane metrics snapshot."""
    typer.echo(json.dumps(_client().call("GatewayService", "GetMetrics", {}), indent=2, default=str))


def main():
    app()


if __name__ == "__main__":
    main()
