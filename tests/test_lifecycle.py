"""Worker lifecycle unit tests (reference analog: pkg/worker/
lifecycle_test.go — spec generation/env/mounts/exit-code normalization with
fake repo clients)."""

import asyncio
import os

import pytest

from beta9_amd.types import CheckpointState, CheckpointStatus, ContainerRequest
from beta9_amd.worker.gpu import GPUManager, make_fake_topology
from beta9_amd.worker.lifecycle import Lifecycle
from beta9_amd.worker.runtime.proc import ProcRuntime, _Proc


class FakeRepo:
    def __init__(self):
        self.spans = []
        self.logs = []
        self.events = []
        self.addresses = {}
        self.exit_codes = {}
        self.stops = set()

    async def push_span(self, cid, span):
        self.spans.append(span)

    async def push_event(self, name, **payload):
        self.events.append({"name": name, **payload})

    async def append_log(self, cid, line):
        self.logs.append(line)

    async def update_container_status(self, cid, status):
        pass

    async def set_container_address(self, cid, stub_id, addr):
        self.addresses[cid] = addr

    async def remove_container_address(self, cid, stub_id):
        self.addresses.pop(cid, None)

    async def set_exit_code(self, cid, code):
        self.exit_codes[cid] = code

    async def delete_container_state(self, cid):
        pass

    async def stop_requested(self, cid):
        return cid in self.stops


def make_lifecycle(tmp_path, gpus=0):
    gm = None
    if gpus:
        topo = make_fake_topology(str(tmp_path / "topo"), n_gpus=gpus)
        gm = GPUManager(topology_path=topo)
    return Lifecycle(
        worker_id="w1",
        repo=FakeRepo(),
        runtime=ProcRuntime(),
        gpu_manager=gm,
        gateway_host="127.0.0.1",
        gateway_grpc_port=1993,
    ), gm


def test_spec_env_contract(tmp_path):
    lc, _ = make_lifecycle(tmp_path)
    req = ContainerRequest(
        container_id="c1",
        stub_id="s1",
        workspace_id="ws1",
        stub_type="endpoint/deployment",
        env={"HANDLER": "m:f"},
        cpu=500,
        memory_mb=256,
        task_id="t1",
    )
    from beta9_amd.worker.lifecycle import ContainerInstance

    inst = ContainerInstance(req)
    spec, needs_addr = lc._spec_from_request(req, inst)
    assert needs_addr and inst.port > 0
    env = spec.env
    assert env["CONTAINER_ID"] == "c1"
    assert env["STUB_ID"] == "s1"
    assert env["STUB_TYPE"] == "endpoint/deployment"
    assert env["WORKSPACE_ID"] == "ws1"
    assert env["B9_GATEWAY_PORT"] == "1993"
    assert env["TASK_ID"] == "t1"
    assert env["BIND_PORT"] == str(inst.port)
    assert env["HANDLER"] == "m:f"
    assert spec.entry_point[-1] == "beta9_amd.runner.endpoint"
    assert spec.cpu_millicores == 500 and spec.memory_mb == 256


def test_runner_module_selection(tmp_path):
    lc, _ = make_lifecycle(tmp_path)
    from beta9_amd.worker.lifecycle import ContainerInstance

    for stub_type, module in [
        ("function", "beta9_amd.runner.function"),
        ("taskqueue/deployment", "beta9_amd.runner.taskqueue"),
        ("sandbox", "beta9_amd.runner.sandbox"),
        ("pod/run", "beta9_amd.runner.pod"),
        ("asgi/serve", "beta9_amd.runner.endpoint"),
    ]:
        req = ContainerRequest(container_id="x", stub_type=stub_type, cpu=1, memory_mb=1)
        spec, _ = lc._spec_from_request(req, ContainerInstance(req))
        assert spec.entry_point[-1] == module, stub_type


@pytest.mark.asyncio
async def test_gpu_assignment_and_rendezvous_injection(tmp_path):
    lc, gm = make_lifecycle(tmp_path, gpus=8)
    req = ContainerRequest(
        container_id="gc1",
        stub_id="s1",
        workspace_id="ws1",
        stub_type="endpoint",
        gpu="MI355X",
        gpu_count=4,
        cpu=100,
        memory_mb=64,
        entry_point=["true"],  # exits immediately
    )
    code = await lc.run_container(req)
    # true exits 0; gpu env must have been injected and released
    assert gm.free_count == 8
    spans = [s["name"] for s in lc.repo.spans]
    assert "worker.gpu_assignment" in spans


@pytest.mark.asyncio
async def test_checkpoint_env_attach(tmp_path):
    lc, _ = make_lifecycle(tmp_path)
    req = ContainerRequest(
        container_id="ck1",
        stub_id="s1",
        stub_type="endpoint",
        cpu=100,
        memory_mb=64,
        checkpoint_enabled=True,
        checkpoint=CheckpointState(
            checkpoint_id="cp1", remote_key="ws-stub", status=CheckpointStatus.available
        ),
        entry_point=["sh", "-c", "echo ck=$B9_CHECKPOINT_KEY"],
    )
    await lc.run_container(req)
    for _ in range(50):
        if any("ck=ws-stub" in l for l in lc.repo.logs):
            break
        await asyncio.sleep(0.02)
    assert any("ck=ws-stub" in l for l in lc.repo.logs)
    assert any(e["name"] == "container.started" for e in lc.repo.events)
    assert any(e["name"] == "container.stopped" for e in lc.repo.events)


@pytest.mark.asyncio
async def test_exit_code_and_stop(tmp_path):
    lc, _ = make_lifecycle(tmp_path)
    req = ContainerRequest(
        container_id="e1", stub_type="function", cpu=100, memory_mb=64,
        entry_point=["sh", "-c", "exit 7"],
    )
    code = await lc.run_container(req)
    assert code == 7
    assert lc.repo.exit_codes["e1"] == 7

    # stop request kills a long-running container
    req2 = ContainerRequest(
        container_id="e2", stub_type="function", cpu=100, memory_mb=64,
        entry_point=["sleep", "300"],
    )
    task = asyncio.create_task(lc.run_container(req2))
    await asyncio.sleep(0.3)
    lc.repo.stops.add("e2")
    code = await asyncio.wait_for(task, timeout=15)
    assert code != 0  # killed


def test_oom_classification(tmp_path):
    rt = ProcRuntime()
    cg = tmp_path / "cg"
    cg.mkdir()
    (cg / "memory.events").write_text("low 0\nhigh 2\nmax 5\noom 1\noom_kill 1\n")

    class P:
        returncode = 1

    entry = _Proc.__new__(_Proc)
    entry.proc = P()
    entry.cgroup_dir = str(cg)
    entry.oom_killed = False
    assert rt._check_oom(entry) is True


@pytest.mark.asyncio
async def test_fractional_gpu_env_injection(tmp_path):
    """A gpu_fraction=0.5 container gets the shared device + the HBM budget
    env the runner enforces (Thunder-analog — pkg/worker/thunder.go)."""
    lc, gm = make_lifecycle(tmp_path, gpus=1)
    req = ContainerRequest(
        container_id="frac1",
        stub_id="s1",
        workspace_id="ws1",
        stub_type="function",
        gpu="MI355X",
        gpu_count=1,
        gpu_fraction=0.5,
        cpu=100,
        memory_mb=64,
        entry_point=["env"],
    )
    code = await lc.run_container(req)
    assert code == 0
    assert gm.free_count == 1 and gm.free_units == 1.0  # released after exit
    logs = "\n".join(lc.repo.logs)
    assert "B9_GPU_MEMORY_FRACTION=0.5" in logs
