"""Failure handling: task retry after container death, autoscaler math,
dispatcher timeout."""

import asyncio

import pytest

import beta9_amd
from beta9_amd.types import StubConfig, TaskPolicy
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_task_retried_after_container_killed():
    async with cluster(n_workers=1) as c:
        # tighten heartbeat so the monitor notices quickly
        c.gateway.dispatcher.heartbeat_ttl_s = 2.0
        c.gateway.dispatcher._monitor_interval_s = 0.3

        from beta9_amd.types import RuntimeConfig

        cfg = StubConfig(
            runtime=RuntimeConfig(cpu=100, memory_mb=128),
            handler="handlers:slow",
            task_policy=TaskPolicy(max_retries=2, timeout=120.0),
        )
        ws = c.creds["workspace_id"]
        stub = c.gateway.db.get_or_create_stub(ws, "retryfn", "function", cfg)
        task = await c.gateway.dispatcher.send_and_execute(
            "function", ws, stub["id"], {"args": [3.0], "kwargs": {}},
            policy=cfg.task_policy,
        )
        # wait for the first container to claim and start running
        for _ in range(200):
            t = await c.gateway.dispatcher.get_task(task.task_id)
            if t.status.value == "RUNNING" and t.container_id:
                break
            await asyncio.sleep(0.05)
        assert t.container_id
        first_container = t.container_id
        # kill the running container out from under the task
        worker = c.workers[0]
        await worker.lifecycle.runtime_of(first_container).kill(first_container, 9)

        done = await c.gateway.dispatcher.wait(task.task_id, timeout=90.0)
        result = await c.gateway.dispatcher.get_result(task.task_id)
        assert done.status.value == "COMPLETE", (done.status, result)
        assert done.retries >= 1
        assert result == "done"


@pytest.mark.asyncio
async def test_task_timeout_enforced():
    async with cluster(n_workers=1) as c:
        c.gateway.dispatcher._monitor_interval_s = 0.3
        from beta9_amd.types import RuntimeConfig

        cfg = StubConfig(
            runtime=RuntimeConfig(cpu=100, memory_mb=128),
            handler="handlers:slow",
            task_policy=TaskPolicy(max_retries=0, timeout=1.0),
        )
        ws = c.creds["workspace_id"]
        stub = c.gateway.db.get_or_create_stub(ws, "slowfn", "function", cfg)
        task = await c.gateway.dispatcher.send_and_execute(
            "function", ws, stub["id"], {"args": [30.0], "kwargs": {}},
            policy=cfg.task_policy,
        )
        done = await c.gateway.dispatcher.wait(task.task_id, timeout=60.0)
        assert done.status.value in ("TIMEOUT", "ERROR")


def test_deployment_scale_fn_math():
    from beta9_amd.abstractions.common.autoscaler import deployment_scale_fn
    from beta9_amd.types import AutoscalerConfig, StubConfig

    cfg = StubConfig(autoscaler=AutoscalerConfig(max_containers=5, tasks_per_container=2, min_containers=1))
    fn = deployment_scale_fn(cfg, stub_limit_max_replicas=10)
    assert fn(0) == 1  # min_containers floor
    assert fn(1) == 1
    assert fn(2) == 1
    assert fn(3) == 2
    assert fn(10) == 5  # max_containers cap
    assert fn(100) == 5

    # gateway stub limit caps below max_containers
    fn2 = deployment_scale_fn(cfg, stub_limit_max_replicas=3)
    assert fn2(100) == 3
