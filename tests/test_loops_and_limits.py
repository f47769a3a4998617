"""Gateway background loops (cron, pool cleaner), workspace concurrency
limits, metrics registry, events pipeline."""

import asyncio
import time

import pytest

import beta9_amd
from beta9_amd.common.metrics import MetricsRegistry
from beta9_amd.scheduler.scheduler import ConcurrencyLimitError
from beta9_amd.types import ContainerRequest
from tests.helpers import cluster


def test_metrics_registry():
    m = MetricsRegistry()
    m.inc("requests_total", 1, route="a")
    m.inc("requests_total", 2, route="a")
    m.set_gauge("backlog", 5)
    for v in [1, 5, 20, 200, 900]:
        m.observe("latency_ms", v)
    snap = m.snapshot()
    assert snap["counters"]['requests_total{route="a"}'] == 3
    assert snap["gauges"]["backlog"] == 5
    h = snap["histograms"]["latency_ms"]
    assert h["count"] == 5 and h["p50"] >= 5
    text = m.prometheus_text()
    assert "latency_ms_count 5" in text


@pytest.mark.asyncio
async def test_events_pipeline():
    async with cluster(n_workers=0) as c:
        await c.gateway.events.emit("container.scheduled", container_id="c1")
        await c.gateway.events.emit("task.created", task_id="t1")
        recent = await c.gateway.events.recent(limit=10)
        assert len(recent) >= 2
        only_containers = await c.gateway.events.recent(name_prefix="container.")
        assert all(e["name"].startswith("container.") for e in only_containers)
        assert only_containers[-1]["payload"]["container_id"] == "c1"


@pytest.mark.asyncio
async def test_scheduler_emits_events_and_metrics():
    async with cluster(n_workers=1) as c:
        import handlers

        fn = beta9_amd.Function(handlers.hello, cpu=100, memory=128)
        await asyncio.to_thread(fn.remote, 1)
        ev = await c.gateway.events.recent(name_prefix="container.scheduled")
        assert len(ev) >= 1
        snap = c.gateway.metrics.snapshot()
        assert snap["counters"].get("scheduler_containers_scheduled", 0) >= 1


@pytest.mark.asyncio
async def test_concurrency_limit_gpu():
    async with cluster(n_workers=1, fake_gpus=8) as c:
        ws_id = c.creds["workspace_id"]
        c.gateway.db._exec(
            "UPDATE workspaces SET concurrency_limit_gpu=2 WHERE id=?", (ws_id,)
        )
        req1 = ContainerRequest(
            container_id="g1", cpu=100, memory_mb=128, gpu="MI355X", gpu_count=2,
            workspace_id=ws_id, stub_id="s", stub_type="endpoint",
        )
        await c.gateway.scheduler.run(req1)  # takes the full GPU budget
        req2 = ContainerRequest(
            container_id="g2", cpu=100, memory_mb=128, gpu="MI355X", gpu_count=1,
            workspace_id=ws_id, stub_id="s", stub_type="endpoint",
        )
        with pytest.raises(ConcurrencyLimitError):
            await c.gateway.scheduler.run(req2)
        await c.gateway.scheduler.stop("g1")


@pytest.mark.asyncio
async def test_multi_gpu_gate():
    async with cluster(n_workers=1, fake_gpus=8) as c:
        ws_id = c.creds["workspace_id"]
        c.gateway.db._exec("UPDATE workspaces SET multi_gpu_enabled=0 WHERE id=?", (ws_id,))
        req = ContainerRequest(
            container_id="mg1", cpu=100, memory_mb=128, gpu="MI355X", gpu_count=4,
            workspace_id=ws_id, stub_id="s", stub_type="endpoint",
        )
        with pytest.raises(ConcurrencyLimitError):
            await c.gateway.scheduler.run(req)


@pytest.mark.asyncio
async def test_cron_loop_fires():
    async with cluster(n_workers=1) as c:
        import handlers

        # register a schedule stub with a very short interval
        from beta9_amd.types import StubConfig

        cfg = StubConfig(handler="handlers:hello", timeout=30)
        stub = c.gateway.db.get_or_create_stub(
            c.creds["workspace_id"], "tick", "schedule/deployment", cfg
        )
        await c.gateway.store.set(
            f"schedule:{c.creds['workspace_id']}:{stub['id']}",
            {"interval_s": 0.5, "last": 0.0},
        )
        # the cron loop should dispatch at least one function task
        for _ in range(100):
            tasks = c.gateway.db.list_tasks(c.creds["workspace_id"], stub["id"])
            c.gateway.db.flush()
            if tasks:
                break
            await asyncio.sleep(0.2)
        assert tasks, "cron never dispatched"


@pytest.mark.asyncio
async def test_pool_cleaner_reaps_dead_worker():
    async with cluster(n_workers=1) as c:
        from beta9_amd.gateway.loops import PoolCleaner
        from beta9_amd.types import Worker, WorkerStatus

        dead = Worker(
            id="deadw", status=WorkerStatus.available, total_cpu=1000,
            total_memory_mb=1024, free_cpu=1000, free_memory_mb=1024,
        )
        await c.gateway.worker_repo.register(dead)
        # kill its keepalive
        from beta9_amd.common import keys

        await c.gateway.store.delete(keys.worker_keepalive("deadw"))
        cleaner = PoolCleaner(c.gateway, interval_s=0.1)
        cleaner.start()
        try:
            for _ in range(50):
                if await c.gateway.worker_repo.get("deadw") is None:
                    break
                await asyncio.sleep(0.1)
            assert await c.gateway.worker_repo.get("deadw") is None
        finally:
            await cleaner.stop()
