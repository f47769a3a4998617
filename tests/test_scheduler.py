"""Scheduler selection/scoring/best-fit tests (reference analog:
pkg/scheduler/scheduler_test.go selection matrices)."""

import asyncio
import time

import pytest

from beta9_amd.backend.repos import ContainerRepo, WorkerRepo
from beta9_amd.common.config import AppConfig
from beta9_amd.common.state import StateStore
from beta9_amd.common import keys
from beta9_amd.scheduler.pool import FakePoolController
from beta9_amd.scheduler.scheduler import Scheduler
from beta9_amd.types import ContainerRequest, Worker, WorkerStatus


def make_worker(wid, cpu=10000, mem=10000, gpus=0, gpu="", pool="default", priority=0, **kw):
    return Worker(
        id=wid,
        status=WorkerStatus.available,
        total_cpu=cpu,
        total_memory_mb=mem,
        total_gpu_count=gpus,
        free_cpu=kw.pop("free_cpu", cpu),
        free_memory_mb=kw.pop("free_mem", mem),
        free_gpu_count=kw.pop("free_gpus", gpus),
        gpu=gpu,
        pool_name=pool,
        priority=priority,
        **kw,
    )


async def make_sched():
    store = StateStore()
    cfg = AppConfig()
    wr = WorkerRepo(store)
    cr = ContainerRepo(store)
    s = Scheduler(cfg, store, wr, cr)
    s.register_pool(FakePoolController())
    return s, wr, cr, store


@pytest.mark.asyncio
async def test_filter_gpu_type_and_resources():
    s, wr, cr, _ = await make_sched()
    cpu_w = make_worker("cpu1")
    gpu_w = make_worker("gpu1", gpus=8, gpu="MI355X")
    small = make_worker("small", cpu=500)
    req = ContainerRequest(container_id="c1", cpu=1000, memory_mb=512, gpu="MI355X", gpu_count=2)
    s._normalize_gpu(req)
    out = s._filter(req, [cpu_w, gpu_w, small])
    assert [w.id for w in out] == ["gpu1"]


@pytest.mark.asyncio
async def test_best_fit_prefers_fullest_worker():
    """Best-fit: fullest worker that fits wins so idle ones drain to zero
    (reference: scheduler.go:1336-1350)."""
    s, wr, cr, _ = await make_sched()
    empty = make_worker("empty")
    half = make_worker("half", free_cpu=5000, free_mem=5000)
    req = ContainerRequest(container_id="c1", cpu=1000, memory_mb=512)
    cands = s._filter(req, [empty, half])
    cands.sort(key=lambda w: (-s._score(req, w), -s._fullness(w)))
    assert cands[0].id == "half"


@pytest.mark.asyncio
async def test_gpu_preference_order_scoring():
    s, wr, cr, _ = await make_sched()
    mi355 = make_worker("a", gpus=8, gpu="MI355X")
    mi300 = make_worker("b", gpus=8, gpu="MI300X")
    req = ContainerRequest(container_id="c1", cpu=100, memory_mb=128, gpu_request=["MI355X", "MI300X"], gpu_count=1)
    s._normalize_gpu(req)
    assert s._score(req, mi355) > s._score(req, mi300)


@pytest.mark.asyncio
async def test_pool_priority_scoring():
    s, *_ = await make_sched()
    hi = make_worker("hi", priority=2)
    lo = make_worker("lo", priority=0)
    req = ContainerRequest(container_id="c1", cpu=100, memory_mb=128)
    assert s._score(req, hi) > s._score(req, lo)


@pytest.mark.asyncio
async def test_machine_pin_and_pool_selector():
    s, *_ = await make_sched()
    w1 = make_worker("w1", machine_id="m1")
    w2 = make_worker("w2", machine_id="m2")
    req = ContainerRequest(container_id="c1", cpu=100, memory_mb=128, machine_id="m2")
    assert [w.id for w in s._filter(req, [w1, w2])] == ["w2"]

    build = make_worker("b1", pool="build", requires_pool_selector=True)
    other = make_worker("o1")
    req2 = ContainerRequest(container_id="c2", cpu=100, memory_mb=128)
    assert [w.id for w in s._filter(req2, [build, other])] == ["o1"]
    req3 = ContainerRequest(container_id="c3", cpu=100, memory_mb=128, pool_selector="build")
    assert [w.id for w in s._filter(req3, [build, other])] == ["b1"]


@pytest.mark.asyncio
async def test_atomic_capacity_commit_and_release():
    s, wr, cr, store = await make_sched()
    w = make_worker("w1", cpu=2000, mem=2048, gpus=4, gpu="MI355X")
    await wr.register(w)
    req = ContainerRequest(container_id="c1", cpu=1500, memory_mb=1024, gpu="MI355X", gpu_count=2)
    s._normalize_gpu(req)
    assert await wr.schedule_container_request(w, req)
    cur = await wr.get("w1")
    assert cur.free_cpu == 500 and cur.free_gpu_count == 2 and cur.resource_version == 1
    # stale snapshot is rejected
    assert not await wr.schedule_container_request(w, req)
    # over-capacity rejected
    req2 = ContainerRequest(container_id="c2", cpu=1000, memory_mb=128)
    assert not await wr.schedule_container_request(cur, req2)
    await wr.release_capacity("w1", req)
    cur = await wr.get("w1")
    assert cur.free_cpu == 2000 and cur.free_gpu_count == 4

    # delivery-token ack
    queued = await store.blpop(keys.worker_request_queue("w1"), 0.1)
    assert queued["container_id"] == "c1"
    token = queued["delivery_token"]
    assert await wr.ack_container("w1", "c1", token)
    assert not await wr.ack_container("w1", "c1", token)  # one-shot


@pytest.mark.asyncio
async def test_end_to_end_backlog_scheduling():
    s, wr, cr, store = await make_sched()
    await wr.register(make_worker("w1"))
    await s.start()
    try:
        req = ContainerRequest(container_id="c-e2e", cpu=1000, memory_mb=256)
        await s.run(req)
        # should land on w1's queue
        d = await store.blpop(keys.worker_request_queue("w1"), 2.0)
        assert d is not None and d["container_id"] == "c-e2e"
        assert s.metrics.scheduled == 1
    finally:
        await s.shutdown()


@pytest.mark.asyncio
async def test_no_fit_provisions_and_retries():
    s, wr, cr, store = await make_sched()
    pool = FakePoolController()
    s.pools = {"default": pool}
    s.cfg.scheduler.retry_backoff_base_s = 0.01
    await s.start()
    try:
        req = ContainerRequest(container_id="c-prov", cpu=1000, memory_mb=256)
        await s.run(req)
        await asyncio.sleep(0.3)
        assert len(pool.added) >= 1
        assert s.metrics.retries >= 1
        # now a worker shows up; request gets scheduled on retry
        await wr.register(make_worker("late"))
        d = await store.blpop(keys.worker_request_queue("late"), 3.0)
        assert d is not None and d["container_id"] == "c-prov"
    finally:
        await s.shutdown()


@pytest.mark.asyncio
async def test_gpu_any_normalization():
    s, *_ = await make_sched()
    req = ContainerRequest(container_id="c", cpu=1, memory_mb=1, gpu="any")
    s._normalize_gpu(req)
    assert "MI355X" in req.gpu_request and "any" not in req.gpu_request
    assert req.gpu_count == 1


@pytest.mark.asyncio
async def test_gpu_failover_chain():
    s, wr, cr, _ = await make_sched()
    s.cfg.scheduler.gpu_failover = {"MI355X": ["MI300X"]}
    req = ContainerRequest(container_id="f1", cpu=1, memory_mb=1, gpu="MI355X", gpu_count=1)
    s._normalize_gpu(req)
    assert req.gpu_request == ["MI355X", "MI300X"]
    # an MI300X worker is now eligible, ranked behind MI355X
    mi300 = make_worker("b", gpus=8, gpu="MI300X")
    assert s._filter(req, [mi300]) == [mi300]


@pytest.mark.asyncio
async def test_reservation_prevents_duplicate_provisioning():
    """Retries of an already-provisioned request must not spawn duplicate
    workers (reference: reserve.go provisioningTracker)."""
    s, wr, cr, store = await make_sched()
    pool = FakePoolController()
    s.pools = {"default": pool}
    s.cfg.scheduler.retry_backoff_base_s = 0.01
    await s.start()
    try:
        req = ContainerRequest(container_id="c-res", cpu=1000, memory_mb=256)
        await s.run(req)
        await asyncio.sleep(0.5)  # several retry rounds
        assert s.metrics.retries >= 2
        assert len(pool.added) == 1  # one reservation, no duplicates
        await wr.register(make_worker("late-res"))
        d = await store.blpop(keys.worker_request_queue("late-res"), 3.0)
        assert d is not None
        assert "c-res" not in s._reservations  # cleared on commit
    finally:
        await s.shutdown()


@pytest.mark.asyncio
async def test_fractional_gpu_capacity():
    """Two half-GPU containers fit a 1-GPU worker; a whole-GPU request is
    then rejected until a share is released (Thunder-analog sharing —
    pkg/worker/thunder.go; capacity in fractional GPU units)."""
    s, wr, cr, store = await make_sched()
    w = make_worker("w1", gpus=1, gpu="MI355X")
    await wr.register(w)
    half = lambda cid: ContainerRequest(
        container_id=cid, cpu=100, memory_mb=128, gpu="MI355X", gpu_count=1,
        gpu_fraction=0.5,
    )
    r1, r2 = half("c1"), half("c2")
    s._normalize_gpu(r1); s._normalize_gpu(r2)
    assert r1.gpu_units() == 0.5
    assert await wr.schedule_container_request(await wr.get("w1"), r1)
    assert await wr.schedule_container_request(await wr.get("w1"), r2)
    cur = await wr.get("w1")
    assert cur.free_gpu_count == pytest.approx(0.0)
    # whole-GPU request no longer fits this worker
    whole = ContainerRequest(container_id="c3", cpu=100, memory_mb=128, gpu="MI355X", gpu_count=1)
    s._normalize_gpu(whole)
    assert s._filter(whole, [cur]) == []
    assert not await wr.schedule_container_request(cur, whole)
    # releasing one share frees half a unit — still not enough for a whole
    await wr.release_capacity("w1", r1)
    cur = await wr.get("w1")
    assert cur.free_gpu_count == pytest.approx(0.5)
    assert not await wr.schedule_container_request(cur, whole)
    # but enough for another half
    r4 = half("c4"); s._normalize_gpu(r4)
    assert await wr.schedule_container_request(cur, r4)
