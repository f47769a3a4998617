import asyncio

import pytest

from beta9_amd.common.state import Lock, StateStore


@pytest.mark.asyncio
async def test_kv_ttl_and_events():
    s = StateStore()
    await s.start()
    try:
        await s.set("a", 1)
        assert await s.get("a") == 1
        await s.set("b", "x", ttl=0.05)
        assert await s.get("b") == "x"
        await asyncio.sleep(0.15)
        assert await s.get("b") is None

        sub = await s.subscribe("watch:*")
        await s.set("watch:1", 5)
        ev = await sub.get(timeout=1.0)
        assert ev == ("set", "watch:1")
        sub.close()
    finally:
        await s.stop()


@pytest.mark.asyncio
async def test_set_nx_lock():
    s = StateStore()
    assert await s.set_nx("k", 1)
    assert not await s.set_nx("k", 2)
    lock = Lock(s, "x", ttl=5.0)
    assert await lock.acquire()
    lock2 = Lock(s, "x", ttl=5.0)
    assert not await lock2.acquire(timeout=0.05)
    await lock.release()
    assert await lock2.acquire()


@pytest.mark.asyncio
async def test_hash_and_zset():
    s = StateStore()
    await s.hset("h", {"a": 1, "b": 2})
    assert await s.hget("h", "a") == 1
    assert await s.hgetall("h") == {"a": 1, "b": 2}
    await s.hdel("h", "a")
    assert await s.hget("h", "a") is None

    await s.zadd("z", "m1", 3.0)
    await s.zadd("z", "m2", 1.0)
    await s.zadd("z", "m3", 2.0)
    assert await s.zcard("z") == 3
    popped = await s.zpop_min("z", 2)
    assert [m for m, _ in popped] == ["m2", "m3"]
    # re-add with new score updates
    await s.zadd("z", "m1", 0.5)
    popped = await s.zpop_min("z", 1)
    assert popped[0][0] == "m1" and popped[0][1] == 0.5


@pytest.mark.asyncio
async def test_list_blocking_pop():
    s = StateStore()
    assert await s.blpop("q", timeout=0.05) is None

    async def pusher():
        await asyncio.sleep(0.05)
        await s.rpush("q", "item")

    t = asyncio.create_task(pusher())
    v = await s.blpop("q", timeout=2.0)
    assert v == "item"
    await t


@pytest.mark.asyncio
async def test_keys_pattern():
    s = StateStore()
    await s.set("task:state:1", 1)
    await s.set("task:state:2", 1)
    await s.set("other", 1)
    ks = await s.keys("task:state:*")
    assert sorted(ks) == ["task:state:1", "task:state:2"]


@pytest.mark.asyncio
async def test_otlp_span_export(tmp_path, monkeypatch):
    """Lifecycle spans export in OTLP/JSON wire shape (SURVEY §5.1 — the
    reference ships OTel to Tempo; we emit collector-ingestible lines)."""
    import json

    from beta9_amd.backend.repos import ContainerRepo
    from beta9_amd.common import otlp
    from beta9_amd.common.state import StateStore

    path = str(tmp_path / "traces.jsonl")
    monkeypatch.setattr(otlp, "_exporter", None)
    monkeypatch.setenv("B9_OTLP_PATH", path)
    store = StateStore()
    repo = ContainerRepo(store)
    await repo.push_span("c-otlp", {"name": "worker.cold_start", "start": 1.0,
                                    "end": 1.5, "duration_ms": 500.0})
    await repo.push_span("c-otlp", {"name": "worker.runtime_start", "start": 1.1,
                                    "end": 1.2, "duration_ms": 100.0})
    lines = [json.loads(l) for l in open(path)]
    assert len(lines) == 2
    span = lines[0]["resourceSpans"][0]["scopeSpans"][0]["spans"][0]
    assert span["name"] == "worker.cold_start"
    assert span["startTimeUnixNano"] == str(int(1.0 * 1e9))
    t1 = lines[0]["resourceSpans"][0]["scopeSpans"][0]["spans"][0]["traceId"]
    t2 = lines[1]["resourceSpans"][0]["scopeSpans"][0]["spans"][0]["traceId"]
    assert t1 == t2  # one trace per container
    rs = lines[0]["resourceSpans"][0]["resource"]["attributes"][0]
    assert rs == {"key": "service.name", "value": {"stringValue": "beta9-amd"}}
    otlp._exporter.close()
    monkeypatch.setattr(otlp, "_exporter", None)
