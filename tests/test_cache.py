"""Content store + raw sendfile transport + cache service tests
(reference analog: pkg/cache/*_test.go)."""

import asyncio
import hashlib
import os

import pytest

from beta9_amd.cache.store import ContentStore
from beta9_amd.cache.transport import RawTransportClient, RawTransportServer


def test_store_roundtrip(tmp_path):
    s = ContentStore(str(tmp_path))
    data = os.urandom(1 << 20)
    digest = s.add_bytes(data)
    assert digest == hashlib.sha256(data).hexdigest()
    assert s.has(digest)
    assert s.size(digest) == len(data)
    assert s.read(digest) == data
    assert s.read(digest, offset=100, length=50) == data[100:150]
    # dedupe
    assert s.add_bytes(data) == digest


def test_store_stream_hash_verify(tmp_path):
    s = ContentStore(str(tmp_path))
    data = os.urandom(100_000)
    digest = hashlib.sha256(data).hexdigest()
    got = s.add_stream(iter([data[:50_000], data[50_000:]]), expected_hash=digest)
    assert got == digest
    with pytest.raises(ValueError):
        s.add_stream(iter([b"wrong"]), expected_hash=digest)


def test_store_file_and_eviction(tmp_path):
    s = ContentStore(str(tmp_path / "store"), quota_bytes=300_000, eviction_watermark=0.5, hot_guard_s=0.0)
    f = tmp_path / "blob.bin"
    digests = []
    for i in range(5):
        f.write_bytes(os.urandom(100_000))
        digests.append(s.add_file(str(f)))
    # quota 300k * 0.5 watermark = 150k -> at most 1-2 of the 100k blobs stay
    remaining = [d for d in digests if s.has(d)]
    assert len(remaining) <= 2
    assert s.used_bytes() <= 300_000


@pytest.mark.asyncio
async def test_raw_transport_sendfile(tmp_path):
    s = ContentStore(str(tmp_path))
    data = os.urandom(5 << 20)
    digest = s.add_bytes(data)
    server = RawTransportServer(s, host="127.0.0.1")
    port = await server.start()
    try:
        client = RawTransportClient("127.0.0.1", port)
        got = await client.fetch(digest)
        assert got == data
        part = await client.fetch(digest, offset=1024, length=2048)
        assert part == data[1024:3072]
        missing = await client.fetch("0" * 64)
        assert missing is None
        # chunked file pull
        dest = str(tmp_path / "out.bin")
        ok = await client.fetch_to_file(digest, dest, chunk=1 << 20)
        assert ok
        with open(dest, "rb") as f:
            assert f.read() == data
        await client.close()
    finally:
        await server.stop()


@pytest.mark.asyncio
async def test_cache_service_via_gateway():
    from tests.helpers import cluster

    async with cluster(n_workers=0) as c:
        import beta9_amd

        client = beta9_amd.Client()
        data = os.urandom(200_000)
        digest = hashlib.sha256(data).hexdigest()
        r = await asyncio.to_thread(
            client.client_stream,
            "CacheService",
            "StoreContent",
            [{"hash": digest, "data": data[:100_000]}, {"data": data[100_000:]}],
        )
        assert r["hash"] == digest
        h = await asyncio.to_thread(client.call, "CacheService", "HasContent", {"hash": digest})
        assert h["exists"] and h["size"] == len(data)
        assert h["transport_port"] > 0
        back = await asyncio.to_thread(client.call, "CacheService", "GetContent", {"hash": digest})
        assert back == data
        # raw transport serves the same content
        t = RawTransportClient("127.0.0.1", h["transport_port"])
        assert await t.fetch(digest) == data
        await t.close()


@pytest.mark.asyncio
async def test_reconciler_pulls_from_peer(tmp_path):
    """Node B's reconciler resolves a digest only node A has: host-map
    lease + RTT-ranked peer + raw-transport pull + hash verification
    (reference analog: pkg/cache hostmap + raw transport)."""
    from beta9_amd.cache.reconciler import CacheReconciler
    from beta9_amd.cache.service import CacheCoordinator
    from beta9_amd.cache.store import ContentStore
    from beta9_amd.cache.transport import RawTransportServer
    from beta9_amd.common.state import StateStore

    store = StateStore()
    coord = CacheCoordinator(store)

    a = ContentStore(str(tmp_path / "a"))
    b = ContentStore(str(tmp_path / "b"))
    payload = os.urandom(9 << 20)  # multi-chunk
    digest = a.add_bytes(payload)

    srv_a = RawTransportServer(a, host="127.0.0.1", port=0)
    await srv_a.start()
    srv_b = RawTransportServer(b, host="127.0.0.1", port=0)
    await srv_b.start()
    try:
        ra = CacheReconciler(a, coord, "node-a", "127.0.0.1", srv_a.port)
        rb = CacheReconciler(b, coord, "node-b", "127.0.0.1", srv_b.port)
        await ra.start()
        await rb.start()
        try:
            assert b.get_path(digest) is None
            path = await rb.ensure(digest)
            assert path is not None and b.has(digest)
            with open(path, "rb") as f:
                assert f.read() == payload
            # peers ranked, self excluded
            peers = await rb.ranked_peers()
            assert [p["host_id"] for p in peers] == ["node-a"]
            # unknown digest -> None (no peer has it)
            assert await rb.ensure("0" * 64) is None
        finally:
            await ra.stop()
            await rb.stop()
    finally:
        await srv_a.stop()
        await srv_b.stop()


@pytest.mark.asyncio
async def test_reconciler_skips_dead_peer(tmp_path):
    """A registered-but-dead peer is ranked out by the RTT probe; content
    still resolves from the live one."""
    from beta9_amd.cache.reconciler import CacheReconciler
    from beta9_amd.cache.service import CacheCoordinator
    from beta9_amd.cache.store import ContentStore
    from beta9_amd.cache.transport import RawTransportServer
    from beta9_amd.common.state import StateStore

    coord = CacheCoordinator(StateStore())
    a = ContentStore(str(tmp_path / "a"))
    b = ContentStore(str(tmp_path / "b"))
    digest = a.add_bytes(b"live-content" * 1000)
    srv_a = RawTransportServer(a, host="127.0.0.1", port=0)
    await srv_a.start()
    try:
        # dead peer: registered with a port nobody listens on
        await coord.register("dead", "127.0.0.1", 1)
        await coord.register("node-a", "127.0.0.1", srv_a.port)
        rb = CacheReconciler(b, coord, "node-b", "127.0.0.1", 0)
        peers = await rb.ranked_peers()
        assert [p["host_id"] for p in peers] == ["node-a"]
        assert await rb.ensure(digest) is not None
    finally:
        await srv_a.stop()


def test_memory_tier_lru(tmp_path):
    """Hot content serves from the RAM tier; LRU bounded by budget
    (reference: pkg/cache memory tier + eviction watermarks)."""
    from beta9_amd.cache.store import ContentStore

    s = ContentStore(str(tmp_path), memory_tier_bytes=3000, memory_max_item=1500)
    d1 = s.add_bytes(b"a" * 1000)
    d2 = s.add_bytes(b"b" * 1000)
    d3 = s.add_bytes(b"c" * 1000)
    big = s.add_bytes(b"X" * 2000)  # > memory_max_item: never promoted

    assert s.read(d1) == b"a" * 1000  # miss -> promote
    assert s.read(d1) == b"a" * 1000  # hit
    st = s.mem_stats()
    assert st["hits"] == 1 and st["items"] == 1

    s.read(d2); s.read(d3)           # fills to 3000
    s.read(big)                      # too big: not promoted
    assert s.mem_stats()["items"] == 3
    # a 4th small item evicts the LRU (d1)
    d4 = s.add_bytes(b"d" * 1000)
    s.read(d4)
    st = s.mem_stats()
    assert st["items"] == 3 and st["bytes"] <= 3000
    hits_before = st["hits"]
    s.read(d1)  # evicted -> miss again
    assert s.mem_stats()["hits"] == hits_before
    # ranged reads from memory
    s.read(d4)
    assert s.read(d4, offset=10, length=5) == b"d" * 5
    # removal drops the memory copy
    s.remove(d4)
    assert s.read(d4) is None
