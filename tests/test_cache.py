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
