"""Content-addressed page store.

Node-local tier of the distributed cache (reference: pkg/cache/storage.go —
4 MiB pages under bucketed dirs keyed by hash, per-page locks, complete
markers; storage_eviction.go — LRU by marker mtime, watermark 0.80,
hot-content guard). Image archives, hipsnap checkpoints and big volume
objects live here so 8 concurrent cold starts on one MI355X node read one
copy.
"""

from __future__ import annotations

import hashlib
import os
import shutil
import threading
import time
from typing import BinaryIO, Iterator, List, Optional


class ContentStore:
    def __init__(
        self,
        root: str,
        page_size: int = 4 * 1024 * 1024,
        quota_bytes: int = 200 * (1 << 30),
        eviction_watermark: float = 0.80,
        hot_guard_s: float = 600.0,
        memory_tier_bytes: int = 512 * (1 << 20),
        memory_max_item: int = 64 * (1 << 20),
    ):
        self.root = root
        self.page_size = page_size
        self.quota_bytes = quota_bytes
        self.eviction_watermark = eviction_watermark
        self.hot_guard_s = hot_guard_s
        os.makedirs(root, exist_ok=True)
        self._lock = threading.RLock()  # remove() runs under _maybe_evict's lock
        # memory tier (reference: pkg/cache disk+memory tiers,
        # storage.go): hot content cached in RAM, LRU-bounded; reads of
        # resident content never touch the disk
        self.memory_tier_bytes = memory_tier_bytes
        self.memory_max_item = memory_max_item
        from collections import OrderedDict

        self._mem: "OrderedDict[str, bytes]" = OrderedDict()
        self._mem_used = 0
        self.mem_hits = 0
        self.mem_misses = 0

    # -- paths -------------------------------------------------------------
    def _dir(self, digest: str) -> str:
        return os.path.join(self.root, digest[:2], digest[2:4], digest)

    def _marker(self, digest: str) -> str:
        return os.path.join(self._dir(digest), ".complete")

    def content_path(self, digest: str) -> str:
        return os.path.join(self._dir(digest), "content")

    # -- writes ------------------------------------------------------------
    def add_bytes(self, data: bytes) -> str:
        digest = hashlib.sha256(data).hexdigest()
        if self.has(digest):
            self._touch(digest)
            return digest
        d = self._dir(digest)
        os.makedirs(d, exist_ok=True)
        tmp = self.content_path(digest) + ".tmp"
        with open(tmp, "wb") as f:
            f.write(data)
        os.rename(tmp, self.content_path(digest))
        self._complete(digest, len(data))
        return digest

    def add_file(self, path: str, move: bool = False) -> str:
        h = hashlib.sha256()
        with open(path, "rb") as f:
            for chunk in iter(lambda: f.read(1 << 20), b""):
                h.update(chunk)
        digest = h.hexdigest()
        if self.has(digest):
            self._touch(digest)
            if move:
                os.unlink(path)
            return digest
        d = self._dir(digest)
        os.makedirs(d, exist_ok=True)
        tmp = self.content_path(digest) + ".tmp"
        if move:
            shutil.move(path, tmp)
        else:
            shutil.copyfile(path, tmp)
        os.rename(tmp, self.content_path(digest))
        self._complete(digest, os.path.getsize(self.content_path(digest)))
        return digest

    def add_stream(self, chunks: Iterator[bytes], expected_hash: str = "") -> str:
        h = hashlib.sha256()
        fd, tmp = _mkstemp(self.root)
        try:
            with os.fdopen(fd, "wb") as f:
                for chunk in chunks:
                    h.update(chunk)
                    f.write(chunk)
            digest = h.hexdigest()
            if expected_hash and digest != expected_hash:
                raise ValueError(f"hash mismatch: got {digest}, want {expected_hash}")
            if self.has(digest):
                os.unlink(tmp)
                self._touch(digest)
                return digest
            os.makedirs(self._dir(digest), exist_ok=True)
            os.rename(tmp, self.content_path(digest))
            self._complete(digest, os.path.getsize(self.content_path(digest)))
            return digest
        except Exception:
            try:
                os.unlink(tmp)
            except OSError:
                pass
            raise

    def _complete(self, digest: str, size: int) -> None:
        with open(self._marker(digest), "w") as f:
            f.write(str(size))
        self._maybe_evict()

    # -- reads ---------------------------------------------------------------
    def has(self, digest: str) -> bool:
        return os.path.exists(self._marker(digest))

    def size(self, digest: str) -> Optional[int]:
        try:
            with open(self._marker(digest)) as f:
                return int(f.read().strip() or 0)
        except OSError:
            return None

    def get_path(self, digest: str) -> Optional[str]:
        if not self.has(digest):
            return None
        self._touch(digest)
        return self.content_path(digest)

    def read(self, digest: str, offset: int = 0, length: int = 0) -> Optional[bytes]:
        with self._lock:
            data = self._mem.get(digest)
            if data is not None:
                self._mem.move_to_end(digest)  # LRU refresh
                self.mem_hits += 1
                if offset or length:
                    return data[offset: offset + length] if length else data[offset:]
                return data
        p = self.get_path(digest)
        if p is None:
            return None
        self.mem_misses += 1
        with open(p, "rb") as f:
            f.seek(offset)
            out = f.read(length) if length else f.read()
        if offset == 0 and not length:
            self._mem_promote(digest, out)
        return out

    def _mem_promote(self, digest: str, data: bytes) -> None:
        if len(data) > self.memory_max_item or self.memory_tier_bytes <= 0:
            return
        with self._lock:
            if digest in self._mem:
                return
            self._mem[digest] = data
            self._mem_used += len(data)
            while self._mem_used > self.memory_tier_bytes and self._mem:
                _, old = self._mem.popitem(last=False)
                self._mem_used -= len(old)

    def mem_stats(self) -> dict:
        return {
            "items": len(self._mem),
            "bytes": self._mem_used,
            "hits": self.mem_hits,
            "misses": self.mem_misses,
        }

    def _touch(self, digest: str) -> None:
        # refresh LRU stamp, throttled (reference: 5 min throttle)
        try:
            m = self._marker(digest)
            if time.time() - os.path.getmtime(m) > 300:
                os.utime(m)
        except OSError:
            pass

    # -- eviction --------------------------------------------------------------
    def used_bytes(self) -> int:
        total = 0
        for digest in self.list_digests():
            total += self.size(digest) or 0
        return total

    def list_digests(self) -> List[str]:
        out = []
        if not os.path.isdir(self.root):
            return out
        for a in os.listdir(self.root):
            pa = os.path.join(self.root, a)
            if not os.path.isdir(pa) or len(a) != 2:
                continue
            for b in os.listdir(pa):
                pb = os.path.join(pa, b)
                if not os.path.isdir(pb):
                    continue
                for digest in os.listdir(pb):
                    if os.path.exists(os.path.join(pb, digest, ".complete")):
                        out.append(digest)
        return out

    def _maybe_evict(self) -> None:
        with self._lock:
            used = self.used_bytes()
            limit = int(self.quota_bytes * self.eviction_watermark)
            if used <= limit:
                return
            entries = []
            now = time.time()
            for digest in self.list_digests():
                try:
                    mtime = os.path.getmtime(self._marker(digest))
                except OSError:
                    continue
                entries.append((mtime, digest))
            entries.sort()
            for mtime, digest in entries:
                if used <= limit:
                    break
                if now - mtime < self.hot_guard_s:
                    continue  # hot-content guard
                used -= self.size(digest) or 0
                self.remove(digest)

    def remove(self, digest: str) -> None:
        with self._lock:
            old = self._mem.pop(digest, None)
            if old is not None:
                self._mem_used -= len(old)
        shutil.rmtree(self._dir(digest), ignore_errors=True)


def _mkstemp(root: str):
    import tempfile

    return tempfile.mkstemp(dir=root, prefix=".ingest-")
