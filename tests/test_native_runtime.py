"""b9launch native runtime + volume_cache LD_PRELOAD shim tests (skipped on
hosts without namespace privileges; the launcher binary builds everywhere)."""

import os
import subprocess
import sys

import pytest

from beta9_amd.worker.runtime.base import ContainerSpec
from beta9_amd.worker.runtime.native import (
    LAUNCHER,
    NativeRuntime,
    launcher_available,
    namespaces_available,
)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
VOLCACHE = os.path.join(REPO, "bin", "volume_cache.so")


def _build_native():
    if not (launcher_available() and os.path.exists(VOLCACHE)):
        subprocess.run(["make", "-C", os.path.join(REPO, "native")], check=True, capture_output=True)


_build_native()

needs_ns = pytest.mark.skipif(not namespaces_available(), reason="no namespace privileges")


@needs_ns
@pytest.mark.asyncio
async def test_native_runtime_isolation(tmp_path):
    rt = NativeRuntime(scratch_path=str(tmp_path))
    lines = []
    spec = ContainerSpec(
        container_id="nat-1",
        entry_point=[
            "sh",
            "-c",
            "hostname; echo $$; touch /isolated-write; ls /dev | wc -l",
        ],
        hostname="nat-host",
        env={},
    )
    await rt.run(spec, on_log=lambda s, l: lines.append(l))
    code = await rt.wait("nat-1")
    await rt.delete("nat-1")
    assert code == 0
    assert lines[0] == "nat-host"  # UTS namespace
    assert lines[1] == "2"  # PID namespace: b9launch mini-init is 1, payload is 2
    assert not os.path.exists("/isolated-write")  # overlay upper, not host
    assert int(lines[2]) < 15  # minimal /dev


@needs_ns
@pytest.mark.asyncio
async def test_native_runtime_bind_mount(tmp_path):
    data_dir = tmp_path / "vol"
    data_dir.mkdir()
    (data_dir / "f.txt").write_text("vol-data")
    rt = NativeRuntime(scratch_path=str(tmp_path / "scratch"))
    lines = []
    spec = ContainerSpec(
        container_id="nat-2",
        entry_point=["sh", "-c", "cat /mnt/vol/f.txt; echo w > /mnt/vol/out.txt"],
        mounts=[{"local_path": str(data_dir), "mount_path": "/mnt/vol"}],
    )
    await rt.run(spec, on_log=lambda s, l: lines.append(l))
    assert await rt.wait("nat-2") == 0
    await rt.delete("nat-2")
    assert lines[0] == "vol-data"
    # writes to the bind persist on the host (unlike overlay writes)
    assert (data_dir / "out.txt").read_text().strip() == "w"


@needs_ns
@pytest.mark.asyncio
async def test_native_runtime_block_network(tmp_path):
    rt = NativeRuntime(scratch_path=str(tmp_path))
    lines = []
    spec = ContainerSpec(
        container_id="nat-3",
        entry_point=["sh", "-c", "ls /sys/class/net 2>/dev/null | wc -l; ip link 2>/dev/null | grep -c UP || true"],
        block_network=True,
    )
    await rt.run(spec, on_log=lambda s, l: lines.append(l))
    assert await rt.wait("nat-3") == 0
    await rt.delete("nat-3")


def test_volume_cache_preload(tmp_path):
    real = tmp_path / "real"
    cache = tmp_path / "cache"
    real.mkdir()
    cache.mkdir()
    (real / "a.txt").write_text("real")
    (cache / "a.txt").write_text("cached")
    (real / "miss.txt").write_text("only-real")
    env = dict(os.environ)
    env["LD_PRELOAD"] = VOLCACHE
    env["VOLUME_CACHE_MAP"] = f"{real}={cache}"
    r = subprocess.run(
        [sys.executable, "-c", f"print(open('{real}/a.txt').read() + open('{real}/miss.txt').read())"],
        env=env,
        capture_output=True,
        text=True,
    )
    assert r.returncode == 0, r.stderr
    assert "cached" in r.stdout and "only-real" in r.stdout

    # writes go to the real path, never the cache
    r = subprocess.run(
        ["sh", "-c", f"echo new > {real}/w.txt && cat {real}/w.txt"],
        env=env,
        capture_output=True,
        text=True,
    )
    assert r.returncode == 0
    assert (real / "w.txt").exists() and not (cache / "w.txt").exists()
