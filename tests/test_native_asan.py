"""AddressSanitizer screens for the first-party native code (SURVEY §5.2,
VERDICT round-1 item #10): b9launch runs its full namespace / overlay /
mask / bind / netns / mini-init path under ASan; volume_cache.so is
ASan-built and loaded. Any heap/stack error aborts with a nonzero exit."""

import os
import subprocess

import pytest

from beta9_amd.worker.runtime.native import namespaces_available

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ASAN_LAUNCHER = os.path.join(REPO, "bin", "b9launch.asan")
ASAN_VOLCACHE = os.path.join(REPO, "bin", "volume_cache.asan.so")

needs_ns = pytest.mark.skipif(not namespaces_available(), reason="no namespace privileges")


def _build():
    if not (os.path.exists(ASAN_LAUNCHER) and os.path.exists(ASAN_VOLCACHE)):
        subprocess.run(
            ["make", "-C", os.path.join(REPO, "native"), "asan"],
            check=True, capture_output=True,
        )


_build()
ENV = {**os.environ, "ASAN_OPTIONS": "detect_leaks=0:abort_on_error=1"}


@needs_ns
def test_asan_launcher_full_path(tmp_path):
    vol = tmp_path / "vol"
    vol.mkdir()
    (vol / "f.txt").write_text("hi")
    masked = tmp_path / "secret"
    masked.mkdir()
    r = subprocess.run(
        [
            ASAN_LAUNCHER,
            "--rootfs-overlay", str(tmp_path / "ov"),
            "--hostname", "asan-test",
            "--cpu", "500", "--mem", "256",
            "--mask", str(masked),
            "--bind", f"{vol}:/data",
            "--env", "X=1",
            "--cwd", "/tmp",
            "--",
            "sh", "-c", "cat /data/f.txt; hostname; exit 7",
        ],
        env=ENV, capture_output=True, text=True, timeout=60,
    )
    assert "AddressSanitizer" not in r.stderr, r.stderr[-2000:]
    assert r.returncode == 7, (r.returncode, r.stderr[-500:])
    assert "hi" in r.stdout and "asan-test" in r.stdout


@needs_ns
def test_asan_launcher_signal_forwarding(tmp_path):
    import signal
    import time

    p = subprocess.Popen(
        [ASAN_LAUNCHER, "--rootfs-overlay", str(tmp_path / "ov2"), "--",
         "sleep", "30"],
        env=ENV, stderr=subprocess.PIPE, text=True,
    )
    time.sleep(1.0)
    p.send_signal(signal.SIGTERM)
    rc = p.wait(timeout=10)
    err = p.stderr.read()
    assert "AddressSanitizer" not in err, err[-2000:]
    assert rc == 143  # 128+SIGTERM through the mini-init


def test_asan_volcache_interposer(tmp_path):
    real = tmp_path / "real.txt"
    real.write_text("cached-bytes")
    vmap = f"{tmp_path}/virtual.txt={real}"
    # the asan runtime must be first in the preload chain when injecting
    # an asan-built .so into a non-asan binary
    libasan = subprocess.run(
        ["gcc", "-print-file-name=libasan.so"], capture_output=True, text=True
    ).stdout.strip()
    r = subprocess.run(
        ["cat", str(tmp_path / "virtual.txt")],
        env={**ENV, "LD_PRELOAD": f"{libasan}:{ASAN_VOLCACHE}",
             "VOLUME_CACHE_MAP": vmap},
        capture_output=True, text=True, timeout=30,
    )
    assert "AddressSanitizer" not in r.stderr, r.stderr[-2000:]
    assert r.returncode == 0 and "cached-bytes" in r.stdout
