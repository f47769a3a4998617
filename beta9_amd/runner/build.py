"""In-container image build runner.

Executes build steps inside the build container (reference:
pkg/abstractions/image/build.go:89-343 — v1 exec'd pip/shell steps in a
base-image container). The step list arrives as JSON in B9_BUILD_STEPS.

Modes (B9_BUILD_MODE):
- prefix: pip installs target $IMAGE_SITE, shell steps run in $PREFIX with
  PREFIX/IMAGE_SITE exported (the dirs are bind-mounted from the host build
  dir so the product survives the container).
- rootfs: steps mutate the container rootfs normally (pip installs into the
  image python's site-packages); the overlay upper is captured by the
  worker as the new layer.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys


def main() -> int:
    steps = json.loads(os.environ.get("B9_BUILD_STEPS", "[]"))
    mode = os.environ.get("B9_BUILD_MODE", "prefix")
    site = os.environ.get("IMAGE_SITE", "")
    prefix = os.environ.get("PREFIX", "")
    if mode == "prefix":
        os.makedirs(site, exist_ok=True)
        os.makedirs(prefix, exist_ok=True)
    for step in steps:
        if step.get("kind") == "pip":
            pkgs = step.get("packages", [])
            cmd = [sys.executable, "-m", "pip", "install", "--no-index", *pkgs]
            if mode == "prefix":
                cmd[4:4] = ["--target", site]
        else:
            cmd = ["bash", "-c", step.get("cmd", "true")]
        print(f"+ {' '.join(cmd)}", flush=True)
        cwd = prefix if (mode == "prefix" and prefix) else None
        env = dict(os.environ)
        r = subprocess.run(cmd, cwd=cwd, env=env)
        if r.returncode != 0:
            print(f"build step failed with exit code {r.returncode}", flush=True)
            return r.returncode
    print("build steps complete", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
