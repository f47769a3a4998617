"""Pod runner: arbitrary entrypoint with the sandbox PM as fallback.

Pods with a user entrypoint (POD_ENTRYPOINT) exec it; otherwise (and for
sandbox stubs) the process-manager server runs so the Sandbox surface works
(reference: worker swaps sandbox entrypoints to goproc —
pkg/worker/lifecycle.go:1690-1712).
"""

from __future__ import annotations

import asyncio
import logging
import os
import shlex
import sys

from .common import GPUBoot, RunnerEnv, setup_environment
from .sandbox import main as sandbox_main

log = logging.getLogger("b9.runner.pod")


def main() -> int:
    entry = os.environ.get("POD_ENTRYPOINT", "")
    if entry:
        env = RunnerEnv.from_env()
        setup_environment(env)
        gpu = GPUBoot(env)
        gpu.boot()
        argv = shlex.split(entry)
        os.execvp(argv[0], argv)
        return 127  # unreachable
    return asyncio.run(sandbox_main())


if __name__ == "__main__":
    logging.basicConfig(level=logging.INFO)
    sys.exit(main())
