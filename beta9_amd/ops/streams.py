"""Per-thread HIP stream pool for concurrent serving requests.

Each concurrent request runs its kernels on its own HIP stream so they
overlap on the GPU (the default stream serializes). The FIRST launch on a
new stream pays lazy HSA hardware-queue creation (~ms), which would land
in the first serving request after a cold start — so the prewarm shell
pre-creates a small pool and pre-launches on each (HIP multiplexes
streams onto GPU_MAX_HW_QUEUES hardware queues, so a small pool covers
the queue-creation cost for any number of threads).
"""

from __future__ import annotations

import threading
from typing import List

_pool: List = []
_tls = threading.local()


def prewarm(n: int = 8, device: str = "cuda:0") -> int:
    """Create n streams and force their hardware-queue setup; called from
    the prewarm shell warm set (runner/prewarm.py)."""
    import torch

    if not torch.cuda.is_available():
        return 0
    made = 0
    for _ in range(n):
        s = torch.cuda.Stream(device=device)
        with torch.cuda.stream(s):
            torch.zeros(1, device=device)
        _pool.append(s)
        made += 1
    torch.cuda.synchronize()
    return made


def get_thread_stream():
    """This thread's serving stream: reuse a prewarmed one when available.
    Returns None without a GPU."""
    import torch

    if not torch.cuda.is_available():
        return None
    s = getattr(_tls, "stream", None)
    if s is None:
        try:
            s = _pool.pop()
        except IndexError:
            s = torch.cuda.Stream()
        _tls.stream = s
    return s
