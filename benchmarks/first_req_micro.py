#!/usr/bin/env python3
"""Fresh-box decomposition of the first serving request: what pays the
~145 ms after the runner is healthy (benchmarks/cold_trace.py)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def t(name, fn):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    r = fn()
    torch.cuda.synchronize()
    print(f"{(time.perf_counter()-t0)*1000:9.2f} ms  {name}", flush=True)
    return r

# ---- shell-warm equivalent ----
t("cuda init + ctx", lambda: torch.zeros(1, device="cuda:0"))
x256 = torch.randn(256, 256, dtype=torch.bfloat16, device="cuda:0")
t("hipBLASLt warm 256^3", lambda: x256 @ x256.t())
from beta9_amd.ops import gemm as g
t("warm_all kernels", lambda: g.warm_all())
from beta9_amd.ops import hipsnap
arena = t("prewarm arena", lambda: hipsnap.Arena(device=0, size_bytes=4 << 30))

# ---- what the first request pays ----
from beta9_amd.models.mlp import SyntheticServingModel
model = t("model build (arena alloc + normal_ queued)",
          lambda: SyntheticServingModel(4096, 14336, 8, "cuda:0", arena=arena))
t("drain init kernels (sync)", lambda: None)
xa = torch.randn(64, 14336, dtype=torch.bfloat16, device="cuda:0")
wd = torch.randn(4096, 14336, dtype=torch.bfloat16, device="cuda:0")
t("hipBLASLt FIRST 64x4096x14336", lambda: xa @ wd.t())
t("hipBLASLt second same shape", lambda: xa @ wd.t())
h = torch.randn(64, 4096, dtype=torch.bfloat16, device="cuda:0")
t("our mlp_gate_up FIRST 64->pad128", lambda: g.mlp_gate_up(h, model.blocks[0].w_gate, model.blocks[0].w_up))
t("first full infer", lambda: model.infer(64))
t("first infer 2048 tokens", lambda: model.infer(2048))
t("second infer", lambda: model.infer(64))
