"""Tensor-parallel bf16 MLP over RCCL/xGMI for gpu_count>1 endpoints.

Column-parallel gate/up projections, row-parallel down projection, one
bf16 all-reduce per block over xGMI (RCCL ring; 7 p2p links x ~153 GB/s per
MI355X GPU). One process per GPU (provisioned by runner/mgpu.py); each rank
keeps its weight shard in its own hipsnap arena, so multi-GPU checkpoints
are per-rank snapshots — the thing the reference disables outright
(reference: endpoint/instance.go:87-89 multi-GPU checkpointing off).
"""

from __future__ import annotations

import time
from typing import Dict

import torch
import torch.distributed as dist

from ..ops import gemm as ops


class TPMLPBlock:
    def __init__(self, hidden: int, inter: int, rank: int, world: int, device: str, arena=None, prefix: str = ""):
        assert inter % world == 0, "inter dim must divide world size"
        shard = inter // world
        self.hidden, self.shard = hidden, shard

        def make(name, shape):
            if arena is not None and getattr(arena, "native", False):
                t = arena.tensor(f"{prefix}{name}", shape, dtype="bfloat16")
            else:
                t = torch.empty(shape, dtype=torch.bfloat16, device=device)
            with torch.no_grad():
                # deterministic per-rank init so restores are verifiable
                t.normal_(0, 0.02)
            return t

        self.w_gate = make("w_gate", (shard, hidden))
        self.w_up = make("w_up", (shard, hidden))
        self.w_down = make("w_down", (hidden, shard))
        self.norm_w = make("norm_w", (hidden,))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        import os

        h = ops.rmsnorm(x, self.norm_w)
        act = ops.mlp_gate_up(h, self.w_gate, self.w_up)  # [tokens, shard]
        if os.environ.get("B9_GEMM_BACKEND", "hybrid") == "native":
            out = ops.gemm_nt(act, self.w_down)  # [tokens, hidden] partial
        else:
            out = act @ self.w_down.t()
        if dist.get_backend() == "gloo":
            # CPU test path: gloo has no bf16 reduction
            out32 = out.float()
            dist.all_reduce(out32)
            out = out32.to(out.dtype)
        else:
            dist.all_reduce(out)  # RCCL over xGMI
        return x + out


class TPServingModel:
    def __init__(self, hidden: int, inter: int, layers: int, rank: int, world: int, device: str = "cuda:0", arena=None):
        self.hidden, self.inter = hidden, inter
        self.rank, self.world = rank, world
        self.device = device
        torch.manual_seed(1234 + rank)
        self.blocks = [
            TPMLPBlock(hidden, inter, rank, world, device, arena=arena, prefix=f"l{i}.")
            for i in range(layers)
        ]
        self.arena = arena
        self._counter = 0

    @property
    def param_bytes(self) -> int:
        return sum(
            b.w_gate.numel() + b.w_up.numel() + b.w_down.numel() + b.norm_w.numel()
            for b in self.blocks
        ) * 2

    def infer(self, batch_tokens: int = 128) -> Dict:
        # all ranks run this in lockstep (broadcast dispatch); a shared
        # deterministic seed keeps x identical across ranks
        self._counter += 1
        gen = torch.Generator(device=self.device)
        gen.manual_seed(77_000 + self._counter)
        x = torch.randn(
            batch_tokens, self.hidden, generator=gen, dtype=torch.float32, device=self.device
        ).to(torch.bfloat16)
        t0 = time.perf_counter()
        for b in self.blocks:
            x = b.forward(x)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        flops = 2 * batch_tokens * 3 * self.hidden * self.inter * len(self.blocks)
        return {
            "tokens": batch_tokens,
            "ms": dt * 1000,
            "tflops_aggregate": flops / dt / 1e12,
            "world": self.world,
            "checksum": float(x.float().abs().mean()),
        }
