"""Synthetic llama-style transformer block serving model.

Attention (grouped-query, torch SDPA — a plain library op) + the
hand-written MLP path (fused gate+up+SiLU, rmsnorm) with weights resident in
the hipsnap arena. A second serving workload family beside models/mlp.py:
bench.py --model transformer.
"""

from __future__ import annotations

import math
import threading
import time
from typing import Dict, Optional

import torch


from ..ops import gemm as ops


class TransformerBlock:
    def __init__(self, hidden: int, inter: int, n_heads: int, n_kv_heads: int, device: str, arena=None, prefix: str = ""):
        self.hidden = hidden
        self.inter = inter
        self.n_heads = n_heads
        self.n_kv_heads = n_kv_heads
        self.head_dim = hidden // n_heads

        def make(name, shape):
            if arena is not None and getattr(arena, "native", False):
                t = arena.tensor(f"{prefix}{name}", shape, dtype="bfloat16")
            else:
                t = torch.empty(shape, dtype=torch.bfloat16, device=device)
            with torch.no_grad():
                t.normal_(0, 0.02)
            return t

        kv_dim = self.head_dim * n_kv_heads
        self.wq = make("wq", (hidden, hidden))
        self.wk = make("wk", (kv_dim, hidden))
        self.wv = make("wv", (kv_dim, hidden))
        self.wo = make("wo", (hidden, hidden))
        self.w_gate = make("w_gate", (inter, hidden))
        self.w_up = make("w_up", (inter, hidden))
        self.w_down = make("w_down", (hidden, inter))
        self.attn_norm = make("attn_norm", (hidden,))
        self.mlp_norm = make("mlp_norm", (hidden,))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        tokens = x.shape[0]
        h = ops.rmsnorm(x, self.attn_norm)
        q = ops.gemm_nt(h, self.wq).view(tokens, self.n_heads, self.head_dim)
        k = (h @ self.wk.t()).view(tokens, self.n_kv_heads, self.head_dim)
        v = (h @ self.wv.t()).view(tokens, self.n_kv_heads, self.head_dim)
        rep = self.n_heads // self.n_kv_heads
        if rep > 1:
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        # [heads, tokens, dim] for SDPA (library attention; the MLP below is
        # the hand-written path)
        q, k, v = (t.transpose(0, 1) for t in (q, k, v))
        attn = torch.nn.functional.scaled_dot_product_attention(q, k, v, is_causal=True)
        attn = attn.transpose(0, 1).reshape(tokens, self.hidden)
        x = x + ops.gemm_nt(attn.contiguous(), self.wo)

        h = ops.rmsnorm(x, self.mlp_norm)
        act = ops.mlp_gate_up(h, self.w_gate, self.w_up)
        import os

        if os.environ.get("B9_GEMM_BACKEND", "hybrid") == "native":
            out = ops.gemm_nt(act, self.w_down)
        else:
            out = act @ self.w_down.t()
        return x + out


class TransformerServingModel:
    def __init__(
        self,
        hidden: int = 4096,
        inter: int = 14336,
        layers: int = 4,
        n_heads: int = 32,
        n_kv_heads: int = 8,
        device: str = "cuda:0",
        arena=None,
    ):
        self.hidden = hidden
        self.inter = inter
        self.device = device
        self.arena = arena
        self.blocks = [
            TransformerBlock(hidden, inter, n_heads, n_kv_heads, device, arena=arena, prefix=f"l{i}.")
            for i in range(layers)
        ]

    @property
    def param_bytes(self) -> int:
        total = 0
        for b in self.blocks:
            for w in (b.wq, b.wk, b.wv, b.wo, b.w_gate, b.w_up, b.w_down, b.attn_norm, b.mlp_norm):
                total += w.numel() * 2
        return total

    def infer(self, batch_tokens: int = 256) -> Dict:
        # per-thread HIP stream: concurrent serving requests overlap on the
        # GPU instead of serializing on the default stream (weights are
        # read-only; torch's allocator is stream-aware) — same pattern as
        # SyntheticServingModel.infer
        stream = None
        if self.device.startswith("cuda"):
            from ..ops import streams as _streams

            stream = _streams.get_thread_stream()
        if stream is not None:
            with torch.cuda.stream(stream):
                x = torch.randn(batch_tokens, self.hidden, dtype=torch.bfloat16, device=self.device)
                t0 = time.perf_counter()
                for b in self.blocks:
                    x = b.forward(x)
            stream.synchronize()
        else:
            x = torch.randn(batch_tokens, self.hidden, dtype=torch.bfloat16, device=self.device)
            t0 = time.perf_counter()
            for b in self.blocks:
                x = b.forward(x)
            if self.device.startswith("cuda"):
                torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        # flops: attention projections + sdpa + mlp
        L = len(self.blocks)
        proj = 2 * batch_tokens * self.hidden * (2 * self.hidden + 2 * self.blocks[0].head_dim * self.blocks[0].n_kv_heads)
        sdpa = 4 * batch_tokens * batch_tokens * self.hidden
        mlp = 2 * batch_tokens * 3 * self.hidden * self.inter
        flops = L * (proj + sdpa + mlp)
        return {
            "tokens": batch_tokens,
            "ms": dt * 1000,
            "tflops": flops / dt / 1e12,
            "checksum": float(x.float().abs().mean()),
        }
