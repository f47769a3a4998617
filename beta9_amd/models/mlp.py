"""Synthetic bf16 serving model for GPU endpoint benches.

A llama-style transformer MLP stack (rmsnorm -> gate/up projections ->
silu*mul -> down projection, residual) with random-init weights, running
entirely on the hand-written gfx950 kernels (beta9_amd/ops) with weights
resident in the hipsnap HBM arena — this is the "single-GPU bf16 handler"
of BASELINE.json config 2, and the payload whose cold-start weight
streaming the framework optimizes.

There is no network in the target environment, so weights are random-init
by design (synthetic handlers / random-init weights).
"""

from __future__ import annotations

import math
import threading
import time
from typing import Dict, List, Optional

import torch


from ..ops import gemm as ops


class MLPBlock:
    def __init__(self, hidden: int, inter: int, device: str, arena=None, prefix: str = ""):
        self.hidden = hidden
        self.inter = inter

        def make(name: str, shape):
            n = math.prod(shape)
            if arena is not None and arena.native:
                t = arena.tensor(f"{prefix}{name}", shape, dtype="bfloat16")
            else:
                t = torch.empty(shape, dtype=torch.bfloat16, device=device)
            with torch.no_grad():
                t.normal_(0, 0.02)
            return t

        # weights in torch Linear layout [out, in] = the GEMM's Bt operand
        self.w_gate = make("w_gate", (inter, hidden))
        self.w_up = make("w_up", (inter, hidden))
        self.w_down = make("w_down", (hidden, inter))
        self.norm_w = make("norm_w", (hidden,))
        self._fp8 = None  # lazily-built w8a8 gate/up/down weights
        self._act_scale = None  # calibrated static activation scale
        self._act2_scale = None  # calibrated silu-output scale (down input)
        self._init_lock = threading.Lock()

    def __getstate__(self):
        # objsnap pickles the live model object (checkpoint flow); locks
        # aren't picklable and are recreated on restore
        d = dict(self.__dict__)
        d.pop("_init_lock", None)
        return d

    def __setstate__(self, d):
        self.__dict__.update(d)
        self._init_lock = threading.Lock()

    def _fp8_weights(self):
        # double-checked: concurrent warmup requests run on per-thread
        # streams (infer), so quantize once under a lock and synchronize
        # BEFORE publishing — a reader that sees non-None must never see
        # half-written weights from another stream
        if self._fp8 is None:
            with self._init_lock:
                if self._fp8 is None:
                    gq, gs = ops.quantize_fp8_rowwise(self.w_gate)
                    uq, us = ops.quantize_fp8_rowwise(self.w_up)
                    dq, ds = ops.quantize_fp8_rowwise(self.w_down)
                    if self.w_gate.is_cuda:
                        torch.cuda.synchronize()
                    self._fp8 = (gq, gs, uq, us, dq, ds)
        return self._fp8

    def _calibrate(self, h: torch.Tensor) -> float:
        # static per-block activation scale from the first (calibration)
        # batch — standard w8a8 practice; keeps the per-call path free of
        # amax reductions
        if self._act_scale is None:
            amax = h.float().abs().amax().clamp(min=1e-8).item()
            self._act_scale = amax * 1.1 / ops.E4M3_MAX
        return self._act_scale

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        import os

        if os.environ.get("B9_GEMM_BACKEND", "native") == "fp8" and x.is_cuda:
            # w8a8 serving path: per-channel fp8 weights (quantized once),
            # STATIC calibrated activation scales, quantization fused into
            # producing kernels' epilogues (rmsnorm_fp8 / gate-up Q8), and
            # the fp8 down-projection with residual+dequant folded into its
            # split-K reduce — the steady-state block runs zero bf16 GEMMs
            # and zero eager quant kernels.
            gq, gs, uq, us, dq, ds = self._fp8_weights()
            if self._act_scale is None:
                self._calibrate(ops.rmsnorm(x, self.norm_w))  # one-time
            a_s = self._act_scale
            hq = ops.rmsnorm_fp8_out(x, self.norm_w, a_s)
            M = hq.shape[0]
            pad = (128 - M % 128) % 128
            if pad:
                import torch.nn.functional as F

                hq = F.pad(hq, (0, 0, 0, pad))
            if self._act2_scale is None:
                # calibration call: bf16 activation out, measure its scale
                act = ops.mlp_gate_up_fp8(hq, gq, uq, gs, us, a_s)
                amax = act.float().abs().amax().clamp(min=1e-8).item()
                self._act2_scale = amax * 1.1 / ops.E4M3_MAX
                act = act[:M] if pad else act
                return ops.gemm_nt_residual(act, self.w_down, x)
            # steady state: the WHOLE block is w8a8 — rmsnorm_fp8 ->
            # gate-up(Q8 epilogue) -> fp8 down + fused residual/dequant.
            # Zero bf16 GEMMs, zero eager quant kernels.
            a2 = self._act2_scale
            act_q = ops.mlp_gate_up_fp8_q8(hq, gq, uq, gs, us, a_s, a2)
            if pad:
                import torch.nn.functional as F

                xr = F.pad(x, (0, 0, 0, pad))
                return ops.mlp_down_fp8(act_q, dq, ds, a2, xr)[:M]
            return ops.mlp_down_fp8(act_q, dq, ds, a2, x)
        else:
            h = ops.rmsnorm(x, self.norm_w)
            # fused hand-written gate+up+silu (one A pass, no intermediates)
            act = ops.mlp_gate_up(h, self.w_gate, self.w_up)
        backend = os.environ.get("B9_GEMM_BACKEND", "native")
        if backend == "hybrid":
            # library GEMM + separate residual add (kept for A/B)
            return x + act @ self.w_down.t()
        # default: our split-K GEMM with the residual fused into the
        # epilogue — measured faster than hipBLASLt-plus-add at serving M
        return ops.gemm_nt_residual(act, self.w_down, x)


class SyntheticServingModel:
    """N MLP blocks; forward = one serving "inference" of the bench handler."""

    def __init__(
        self,
        hidden: int = 4096,
        inter: int = 14336,
        layers: int = 4,
        device: str = "cuda:0",
        use_arena: bool = True,
        arena=None,
    ):
        self.hidden = hidden
        self.inter = inter
        self.device = device
        need = layers * (2 * inter * hidden + hidden * inter + hidden) * 2
        if arena is not None and getattr(arena, "native", False):
            free = arena.size_bytes - arena._offset
            self.arena = arena if free >= int(need * 1.1) else None
        else:
            self.arena = None
        if self.arena is None and use_arena and torch.cuda.is_available():
            try:
                from ..ops import hipsnap

                nbytes = layers * (2 * inter * hidden + hidden * inter + hidden) * 2
                self.arena = hipsnap.Arena(
                    device=int(device.split(":")[-1]) if ":" in device else 0,
                    size_bytes=int(nbytes * 1.25) + (64 << 20),
                )
                if not self.arena.native:
                    self.arena = None
            except Exception:
                self.arena = None
        self.blocks = [
            MLPBlock(hidden, inter, device, arena=self.arena, prefix=f"l{i}.")
            for i in range(layers)
        ]

    @classmethod
    def from_arena(cls, arena, hidden: int, inter: int, layers: int, device: str = "cuda:0"):
        """Rebuild the model as views over a RESTORED hipsnap arena — the
        scale-to-zero resume path: no weight init, no weight load; the HBM
        snapshot already streamed in during GPU boot."""
        import torch

        self = cls.__new__(cls)
        self.hidden, self.inter, self.device, self.arena = hidden, inter, device, arena
        regions = {n: (o, s) for (n, o, s) in arena.regions()}

        def view(name, shape):
            off, size = regions[name]
            return arena._impl.tensor_view(off, size).view(torch.bfloat16).reshape(shape)

        self.blocks = []
        for i in range(layers):
            b = MLPBlock.__new__(MLPBlock)
            b.hidden, b.inter = hidden, inter
            b.w_gate = view(f"l{i}.w_gate", (inter, hidden))
            b.w_up = view(f"l{i}.w_up", (inter, hidden))
            b.w_down = view(f"l{i}.w_down", (hidden, inter))
            b.norm_w = view(f"l{i}.norm_w", (hidden,))
            b._fp8 = None
            b._act_scale = None
            b._act2_scale = None
            b._init_lock = threading.Lock()
            self.blocks.append(b)
        return self

    @property
    def param_bytes(self) -> int:
        return sum(
            b.w_gate.numel() + b.w_up.numel() + b.w_down.numel() + b.norm_w.numel()
            for b in self.blocks
        ) * 2

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        for b in self.blocks:
            x = b.forward(x)
        return x

    def infer(self, batch_tokens: int = 128) -> Dict:
        # Concurrent requests land here on runner threads; a per-thread HIP
        # stream lets their kernels overlap on the GPU (the default stream
        # would serialize them). Weights are read-only so cross-stream
        # sharing is safe; torch's caching allocator is stream-aware.
        stream = None
        if self.device.startswith("cuda"):
            from ..ops import streams as _streams

            stream = _streams.get_thread_stream()
        if stream is not None:
            with torch.cuda.stream(stream):
                # input created on the same stream its consumers run on
                x = torch.randn(batch_tokens, self.hidden, dtype=torch.bfloat16, device=self.device)
                t0 = time.perf_counter()
                y = self.forward(x)
            stream.synchronize()
        else:
            x = torch.randn(batch_tokens, self.hidden, dtype=torch.bfloat16, device=self.device)
            t0 = time.perf_counter()
            y = self.forward(x)
            if self.device.startswith("cuda"):
                torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        flops = 2 * batch_tokens * 3 * self.hidden * self.inter * len(self.blocks)
        return {
            "tokens": batch_tokens,
            "ms": dt * 1000,
            "tflops": flops / dt / 1e12,
            "checksum": float(y.float().abs().mean()),
        }
