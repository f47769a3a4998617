"""Python surface over the hand-written gfx950 kernels.

On a GPU host a missing extension is a hard error (no silent eager
fallback); on CPU-only hosts (dev container, CPU tests) the torch reference
path runs so unit tests exercise the same call sites.
"""

from __future__ import annotations

from typing import Optional

import torch

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if not _tried:
        _tried = True
        try:
            from . import _gemm_ext

            _ext = _gemm_ext
        except ImportError as e:
            if torch.cuda.is_available():
                raise RuntimeError(
                    f"gfx950 GEMM extension missing on a GPU host: {e}; "
                    "run __graft_entry__.build()"
                )
    return _ext


def has_native() -> bool:
    return _load() is not None


def gemm_nt(a: torch.Tensor, w: torch.Tensor, xcd_swizzle: bool = False) -> torch.Tensor:
    """C = a @ w.T — a [M,K] bf16, w [N,K] bf16 (torch Linear weight layout).

    Hand-written MFMA path for aligned shapes (M%128, N%128, K%64 == 0);
    torch fallback only on CPU."""
    ext = _load()
    if ext is not None and a.is_cuda:
        M, K = a.shape
        N = w.shape[0]
        if M % 128 == 0 and N % 128 == 0 and K % 64 == 0:
            return ext.gemm_bf16_nt(a.contiguous(), w.contiguous(), xcd_swizzle)
        # pad M up to 128 (weights stay aligned by construction)
        if N % 128 == 0 and K % 64 == 0:
            pad = (128 - M % 128) % 128
            a2 = torch.nn.functional.pad(a, (0, 0, 0, pad))
            return ext.gemm_bf16_nt(a2.contiguous(), w.contiguous(), xcd_swizzle)[:M]
        raise ValueError(f"unsupported GEMM shape M={M} N={N} K={K}")
    return a @ w.t()


def mlp_gate_up(a: torch.Tensor, w_gate: torch.Tensor, w_up: torch.Tensor) -> torch.Tensor:
    """Fused act = silu(a @ w_gate.T) * (a @ w_up.T) — one hand-written
    kernel; A staged once, no gate/up intermediates in HBM."""
    ext = _load()
    if ext is not None and a.is_cuda:
        M, K = a.shape
        N = w_gate.shape[0]
        if N % 128 == 0 and K % 64 == 0:
            if M % 128 != 0:
                pad = (128 - M % 128) % 128
                a2 = torch.nn.functional.pad(a, (0, 0, 0, pad))
                return ext.gemm_gate_up_silu(a2.contiguous(), w_gate.contiguous(), w_up.contiguous(), False)[:M]
            return ext.gemm_gate_up_silu(a.contiguous(), w_gate.contiguous(), w_up.contiguous(), False)
    return torch.nn.functional.silu(a @ w_gate.t()) * (a @ w_up.t())


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    ext = _load()
    if ext is not None and gate.is_cuda:
        return ext.silu_mul(gate.contiguous(), up.contiguous())
    return torch.nn.functional.silu(gate) * up


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    ext = _load()
    if ext is not None and x.is_cuda and x.shape[-1] % 8 == 0:
        shape = x.shape
        return ext.rmsnorm(x.reshape(-1, shape[-1]).contiguous(), w.contiguous(), eps).reshape(shape)
    xf = x.float()
    out = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (out * w.float()).to(x.dtype)
