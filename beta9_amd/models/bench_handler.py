"""Bench endpoint handler: the GPU bf16 serving workload dispatched through
the full platform path (gateway -> scheduler -> worker -> container -> this
module). Importable by the endpoint runner as
`beta9_amd.models.bench_handler:infer` with `:load` as on_start.
"""

from __future__ import annotations

import os
import time
from typing import Any, Dict, Optional

_model = None
_load_info: Dict[str, Any] = {}


def load() -> Dict[str, Any]:
    """on_start: build the model (random-init weights into the HBM arena).
    The time spent here IS the cold-start weight phase the platform
    overlaps/snapshots."""
    global _model, _load_info
    if _model is not None:
        return _load_info
    t0 = time.perf_counter()
    import torch

    hidden = int(os.environ.get("B9_BENCH_HIDDEN", "4096"))
    inter = int(os.environ.get("B9_BENCH_INTER", "14336"))
    layers = int(os.environ.get("B9_BENCH_LAYERS", "8"))
    world = int(os.environ.get("WORLD_SIZE", "1") or 1)
    if world > 1:
        # tensor-parallel over the container's RCCL rendezvous (mgpu runner)
        from ..runner.common import get_gpu_boot
        from ..runner.mgpu import get_rendezvous
        from .mlp_tp import TPServingModel

        rdzv = get_rendezvous()
        assert rdzv is not None, "WORLD_SIZE>1 but no rendezvous initialized"
        gb = get_gpu_boot()
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
        _model = TPServingModel(
            hidden=hidden, inter=inter, layers=layers,
            rank=rdzv.rank, world=rdzv.world, device=device,
            arena=(gb.arena if gb is not None else None),
        )
        _load_info = {
            "load_ms": (time.perf_counter() - t0) * 1000,
            "native": torch.cuda.is_available(),
            "param_bytes": _model.param_bytes,
            "restored": False,
            "world": world,
            "backend": rdzv.backend,
        }
        return _load_info
    model_kind = os.environ.get("B9_BENCH_MODEL", "mlp")
    # objsnap fast path: the model OBJECT came back with the checkpoint —
    # no rebuild, no on_start work at all
    from ..runner.common import get_checkpoint_state, register_checkpoint_state

    restored_model = get_checkpoint_state("bench_model")
    if restored_model is not None:
        _model = restored_model
        _load_info = {
            "load_ms": (time.perf_counter() - t0) * 1000,
            "native": True,
            "param_bytes": getattr(_model, "param_bytes", 0),
            "restored": True,
            "object_restored": True,
        }
        return _load_info
    if torch.cuda.is_available():
        from ..runner.common import get_gpu_boot
        from .mlp import SyntheticServingModel

        gb = get_gpu_boot()
        restored = False
        if model_kind == "transformer":
            from .transformer import TransformerServingModel

            _model = TransformerServingModel(
                hidden=hidden, inter=inter, layers=layers, device="cuda:0",
                arena=(gb.arena if gb is not None else None),
            )
        elif gb is not None and gb.restored and gb.arena is not None:
            # scale-to-zero resume: weights already streamed into HBM from
            # the hipsnap snapshot during GPU boot — just remap views.
            _model = SyntheticServingModel.from_arena(gb.arena, hidden, inter, layers)
            restored = True
        else:
            _model = SyntheticServingModel(
                hidden=hidden, inter=inter, layers=layers, device="cuda:0",
                arena=(gb.arena if gb is not None else None),
            )
        native = True
        param_bytes = _model.param_bytes
    else:
        # CPU dev-path: tiny torch model so the full dispatch loop still runs
        _model = _CPUModel(hidden=min(hidden, 256), layers=2)
        native = False
        param_bytes = _model.param_bytes
    from ..runner.common import get_gpu_boot as _ggb

    gb = _ggb()
    _load_info = {
        "load_ms": (time.perf_counter() - t0) * 1000,
        "native": native,
        "param_bytes": param_bytes,
        "restored": bool(gb is not None and gb.restored),
        "gpu_phases": dict(gb.phases) if gb is not None else {},
    }
    register_checkpoint_state("bench_model", _model)
    return _load_info


class _CPUModel:
    def __init__(self, hidden: int = 256, layers: int = 2):
        import torch

        self.ws = [torch.randn(hidden, hidden) * 0.02 for _ in range(layers)]
        self.param_bytes = sum(w.numel() * 4 for w in self.ws)
        self.hidden = hidden

    def infer(self, batch_tokens: int = 16) -> Dict:
        import torch

        x = torch.randn(min(batch_tokens, 32), self.hidden)
        t0 = time.perf_counter()
        for w in self.ws:
            x = torch.relu(x @ w)
        return {"ms": (time.perf_counter() - t0) * 1000, "tflops": 0.0, "checksum": float(x.abs().mean())}


def infer(tokens: int = 256, **kwargs) -> Dict[str, Any]:
    if _model is None:
        load()
    r = _model.infer(batch_tokens=int(tokens))
    r["load"] = _load_info
    return r


def infer_batched(batch=None, **kw):
    """Dynamic-batching variant: one model forward serves every coalesced
    request (total tokens = sum of per-request tokens; no work skipped).
    Used by `bench.py --dyn-batch`."""
    load()
    import torch

    counts = [int((a or {}).get("tokens", 256) or 256) for a in (batch or [{}])]
    total = sum(counts)
    r = _model.infer(total)
    out = []
    for c in counts:
        out.append({"tokens": c, "ms": r["ms"], "tflops": r["tflops"],
                    "checksum": r["checksum"], "batched": len(counts)})
    return out


infer_batched._b9_batch = {"max_size": 64, "wait_ms": 2.0}
