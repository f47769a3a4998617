"""Fixture user handlers for e2e tests (reference analog:
sdk/tests/fixtures/customer_apps)."""

import os
import time


def hello(x=1):
    return {"hello": x * 2}


def add(a, b):
    return a + b


def envdump():
    return {
        "container_id": os.environ.get("CONTAINER_ID", ""),
        "stub_type": os.environ.get("STUB_TYPE", ""),
        "secret": os.environ.get("TEST_SECRET", ""),
    }


def slow(seconds=1.0):
    time.sleep(float(seconds))
    return "done"


def boom():
    raise ValueError("expected failure")


def echo(**kwargs):
    return kwargs


_started = {"on_start_ran": False}


def on_start():
    _started["on_start_ran"] = True
    return "warm"


def check_started(**kwargs):
    return _started["on_start_ran"]


def image_files_env():
    import os

    return os.environ.get("B9_IMAGE_FILES", "")


def bot_answer(markers=None, **kw):
    loc = next(iter(markers))
    q = markers[loc][0].get("q", "?")
    # answers go to whatever single output the net wires up
    return {"markers": {"answers": [{"text": f"answer:{q}"}],
                        "outbox": [{"text": f"answer:{q}"}]}}


def bot_summarize(markers=None, **kw):
    texts = [m["text"] for m in markers.get("answers", [])]
    return {"markers": {"summaries": [{"text": " | ".join(sorted(texts))}]}}


def batched_echo(batch=None, **kw):
    out = []
    for args in batch:
        pos = args.get("args") or []
        kws = args.get("kwargs") or {}
        x = pos[0] if pos else kws.get("x", args.get("x"))
        out.append({"x": x, "batch_size": len(batch)})
    return out


batched_echo._b9_batch = {"max_size": 16, "wait_ms": 20.0}


def probe_paths(paths=None, **kw):
    """Existence probe used by isolation tests."""
    import os as _os

    return {p: _os.path.exists(p) for p in (paths or [])}


def pidns_probe(**kw):
    """PID-namespace + overlay-write probe used by isolation tests."""
    import os as _os

    with open("/isolation-canary", "w") as f:
        f.write("x")
    return {"pid": _os.getpid(), "canary": _os.path.exists("/isolation-canary")}


def gpu_envdump():
    return {
        "visible": os.environ.get("HIP_VISIBLE_DEVICES", ""),
        "fraction": os.environ.get("B9_GPU_MEMORY_FRACTION", ""),
    }


def frac_serving_steps(steps=100, tokens=256, hidden=2048, inter=4096, layers=4):
    """GPU serving burst for the fractional-sharing demo: N inference steps
    of the synthetic serving model, returns per-container throughput."""
    import time as _time

    import torch

    from beta9_amd.models.mlp import SyntheticServingModel

    m = SyntheticServingModel(int(hidden), int(inter), int(layers), "cuda:0")
    for _ in range(3):
        m.infer(batch_tokens=int(tokens))
    torch.cuda.synchronize()
    t0 = _time.perf_counter()
    for _ in range(int(steps)):
        m.infer(batch_tokens=int(tokens))
    torch.cuda.synchronize()
    dt = _time.perf_counter() - t0
    return {
        "steps_per_s": int(steps) / dt,
        "ms_per_step": dt / int(steps) * 1000,
        "visible": __import__("os").environ.get("HIP_VISIBLE_DEVICES", ""),
        "fraction": __import__("os").environ.get("B9_GPU_MEMORY_FRACTION", ""),
    }
