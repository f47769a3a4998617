"""beta9_amd — MI355X-native serverless GPU runtime.

A from-scratch framework with the capabilities of beam-cloud/beta9 (Python
SDK with @endpoint/@function/@task_queue/Pod/Sandbox, gRPC gateway,
bin-packing scheduler, worker data plane) designed for AMD MI355X nodes:
KFD/amdgpu device plane, xGMI-topology-aware multi-GPU placement, RCCL
rendezvous provisioning, and a hand-written HIP cold-start path (HBM arena
pre-alloc + pinned weight streaming + GPU snapshot/restore).
"""

import os as _os

if _os.environ.get("B9_LIGHT_IMPORT"):
    # container runners set this: skip the SDK surface so interpreter
    # startup stays light on the 100-way-parallel cold-start path
    __all__ = []
else:
    from .sdk import (  # noqa: F401
        Client,
        ClientConfig,
        Endpoint,
        Function,
        Image,
        MapClient,
        Output,
        Pod,
        PodInstance,
        QueueClient,
        QueueDepthAutoscaler,
        AsyncSandbox,
        AsyncSandboxInstance,
        Sandbox,
        SandboxInstance,
        SandboxTerminal,
        SignalClient,
        TaskQueue,
        Volume,
        asgi,
        configure,
        endpoint,
        function,
        get_config,
        realtime,
        schedule,
        task_queue,
    )
    from .schema import Schema  # noqa: F401
    from .sdk import Bot, LLMConfig, LLMTokenPressureAutoscaler, PricingPolicy, Service  # noqa: F401
    from .runner.batching import batchable  # noqa: F401
    from .types import TaskPolicy  # noqa: F401

__version__ = "0.1.0"
