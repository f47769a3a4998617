"""gpu_count=8 endpoint with RCCL-over-xGMI rendezvous (BASELINE config 5).

The platform provisions one process per GPU with a loopback RCCL
rendezvous; the handler runs tensor-parallel with plain torch.distributed
collectives (see beta9_amd/models/mlp_tp.py).

    b9 deploy examples/multi_gpu_endpoint.py:infer --name tp8-api
"""

import beta9_amd
from beta9_amd.models import bench_handler


infer = beta9_amd.endpoint(
    gpu="MI355X",
    gpu_count=8,
    cpu=32000,
    memory=262144,
    keep_warm_seconds=120,
    concurrent_requests=16,
)(bench_handler.infer)
infer.on_start = bench_handler.load
