"""Single-GPU bf16 serving endpoint (BASELINE config 2 shape).

    b9 deploy examples/gpu_endpoint.py:infer --name llm-api
    curl -H "Authorization: Bearer $TOKEN" -X POST \
         http://gateway/deployment/llm-api/ -d '{"tokens": 256}'
"""

import beta9_amd
from beta9_amd.models import bench_handler


infer = beta9_amd.endpoint(
    gpu="MI355X",
    cpu=8000,
    memory=65536,
    keep_warm_seconds=60,
    checkpoint_enabled=True,  # hipsnap scale-to-zero resume
    concurrent_requests=32,
)(bench_handler.infer)
infer.on_start = bench_handler.load
