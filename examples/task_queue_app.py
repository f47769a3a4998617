"""@task_queue with queue-depth autoscaling across 8 MI355X GPUs
(BASELINE config 4).

    b9 deploy examples/task_queue_app.py:work --name workers
"""

import beta9_amd


@beta9_amd.task_queue(
    gpu="MI355X",
    cpu=4000,
    memory=32768,
    autoscaler=beta9_amd.QueueDepthAutoscaler(max_containers=8, tasks_per_container=4),
)
def work(batch_id: int):
    from beta9_amd.models import bench_handler

    return bench_handler.infer(tokens=256) | {"batch_id": batch_id}
