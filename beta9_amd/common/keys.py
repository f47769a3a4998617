"""Centralized state-store key registry.

Every state key used anywhere in the framework is minted here, mirroring the
reference's single key registry (reference: pkg/common/keys.go) so key shapes
are auditable in one place.
"""


# -- scheduler -------------------------------------------------------------
def scheduler_backlog() -> str:
    return "scheduler:container_requests"


def scheduler_request(container_id: str) -> str:
    return f"scheduler:request:{container_id}"


def scheduler_serve_lock(workspace_id: str, stub_id: str) -> str:
    return f"scheduler:serve:lock:{workspace_id}:{stub_id}"


def scheduler_checkpoint_state(workspace_id: str, stub_id: str) -> str:
    return f"scheduler:checkpoint_state:{workspace_id}:{stub_id}"


# -- workers ---------------------------------------------------------------
def worker_state(worker_id: str) -> str:
    return f"worker:state:{worker_id}"


def worker_keepalive(worker_id: str) -> str:
    return f"worker:keepalive:{worker_id}"


def worker_request_queue(worker_id: str) -> str:
    return f"worker:requests:{worker_id}"


def worker_pending_acks(worker_id: str) -> str:
    return f"worker:pending_acks:{worker_id}"


def worker_index() -> str:
    return "worker:index"


# -- containers ------------------------------------------------------------
def container_state(container_id: str) -> str:
    return f"container:state:{container_id}"


def container_address(container_id: str) -> str:
    return f"container:address:{container_id}"


def container_addresses_for_stub(stub_id: str) -> str:
    return f"container:addresses:{stub_id}"


def container_exit_code(container_id: str) -> str:
    return f"container:exit:{container_id}"


def container_logs(container_id: str) -> str:
    return f"container:logs:{container_id}"


def container_stop_requested(container_id: str) -> str:
    return f"container:stop:{container_id}"


def containers_by_stub(stub_id: str) -> str:
    return f"container:by_stub:{stub_id}"


def container_keep_warm(stub_id: str, container_id: str) -> str:
    return f"container:keepwarm:{stub_id}:{container_id}"


def container_concurrency(container_id: str) -> str:
    return f"container:tokens:{container_id}"


def container_lifecycle_spans(container_id: str) -> str:
    return f"container:spans:{container_id}"


# -- tasks -----------------------------------------------------------------
def task_state(task_id: str) -> str:
    return f"task:state:{task_id}"


def task_claim(task_id: str) -> str:
    return f"task:claim:{task_id}"


def task_heartbeat(task_id: str) -> str:
    return f"task:heartbeat:{task_id}"


def task_result(task_id: str) -> str:
    return f"task:result:{task_id}"


def task_args(task_id: str) -> str:
    return f"task:args:{task_id}"


def task_index_for_stub(stub_id: str) -> str:
    return f"task:index:{stub_id}"


def task_cancel(task_id: str) -> str:
    return f"task:cancel:{task_id}"


def taskqueue_list(workspace_id: str, stub_id: str) -> str:
    return f"taskqueue:{workspace_id}:{stub_id}"


# -- abstractions ----------------------------------------------------------
def map_key(workspace_id: str, name: str) -> str:
    return f"map:{workspace_id}:{name}"


def queue_key(workspace_id: str, name: str) -> str:
    return f"queue:{workspace_id}:{name}"


def signal_key(workspace_id: str, name: str) -> str:
    return f"signal:{workspace_id}:{name}"


def endpoint_inflight(stub_id: str) -> str:
    return f"endpoint:inflight:{stub_id}"


# -- concurrency limits ----------------------------------------------------
def concurrency_limit(workspace_id: str) -> str:
    return f"limits:concurrency:{workspace_id}"


# -- gpu -------------------------------------------------------------------
def gpu_assignments(worker_id: str) -> str:
    return f"gpu:assignments:{worker_id}"


def rendezvous_port(container_id: str) -> str:
    return f"gpu:rdzv:{container_id}"


# -- metrics / events ------------------------------------------------------
def metrics_counter(name: str) -> str:
    return f"metrics:counter:{name}"


def metrics_series(name: str) -> str:
    return f"metrics:series:{name}"


def events_stream() -> str:
    return "events:stream"


def workspace_usage(workspace_id: str) -> str:
    """Accumulated usage counters hash: task_count, task_duration_ms,
    cost_cents (reference: usage metrics repo counters)."""
    return f"usage:{workspace_id}"
