from .pod import (  # noqa: F401
    Pod,
    extract_pod_bind_info,
    extract_pod_scheduling_spec,
    is_bound,
    is_interested,
    is_node_healthy,
    new_binding_pod,
    pod_key,
)
from .types import (  # noqa: F401
    PodScheduleStatus,
    POD_WAITING,
    POD_PREEMPTING,
    POD_BINDING,
    POD_BOUND,
)
