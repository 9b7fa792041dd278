from .health import NodeHealthAgent, collect_node_health  # noqa: F401
