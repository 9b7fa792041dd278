from .harness import SimScheduler, mi355x_cluster_config  # noqa: F401
