from .allreduce import allreduce_probe, busbw_from_algbw, CellProbeRunner  # noqa: F401
