"""MI355X-native topology-aware gang scheduler (HiveD-capable, built from scratch).

Components:
- hivedcore (C++): cell model, buddy allocation, topology-aware placement,
  VC-safety accounting, preemption state machine
- api: YAML config + pod-annotation wire formats (HiveD-compatible)
- scheduler/webserver: K8s scheduler-extender bridge (filter/bind/preempt)
- topo: CDNA4 topology (MI355X -> xGMI pair -> quad -> 8-GPU node) + rocm-smi
  discovery
- probe: RCCL-over-xGMI all-reduce placement probe
- ops: HIP (gfx950) GPU health-check kernels
"""

__version__ = "0.1.0"


def _ensure_core():
    try:
        from . import hivedcore  # noqa: F401
    except ImportError:
        from .core import build as _build

        _build.build()


_ensure_core()

from . import hivedcore  # noqa: E402,F401
from .algorithm import HivedAlgorithm  # noqa: E402,F401
