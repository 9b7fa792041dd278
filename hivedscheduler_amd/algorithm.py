"""Python facade over the C++ scheduling core.

Mirrors the reference's internal.SchedulerAlgorithm contract
(pkg/internal/types.go:57-100): Schedule, Add/Update/DeleteNode,
Add/DeleteUnallocatedPod, Add/DeleteAllocatedPod plus inspect getters.
The sequencing contract is the caller's job (HivedScheduler serializes calls).
"""
from __future__ import annotations

import threading
from typing import Dict, List, Optional

from .api import constants
from .api.types import (
    Config,
    PodBindInfo,
    PodSchedulingSpec,
    WebServerError,
    dataclass_to_plain,
)

FILTERING = "Filtering"
PREEMPTING = "Preempting"


def normalize_spec(config: Config) -> dict:
    """Convert a parsed (and inferred) Config into the C++ core's spec dict."""
    pc = config.physicalCluster
    return {
        "cellTypes": {
            name: {
                "childCellType": ct.childCellType or "",
                "childCellNumber": ct.childCellNumber,
                "isNodeLevel": ct.isNodeLevel,
            }
            for name, ct in pc.cellTypes.items()
        },
        "physicalCells": [c.to_dict() for c in pc.physicalCells],
        "virtualClusters": {
            vc: {
                "virtualCells": [
                    {"cellType": v.cellType, "cellNumber": v.cellNumber} for v in spec.virtualCells
                ],
                "pinnedCells": [{"pinnedCellId": p.pinnedCellId} for p in spec.pinnedCells],
            }
            for vc, spec in config.virtualClusters.items()
        },
    }


class ScheduleResult:
    """One of bind / preempt / wait."""

    def __init__(self, raw: dict):
        self.kind: str = raw["kind"]
        self.bind_info: Optional[PodBindInfo] = None
        self.victim_node: str = ""
        self.victim_pod_keys: List[str] = []
        self.wait_reason: str = ""
        if self.kind == "bind":
            self.bind_info = PodBindInfo.from_dict(raw["bindInfo"])
        elif self.kind == "preempt":
            self.victim_node = raw.get("victimNode", "")
            self.victim_pod_keys = list(raw.get("victimPodKeys", []))
        else:
            self.wait_reason = raw.get("reason", "")

    def __repr__(self) -> str:  # pragma: no cover
        if self.kind == "bind":
            return f"ScheduleResult(bind node={self.bind_info.node} cells={self.bind_info.leafCellIsolation})"
        if self.kind == "preempt":
            return f"ScheduleResult(preempt victims={self.victim_pod_keys})"
        return f"ScheduleResult(wait reason={self.wait_reason!r})"


class HivedAlgorithm:
    """Thread-safe wrapper: a single lock serializes all calls, matching the
    reference's algorithmLock (hived_algorithm.go:104)."""

    def __init__(self, config: Config):
        from . import hivedcore

        self._lock = threading.RLock()
        try:
            self._core = hivedcore.HivedCore(normalize_spec(config))
        except hivedcore.CoreError as e:
            raise WebServerError(getattr(e, "code", 500), str(e)) from e
        self._config = config

    def _call(self, fn, *args):
        from . import hivedcore

        with self._lock:
            try:
                return fn(*args)
            except hivedcore.CoreError as e:
                raise WebServerError(getattr(e, "code", 400), str(e)) from e

    # --- node events ---
    def add_node(self, name: str, healthy: bool = True) -> None:
        self._call(self._core.set_node_healthy, name, healthy)

    def update_node(self, name: str, healthy: bool) -> None:
        self._call(self._core.set_node_healthy, name, healthy)

    def delete_node(self, name: str) -> None:
        self._call(self._core.set_node_healthy, name, False)

    def set_bad_node(self, name: str) -> None:
        self._call(self._core.set_node_healthy, name, False)

    def set_healthy_node(self, name: str) -> None:
        self._call(self._core.set_node_healthy, name, True)

    def set_leaf_cell_healthy(self, node: str, leaf_index: int, healthy: bool) -> None:
        """GPU/xGMI-level health: marks one leaf cell; badness rolls up to the
        pair/quad/node cells. Independent of node-level health."""
        self._call(self._core.set_leaf_cell_healthy, node, leaf_index, healthy)

    def set_xgmi_link_healthy(self, node: str, a: int, b: int, healthy: bool,
                              gbps: float = 0.0) -> None:
        """First-class xGMI link health: a degraded link between GPUs a and b
        of one node makes multi-GPU placements avoid co-placing the two
        endpoints, while both GPUs stay schedulable for 1-GPU work (unlike
        set_leaf_cell_healthy, which removes a GPU entirely)."""
        self._call(self._core.set_xgmi_link_healthy, node, a, b, healthy, gbps)

    def get_xgmi_links(self, node: str) -> List[dict]:
        """Per-node link table: [{a, b, gbps, healthy}, ...]."""
        return self._call(self._core.xgmi_links, node)

    def all_nodes(self) -> List[str]:
        return self._call(self._core.all_nodes)

    def bad_nodes(self) -> List[str]:
        return self._call(self._core.bad_nodes)

    # --- scheduling ---
    def schedule(
        self,
        spec: PodSchedulingSpec,
        pod_key: str,
        suggested_nodes: List[str],
        phase: str = FILTERING,
    ) -> ScheduleResult:
        raw = self._call(self._core.schedule, spec.to_dict(), pod_key, suggested_nodes, phase)
        return ScheduleResult(raw)

    def add_unallocated_pod(self, spec: PodSchedulingSpec, pod_key: str) -> None:
        pass  # parity: reference AddUnallocatedPod is a no-op

    def delete_unallocated_pod(self, spec: PodSchedulingSpec, pod_key: str) -> None:
        self._call(self._core.delete_unallocated_pod, spec.to_dict(), pod_key)

    def add_allocated_pod(self, spec: PodSchedulingSpec, info: PodBindInfo, pod_key: str) -> None:
        self._call(self._core.add_allocated_pod, spec.to_dict(), info.to_dict(), pod_key)

    def delete_allocated_pod(self, spec: PodSchedulingSpec, info: PodBindInfo, pod_key: str) -> None:
        self._call(self._core.delete_allocated_pod, spec.to_dict(), info.to_dict(), pod_key)

    # --- inspect ---
    def get_all_affinity_groups(self) -> List[dict]:
        return self._call(self._core.get_all_affinity_groups)

    def get_affinity_group(self, name: str) -> dict:
        return self._call(self._core.get_affinity_group, name)

    def get_cluster_status(self) -> dict:
        return self._call(self._core.get_cluster_status)

    def get_physical_cluster_status(self) -> list:
        return self._call(self._core.get_physical_cluster_status)

    def get_all_virtual_clusters_status(self) -> dict:
        return self._call(self._core.get_all_virtual_clusters_status)

    def get_virtual_cluster_status(self, vc: str) -> list:
        return self._call(self._core.get_virtual_cluster_status, vc)

    def schedule_count(self) -> int:
        return self._call(self._core.schedule_count)
