"""In-memory simulation harness: drives HivedAlgorithm exactly like the real
scheduler framework does (optimistic AddAllocatedPod right after a bind
decision, K8s-style victim deletion for preemption), with no Kubernetes.

This mirrors the reference's test strategy (pkg/algorithm/hived_algorithm_test.go
drives the algorithm directly) and powers bench.py configs 1 and 4.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..algorithm import FILTERING, PREEMPTING, HivedAlgorithm, ScheduleResult
from ..api.types import (
    Config,
    PhysicalClusterSpec,
    PodBindInfo,
    PodSchedulingSpec,
    VirtualCellSpec,
    VirtualClusterSpec,
)
from ..api import config as apicfg
from ..internal.pod import validate_pod_scheduling_spec
from ..topo.mi355x import mi355x_cell_types, mi355x_node_cell


def mi355x_cluster_config(
    num_nodes: int = 1,
    vcs: Optional[Dict[str, List[Tuple[str, int]]]] = None,
    node_prefix: str = "node",
) -> Config:
    """A homogeneous cluster of 8-GPU MI355X nodes.

    vcs: VC name -> list of (cellTypePath, number). Default: one VC owning all
    nodes.
    """
    if vcs is None:
        vcs = {"VC1": [("MI355X-NODE", num_nodes)]}
    cfg = Config(
        physicalCluster=PhysicalClusterSpec(
            cellTypes=mi355x_cell_types(),
            physicalCells=[mi355x_node_cell(f"{node_prefix}{i + 1}") for i in range(num_nodes)],
        ),
        virtualClusters={
            vc: VirtualClusterSpec(
                virtualCells=[VirtualCellSpec(cellType=t, cellNumber=n) for t, n in quota]
            )
            for vc, quota in vcs.items()
        },
    )
    apicfg.infer_physical_cluster(cfg.physicalCluster)
    return cfg


class SimScheduler:
    """Simulated cluster driving the algorithm with the scheduler framework's
    protocol (schedule -> optimistic commit -> bind; preempt -> victim delete
    -> re-filter)."""

    def __init__(self, config: Config, all_healthy: bool = True):
        self.alg = HivedAlgorithm(config)
        self.config = config
        self.pods: Dict[str, Tuple[PodSchedulingSpec, PodBindInfo]] = {}
        if all_healthy:
            self.set_all_nodes_healthy()

    def set_all_nodes_healthy(self) -> None:
        for n in self.alg.all_nodes():
            self.alg.set_healthy_node(n)

    # -- request construction -----------------------------------------------
    @staticmethod
    def pod_spec(
        vc: str = "VC1",
        priority: int = 0,
        leaf_cells: int = 1,
        group: Optional[str] = None,
        members: Optional[List[Tuple[int, int]]] = None,  # (podNumber, leafCellNumber)
        leaf_cell_type: str = "",
        pinned_cell_id: str = "",
        lazy_preemption: bool = False,
        ignore_suggested: bool = True,
        hbm_bytes_per_cell: int = 0,
    ) -> PodSchedulingSpec:
        from ..api.types import AffinityGroupMemberSpec, AffinityGroupSpec

        spec = PodSchedulingSpec(
            virtualCluster=vc,
            priority=priority,
            pinnedCellId=pinned_cell_id,
            leafCellType=leaf_cell_type,
            leafCellNumber=leaf_cells,
            lazyPreemptionEnable=lazy_preemption,
            ignoreK8sSuggestedNodes=ignore_suggested,
            hbmBytesPerCell=hbm_bytes_per_cell,
        )
        if group is not None:
            spec.affinityGroup = AffinityGroupSpec(
                name=group,
                members=[
                    AffinityGroupMemberSpec(podNumber=p, leafCellNumber=g)
                    for p, g in (members or [(1, leaf_cells)])
                ],
            )
        return spec

    # -- scheduling protocol --------------------------------------------------
    def schedule(
        self,
        key: str,
        spec: PodSchedulingSpec,
        suggested: Optional[List[str]] = None,
        phase: str = FILTERING,
        commit: bool = True,
    ) -> ScheduleResult:
        """Schedule a pod; on a bind decision, optimistically commit it
        (AddAllocatedPod) like scheduler.go:518-539 does."""
        validate_pod_scheduling_spec(spec, key)
        if suggested is None:
            suggested = self.alg.all_nodes()
        result = self.alg.schedule(spec, key, suggested, phase)
        if result.kind == "bind" and commit:
            self.alg.add_allocated_pod(spec, result.bind_info, key)
            self.pods[key] = (spec, result.bind_info)
        return result

    def preempt(self, key: str, spec: PodSchedulingSpec,
                suggested: Optional[List[str]] = None, commit: bool = True) -> ScheduleResult:
        return self.schedule(key, spec, suggested, phase=PREEMPTING, commit=commit)

    def delete_pod(self, key: str) -> None:
        spec, info = self.pods.pop(key)
        self.alg.delete_allocated_pod(spec, info, key)

    def delete_unallocated(self, key: str, spec: PodSchedulingSpec) -> None:
        validate_pod_scheduling_spec(spec, key)
        self.alg.delete_unallocated_pod(spec, key)

    def run_preemption_to_completion(
        self, key: str, spec: PodSchedulingSpec, suggested: Optional[List[str]] = None
    ) -> ScheduleResult:
        """Full preemption protocol: preempt phase -> delete victims (K8s's
        role) -> re-filter until bound. Returns the final result."""
        for _ in range(64):  # bounded; each round deletes at least one node's victims
            result = self.preempt(key, spec, suggested)
            if result.kind != "preempt":
                return result
            for victim in result.victim_pod_keys:
                if victim in self.pods:
                    self.delete_pod(victim)
        raise RuntimeError("preemption did not converge")

    # -- assertions ------------------------------------------------------------
    def assert_empty(self) -> None:
        groups = self.alg.get_all_affinity_groups()
        assert groups == [], f"groups not empty: {groups}"
