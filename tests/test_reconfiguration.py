"""Work-preserving reconfiguration: rebuild the algorithm with a changed
config and replay bound pods via add_allocated_pod (the pods ARE the
database). Parity with reference testReconfiguration
(hived_algorithm_test.go:1042-1092) and the recovery path (scheduler.go:306-337).
"""
import pytest

from hivedscheduler_amd.algorithm import HivedAlgorithm
from hivedscheduler_amd.api.types import WebServerError
from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config


def replay(sim_old: SimScheduler, new_config) -> SimScheduler:
    """Restart: build a fresh algorithm from new_config and replay all bound
    pods from their bind-info annotations."""
    new_sim = SimScheduler(new_config)
    for key, (spec, info) in sim_old.pods.items():
        new_sim.alg.add_allocated_pod(spec, info, key)
        new_sim.pods[key] = (spec, info)
    return new_sim


def test_recovery_same_config():
    """Crash/restart with unchanged config: full state reconstruction."""
    cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]})
    sim = SimScheduler(cfg)
    assert sim.schedule("ns/a", sim.pod_spec(leaf_cells=8)).kind == "bind"
    assert sim.schedule("ns/b", sim.pod_spec(leaf_cells=4)).kind == "bind"

    sim2 = replay(sim, cfg)
    groups = {g["name"]: g for g in sim2.alg.get_all_affinity_groups()}
    assert set(groups) == {"ns/a", "ns/b"}
    assert all(g["state"] == "Allocated" for g in groups.values())
    assert all(g["lazyPreemptionStatus"] is None for g in groups.values())
    # capacity is fully accounted: only 4 GPUs left
    assert sim2.schedule("ns/c", sim2.pod_spec(leaf_cells=8)).kind == "wait"
    assert sim2.schedule("ns/d", sim2.pod_spec(leaf_cells=4)).kind == "bind"


def test_reconfig_shrunk_vc_lazy_preempts():
    """VC shrinks below its running jobs: replay lazy-preempts the overflow
    group instead of killing it."""
    cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]})
    sim = SimScheduler(cfg)
    assert sim.schedule("ns/a", sim.pod_spec(leaf_cells=8)).kind == "bind"
    assert sim.schedule("ns/b", sim.pod_spec(leaf_cells=8)).kind == "bind"

    new_cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 1)]})
    sim2 = replay(sim, new_cfg)
    groups = {g["name"]: g for g in sim2.alg.get_all_affinity_groups()}
    assert set(groups) == {"ns/a", "ns/b"}
    lazy = [g for g in groups.values() if g["lazyPreemptionStatus"] is not None]
    assert len(lazy) == 1, f"exactly one group should be lazy-preempted: {groups}"


def test_reconfig_removed_node_keeps_group_running():
    """A node disappears from the config: pods on it are insisted (their cells
    ignored), the rest of the group keeps its placement."""
    cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]})
    sim = SimScheduler(cfg)
    spec = sim.pod_spec(leaf_cells=8, group="g", members=[(2, 8)])
    assert sim.schedule("ns/g-0", spec).kind == "bind"
    assert sim.schedule("ns/g-1", spec).kind == "bind"

    new_cfg = mi355x_cluster_config(num_nodes=1, vcs={"VC1": [("MI355X-NODE", 1)]})
    sim2 = replay(sim, new_cfg)
    groups = {g["name"]: g for g in sim2.alg.get_all_affinity_groups()}
    assert "g" in groups
    # the group survives; the node1 half is still tracked
    assert "node1" in groups["g"]["physicalPlacement"]


def test_reconfig_split_chain():
    """Physical cells regrouped into a different chain: pods are recovered by
    cross-chain leaf-cell search."""
    cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]})
    sim = SimScheduler(cfg)
    assert sim.schedule("ns/a", sim.pod_spec(leaf_cells=8)).kind == "bind"

    # new config: the two nodes now form a 2-node rack chain
    from hivedscheduler_amd.api import config as apicfg
    from hivedscheduler_amd.api.types import Config, PhysicalCellSpec, PhysicalClusterSpec, \
        VirtualCellSpec, VirtualClusterSpec
    from hivedscheduler_amd.topo.mi355x import mi355x_cell_types, mi355x_node_cell

    new_cfg = Config(
        physicalCluster=PhysicalClusterSpec(
            cellTypes=mi355x_cell_types(pool_sizes=(2,)),
            physicalCells=[PhysicalCellSpec(
                cellType="2-MI355X-NODE",
                cellChildren=[mi355x_node_cell("node1"), mi355x_node_cell("node2")],
            )],
        ),
        virtualClusters={"VC1": VirtualClusterSpec(
            virtualCells=[VirtualCellSpec(cellType="2-MI355X-NODE.MI355X-NODE", cellNumber=2)])},
    )
    apicfg.infer_physical_cluster(new_cfg.physicalCluster)
    sim2 = replay(sim, new_cfg)
    groups = {g["name"]: g for g in sim2.alg.get_all_affinity_groups()}
    assert "ns/a" in groups
    # old chain name is gone; recovery found the cells in the new chain
    assert groups["ns/a"]["physicalPlacement"].get("node1") is not None


def test_vc_over_physical_capacity_rejected():
    """Invalid initial assignment (VC quota > physical) fails construction."""
    with pytest.raises(WebServerError):
        SimScheduler(mi355x_cluster_config(num_nodes=1, vcs={"VC1": [("MI355X-NODE", 2)]}))
    with pytest.raises(WebServerError):
        SimScheduler(mi355x_cluster_config(num_nodes=1, vcs={
            "VC1": [("MI355X-NODE", 1)],
            "VC2": [("MI355X-NODE.MI355X-QUAD", 1)],
        }))
