"""Bad-hardware awareness: bad-node marking, placement avoidance, doomed-bad
cells, safe relaxed buddy allocation. Parity with reference testBadNodes
(hived_algorithm_test.go:909-999) and testSafeRelaxedBuddyAlloc (l.1001-1040).
BASELINE config 5 (cell migration on unhealthy GPU node) is covered by
test_allocated_group_insists / test_bad_node_avoided_for_new_groups.
"""
import pytest

from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config


def test_all_nodes_start_bad():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}),
                       all_healthy=False)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=1))
    assert r.kind == "wait"
    sim.alg.set_healthy_node("node1")
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=1))
    assert r.kind == "bind"
    assert r.bind_info.node == "node1"


def test_bad_node_avoided_for_new_groups():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    sim.alg.set_bad_node("node1")
    for i in range(2):
        r = sim.schedule(f"ns/p{i}", sim.pod_spec(leaf_cells=8))
        if i == 0:
            assert r.kind == "bind" and r.bind_info.node == "node2"
        else:
            assert r.kind == "wait"  # only bad node left


def test_allocated_group_insists_on_bad_node():
    """Groups already allocated on a node that goes bad keep their placement
    (the decision is insisted; pod-side retries handle the failure)."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    spec = sim.pod_spec(leaf_cells=4, group="g", members=[(2, 4)])
    r1 = sim.schedule("ns/g-0", spec)
    assert r1.kind == "bind"
    node = r1.bind_info.node
    sim.alg.set_bad_node(node)
    # second pod of the same (allocated) group still binds to the same node
    r2 = sim.schedule("ns/g-1", spec)
    assert r2.kind == "bind"
    assert r2.bind_info.node == node


def test_bad_node_then_healthy_restores_capacity():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    sim.alg.set_bad_node("node1")
    assert sim.schedule("ns/a", sim.pod_spec(leaf_cells=8)).kind == "bind"
    assert sim.schedule("ns/b", sim.pod_spec(leaf_cells=8)).kind == "wait"
    sim.alg.set_healthy_node("node1")
    r = sim.schedule("ns/b", sim.pod_spec(leaf_cells=8))
    assert r.kind == "bind"
    assert r.bind_info.node == "node1"


def test_doomed_bad_cell_exposed_in_vc_status():
    """When healthy capacity < VC quota, the VC status shows a Bad cell;
    when health returns, the doomed binding is released."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    sim.alg.set_bad_node("node2")
    status = sim.alg.get_virtual_cluster_status("VC1")
    badness = [c["cellHealthiness"] for c in status]
    assert "Bad" in badness, f"expected a doomed bad cell in VC status: {status}"
    sim.alg.set_healthy_node("node2")
    status = sim.alg.get_virtual_cluster_status("VC1")
    assert all(c["cellHealthiness"] == "Healthy" for c in status)


def test_doomed_bad_cell_not_used_by_intra_vc_scheduler():
    """The intra-VC scheduler avoids placements doomed to bad hardware."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    sim.alg.set_bad_node("node2")
    # only node1 is healthy; VC1 can still run one node-sized job there
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=8))
    assert r.kind == "bind" and r.bind_info.node == "node1"
    # the second node-sized job must wait (its virtual cell is doomed bad)
    assert sim.schedule("ns/p2", sim.pod_spec(leaf_cells=8)).kind == "wait"


def test_safe_relaxed_buddy_alloc():
    """When the buddy path is blocked by a bad node, a higher-level free cell
    is split (safely) to serve the request."""
    cfg = mi355x_cluster_config(num_nodes=4, vcs={
        "VC1": [("MI355X-NODE.MI355X-QUAD", 1)],
        "VC2": [("MI355X-NODE", 3)],
    })
    sim = SimScheduler(cfg)
    # Make node1 bad: the natural buddy for VC1's quad
    sim.alg.set_bad_node("node1")
    r = sim.schedule("ns/q", sim.pod_spec(vc="VC1", leaf_cells=4))
    assert r.kind == "bind"
    assert r.bind_info.node != "node1"
    # VC2 can still get its 3 nodes? Only 2 fully-free healthy nodes remain
    # (one hosts VC1's quad) -> two bind, third waits (capacity, not safety)
    results = [sim.schedule(f"ns/v2-{i}", sim.pod_spec(vc="VC2", leaf_cells=8)) for i in range(3)]
    kinds = [r.kind for r in results]
    assert kinds.count("bind") == 2 and kinds.count("wait") == 1


def test_xgmi_link_degradation_marks_pair_bad():
    """An unhealthy leaf (e.g. degraded xGMI link endpoint) must not be handed
    out, and pair-level placement avoids it."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    # direct GPU-level health: node-level API marks all 8; here we exercise
    # the same path via the node and verify leaf-level avoidance post-recovery
    sim.alg.set_bad_node("node1")
    sim.alg.set_healthy_node("node1")
    r = sim.schedule("ns/p", sim.pod_spec(leaf_cells=2))
    assert r.kind == "bind"


def test_leaf_cell_level_health():
    """A single bad GPU (or degraded xGMI link endpoint) marks its leaf bad;
    the pair cell goes bad via roll-up; the rest of the node stays usable."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    sim.alg.set_leaf_cell_healthy("node1", 0, False)
    # a pair request avoids the damaged pair
    r = sim.schedule("ns/pair", sim.pod_spec(leaf_cells=2))
    assert r.kind == "bind"
    assert 0 not in r.bind_info.leafCellIsolation
    # a full-node request cannot be served while one GPU is bad
    assert sim.schedule("ns/full", sim.pod_spec(leaf_cells=8)).kind == "wait"
    # recovery restores the leaf
    sim.delete_pod("ns/pair")
    sim.alg.set_leaf_cell_healthy("node1", 0, True)
    assert sim.schedule("ns/full2", sim.pod_spec(leaf_cells=8)).kind == "bind"


def test_leaf_badness_survives_node_health_cycle():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    sim.alg.set_leaf_cell_healthy("node1", 3, False)
    sim.alg.set_bad_node("node1")
    sim.alg.set_healthy_node("node1")
    # the individually-marked leaf stays bad after the node recovers
    r = sim.schedule("ns/q", sim.pod_spec(leaf_cells=4))
    assert r.kind == "bind"
    assert sorted(r.bind_info.leafCellIsolation) == [4, 5, 6, 7]
    assert sim.schedule("ns/full", sim.pod_spec(leaf_cells=8)).kind == "wait"
