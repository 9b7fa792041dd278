"""Feature-demo parity suite: every demo in the reference's
example/feature/README.md, executed 1:1 against the in-memory simulation
harness (no Kubernetes, like the reference's own test vehicle).

Reference sections (example/feature/README.md line numbers):
  VC Safety (l.7), Pinned Cells (l.22), SKU Type (l.33), Gang Scheduling
  (l.52), Incremental Scheduling (l.75), Guaranteed Job (l.88), Opportunistic
  Job (l.97), Intra-VC Preemption (l.106), Inter-VC Preemption (l.128),
  Topology-Aware Intra-VC Scheduling (l.139), Work-Preserving Reconfiguration
  (l.151), Bad Hardware Awareness (l.210).

docs/features.md documents the same 12 demos for users.
"""
import pytest

from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config


def two_vc_two_node_sim():
    """Two 8-GPU MI355X nodes; VC1 and VC2 each own one node cell."""
    return SimScheduler(
        mi355x_cluster_config(
            num_nodes=2,
            vcs={"VC1": [("MI355X-NODE", 1)], "VC2": [("MI355X-NODE", 1)]},
        )
    )


# -- 1. VC Safety (README l.7-20) -------------------------------------------
def test_feature_vc_safety_no_fragmentation_across_vcs():
    """VC1 floods 8 single-GPU jobs; VC2 must still get one ENTIRE node for an
    8-GPU gang (the reference's two-DGX-2 safety story, l.14-16)."""
    sim = two_vc_two_node_sim()
    for i in range(8):
        r = sim.schedule(f"vc1/one{i}", sim.pod_spec(vc="VC1", priority=0, leaf_cells=1))
        assert r.kind == "bind"
    # all eight VC1 singles must have been packed into ONE node
    vc1_nodes = {sim.pods[f"vc1/one{i}"][1].node for i in range(8)}
    assert len(vc1_nodes) == 1, f"VC1 singles fragmented across {vc1_nodes}"
    # VC2's whole-node gang still fits
    r = sim.schedule("vc2/full", sim.pod_spec(vc="VC2", priority=0, leaf_cells=8))
    assert r.kind == "bind"
    assert len(r.bind_info.leafCellIsolation) == 8
    assert r.bind_info.node not in vc1_nodes


# -- 2. Pinned Cells (README l.22-31) ----------------------------------------
def test_feature_pinned_cells(design_sim):
    """Jobs with pinnedCellId land on the pinned node; jobs without it never
    use the pinned node (reference l.29-31: vc1pinned on the pinned node,
    vc1nopinned NOT on it)."""
    sim = design_sim
    r = sim.schedule("vc1/pin", sim.pod_spec(vc="VC1", priority=0, leaf_cells=8,
                                             pinned_cell_id="VC1-PIN"))
    assert r.kind == "bind"
    pinned_node = r.bind_info.node  # n4 per conftest design config
    assert pinned_node == "n4"
    for i in range(3):
        r = sim.schedule(f"vc1/nopin{i}", sim.pod_spec(vc="VC1", priority=0, leaf_cells=8))
        if r.kind == "bind":
            assert r.bind_info.node != pinned_node


# -- 3. SKU Type (README l.33-50) --------------------------------------------
def test_feature_sku_type_specified(design_sim):
    """leafCellType given -> only that type is allocated (reference l.44-46)."""
    sim = design_sim
    r = sim.schedule("vc2/ct1", sim.pod_spec(vc="VC2", priority=0, leaf_cells=2,
                                             leaf_cell_type="CT1"))
    assert r.kind == "bind"
    assert r.bind_info.node in ("c1", "c2")
    # CT1 quota in VC2 is one 2-GPU node; a second CT1 request must wait
    r2 = sim.schedule("vc2/ct1b", sim.pod_spec(vc="VC2", priority=0, leaf_cells=2,
                                               leaf_cell_type="CT1"))
    assert r2.kind == "wait"


def test_feature_sku_type_not_specified(design_sim):
    """No leafCellType -> any chain may serve the request (reference l.47-50)."""
    sim = design_sim
    nodes = set()
    # VC2 owns 2 MI355X nodes (16 leaves) + 1 CT1 node (2 leaves) = 18 leaves.
    # Nine 2-GPU typeless requests need both SKUs to all bind.
    for i in range(9):
        r = sim.schedule(f"vc2/any{i}", sim.pod_spec(vc="VC2", priority=0, leaf_cells=2))
        assert r.kind == "bind", f"request {i} should bind: {r}"
        nodes.add(r.bind_info.node)
    assert nodes & {"c1", "c2"}, "CT1 chain never used"
    assert nodes - {"c1", "c2"}, "MI355X chain never used"


# -- 4. Gang Scheduling (README l.52-73) --------------------------------------
def test_feature_gang_all_or_nothing_no_hol_blocking():
    """A 6-pod gang on a 4-GPU quota waits entirely; a later 4-pod gang runs
    anyway — no head-of-line blocking (reference l.63-66)."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=1, vcs={"VC1": [("MI355X-NODE.MI355X-QUAD", 1)]}))
    big = sim.pod_spec(vc="VC1", priority=0, leaf_cells=1, group="gang6",
                       members=[(6, 1)])
    assert sim.schedule("ns/gang6-0", big).kind == "wait"
    ok = sim.pod_spec(vc="VC1", priority=0, leaf_cells=1, group="gang4",
                      members=[(4, 1)])
    for i in range(4):
        assert sim.schedule(f"ns/gang4-{i}", ok).kind == "bind"
    # the big gang still waits (not partially allocated); the group does not
    # exist in the allocated registry (404 like the reference's inspect API)
    assert sim.schedule("ns/gang6-0", big).kind == "wait"
    from hivedscheduler_amd.api.types import WebServerError

    with pytest.raises(WebServerError):
        sim.alg.get_affinity_group("gang6")


# -- 5. Incremental Scheduling (README l.75-86) -------------------------------
def test_feature_incremental_scheduling():
    """Pods in separate singleton groups: a job larger than its VC quota still
    partially runs (reference itc-elastic, l.84-86)."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=1, vcs={"VC1": [("MI355X-NODE.MI355X-QUAD", 1)]}))
    bound = waited = 0
    for i in range(6):  # 6 x 1-GPU, quota 4
        r = sim.schedule(f"ns/elastic{i}", sim.pod_spec(vc="VC1", priority=0, leaf_cells=1))
        bound += r.kind == "bind"
        waited += r.kind == "wait"
    assert bound == 4 and waited == 2


# -- 6. Guaranteed Job (README l.88-95) ---------------------------------------
def test_feature_guaranteed_job_capped_by_vc_quota():
    """priority >= 0 can only use its own VC's quota even when the cluster has
    free cells elsewhere (reference l.90)."""
    sim = two_vc_two_node_sim()
    assert sim.schedule("vc1/g8", sim.pod_spec(vc="VC1", priority=0, leaf_cells=8)).kind == "bind"
    # VC1 quota exhausted; node 2 is free but belongs to VC2
    assert sim.schedule("vc1/g1", sim.pod_spec(vc="VC1", priority=0, leaf_cells=1)).kind == "wait"


# -- 7. Opportunistic Job (README l.97-104) -----------------------------------
def test_feature_opportunistic_job_uses_other_vcs_quota():
    """priority -1 may exceed its VC quota, spreading over other VCs' free
    cells (reference l.99: 'it will use more than one node')."""
    sim = two_vc_two_node_sim()
    nodes = set()
    for i in range(16):  # both nodes' worth of GPUs
        r = sim.schedule(f"vc1/o{i}", sim.pod_spec(vc="VC1", priority=-1, leaf_cells=1))
        assert r.kind == "bind"
        nodes.add(r.bind_info.node)
    assert len(nodes) == 2


# -- 8. Intra-VC Preemption (README l.106-127) --------------------------------
def test_feature_intra_vc_immediate_preemption():
    """Same VC: prod (100) preempts test (0) immediately (reference l.113-117)."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    assert sim.schedule("vc1/test", sim.pod_spec(vc="VC1", priority=0, leaf_cells=8)).kind == "bind"
    r = sim.run_preemption_to_completion(
        "vc1/prod", sim.pod_spec(vc="VC1", priority=100, leaf_cells=8))
    assert r.kind == "bind"
    assert "vc1/test" not in sim.pods


def test_feature_intra_vc_lazy_preemption():
    """Lazy preemption: the victim's VIRTUAL cells go to the preemptor while
    the victim keeps running on its physical cells as an opportunistic group
    — possible here because physical room remains (reference l.118-127)."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 1)],
                          "VC2": [("MI355X-NODE", 1)]}))
    lazy = sim.pod_spec(vc="VC1", priority=0, leaf_cells=8, group="lazyg",
                        lazy_preemption=True)
    assert sim.schedule("vc1/lazy", lazy).kind == "bind"
    # prod job in the SAME VC needs VC1's only node cell -> lazy-preempts it;
    # physically it can land on the free node (VC2's, now squatted legally
    # since lazyg became opportunistic / prod is guaranteed)
    r = sim.run_preemption_to_completion(
        "vc1/prod", sim.pod_spec(vc="VC1", priority=100, leaf_cells=8))
    assert r.kind == "bind"
    g = sim.alg.get_affinity_group("lazyg")
    assert g.get("lazyPreemptionStatus") is not None
    assert g["lazyPreemptionStatus"]["preemptor"] == "vc1/prod"
    # the downgraded group keeps running
    assert g["state"] in ("Allocated", "BeingPreempted")
    assert "vc1/lazy" in sim.pods


# -- 9. Inter-VC Preemption (README l.128-137) --------------------------------
def test_feature_inter_vc_preemption():
    """A guaranteed job preempts another VC's opportunistic job squatting on
    its quota (reference l.130)."""
    sim = two_vc_two_node_sim()
    # VC2 opportunistic fills both nodes
    for i in range(2):
        assert sim.schedule(f"vc2/o{i}", sim.pod_spec(vc="VC2", priority=-1,
                                                      leaf_cells=8)).kind == "bind"
    r = sim.run_preemption_to_completion(
        "vc1/g", sim.pod_spec(vc="VC1", priority=0, leaf_cells=8))
    assert r.kind == "bind"
    assert len(sim.pods) == 2  # one opportunistic victim was deleted


# -- 10. Topology-Aware Intra-VC Scheduling (README l.139-149) ----------------
def test_feature_topology_aware_packing():
    """Multi-pod gang lands with minimal LCA: a 2x4-GPU gang fills ONE node
    (quads of the same node), not 4 GPUs on each of two nodes."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    spec = sim.pod_spec(vc="VC1", priority=0, leaf_cells=4, group="tp",
                        members=[(2, 4)])
    r1 = sim.schedule("ns/tp-0", spec)
    r2 = sim.schedule("ns/tp-1", spec)
    assert r1.kind == r2.kind == "bind"
    assert r1.bind_info.node == r2.bind_info.node
    # and each member got a whole quad (0-3 / 4-7)
    for r in (r1, r2):
        idx = sorted(r.bind_info.leafCellIsolation)
        assert idx in ([0, 1, 2, 3], [4, 5, 6, 7])


# -- 11. Work-Preserving Reconfiguration (README l.151-208) -------------------
def test_feature_work_preserving_reconfiguration():
    """Restart with a GROWN cluster config: bound pods are replayed via
    AddAllocatedPod and keep their exact placement (reference l.151-208;
    shrink/mismatch cases in tests/test_reconfiguration.py)."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    r = sim.schedule("ns/keep", sim.pod_spec(vc="VC1", priority=0, leaf_cells=4))
    assert r.kind == "bind"
    old = sim.pods["ns/keep"]

    sim2 = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    sim2.alg.add_allocated_pod(old[0], old[1], "ns/keep")
    g = sim2.alg.get_all_affinity_groups()
    assert len(g) == 1 and g[0]["state"] == "Allocated"
    assert g[0].get("lazyPreemptionStatus") is None  # placement fit -> kept


# -- 12. Bad Hardware Awareness (README l.210+) -------------------------------
def test_feature_bad_hardware_awareness():
    """An unhealthy node is avoided for new groups; when it recovers the
    capacity returns (deeper cases in tests/test_bad_nodes.py)."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    sim.alg.set_bad_node("node1")
    for i in range(2):
        r = sim.schedule(f"ns/b{i}", sim.pod_spec(vc="VC1", priority=0, leaf_cells=8))
        if i == 0:
            assert r.kind == "bind" and r.bind_info.node == "node2"
        else:
            assert r.kind == "wait"  # only bad capacity left
    sim.alg.set_healthy_node("node1")
    r = sim.schedule("ns/b1", sim.pod_spec(vc="VC1", priority=0, leaf_cells=8))
    assert r.kind == "bind" and r.bind_info.node == "node1"
