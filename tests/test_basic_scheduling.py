"""Normal-operation scenarios: placement quality, gang scheduling, group
lifecycle, user errors. Parity with reference testNormalOperations
(pkg/algorithm/hived_algorithm_test.go:678-751)."""
import pytest

from hivedscheduler_amd.api.types import WebServerError
from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config


def test_pair_and_quad_affinity():
    """Requests land on xGMI-aligned cells: 2 GPUs -> a pair, 4 -> a quad."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=2))
    assert r.kind == "bind"
    assert r.bind_info.node == "node1"
    assert sorted(r.bind_info.leafCellIsolation) == [0, 1]  # one pair

    r2 = sim.schedule("ns/p2", sim.pod_spec(leaf_cells=2))
    assert sorted(r2.bind_info.leafCellIsolation) == [2, 3]  # buddy pair, same quad

    r3 = sim.schedule("ns/p3", sim.pod_spec(leaf_cells=4))
    assert sorted(r3.bind_info.leafCellIsolation) == [4, 5, 6, 7]  # the other quad


def test_one_gpu_packing():
    """1-GPU pods pack into the same pair/quad before spreading."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    cells = []
    for i in range(4):
        r = sim.schedule(f"ns/p{i}", sim.pod_spec(leaf_cells=1))
        cells.extend(r.bind_info.leafCellIsolation)
    assert sorted(cells) == [0, 1, 2, 3]  # first quad fully packed


def test_gang_group_lands_on_one_node():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    spec = sim.pod_spec(leaf_cells=4, group="g1", members=[(2, 4)])
    r1 = sim.schedule("ns/g1-0", spec)
    r2 = sim.schedule("ns/g1-1", spec)
    assert r1.kind == r2.kind == "bind"
    assert r1.bind_info.node == r2.bind_info.node  # 8 GPUs -> one node
    used = sorted(r1.bind_info.leafCellIsolation + r2.bind_info.leafCellIsolation)
    assert used == list(range(8))


def test_gang_all_or_nothing():
    """A group needing more than the VC has waits instead of partially landing."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    spec = sim.pod_spec(leaf_cells=8, group="g1", members=[(2, 8)])
    r = sim.schedule("ns/g1-0", spec)
    assert r.kind == "wait"
    assert r.wait_reason


def test_group_more_pods_than_configured():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    spec = sim.pod_spec(leaf_cells=2, group="g1", members=[(1, 2)])
    assert sim.schedule("ns/p1", spec).kind == "bind"
    with pytest.raises(WebServerError) as e:
        sim.schedule("ns/p2", spec)
    assert e.value.code == 400


def test_heterogeneous_group():
    """A group mixing 2-GPU and 4-GPU members (reference pod8/pod9 scenario)."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    members = [(1, 2), (1, 4)]
    r2 = sim.schedule("ns/h-2", sim.pod_spec(leaf_cells=2, group="h", members=members))
    r4 = sim.schedule("ns/h-4", sim.pod_spec(leaf_cells=4, group="h", members=members))
    assert r2.kind == r4.kind == "bind"
    all_cells = r2.bind_info.leafCellIsolation + r4.bind_info.leafCellIsolation
    assert len(set(all_cells)) == 6
    # the 4-GPU member gets a whole quad
    quad = sorted(r4.bind_info.leafCellIsolation)
    assert quad in ([0, 1, 2, 3], [4, 5, 6, 7])


def test_delete_and_reschedule():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    r1 = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=8))
    assert r1.kind == "bind"
    assert sim.schedule("ns/p2", sim.pod_spec(leaf_cells=8)).kind == "wait"
    sim.delete_pod("ns/p1")
    sim.assert_empty()
    r2 = sim.schedule("ns/p2b", sim.pod_spec(leaf_cells=8))
    assert r2.kind == "bind"


def test_opportunistic_scheduling_beyond_quota(design_sim):
    """Opportunistic pods run on idle cells beyond the VC quota."""
    sim = design_sim
    # VC2 only owns 1 CT1-NODE, but can use both opportunistically
    r1 = sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=2,
                                            leaf_cell_type="CT1"))
    r2 = sim.schedule("ns/o2", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=2,
                                            leaf_cell_type="CT1"))
    assert r1.kind == r2.kind == "bind"
    assert {r1.bind_info.node, r2.bind_info.node} == {"c1", "c2"}


def test_user_errors(design_sim):
    sim = design_sim
    with pytest.raises(WebServerError):
        sim.schedule("ns/bad-vc", sim.pod_spec(vc="NOPE", leaf_cells=1))
    with pytest.raises(WebServerError):
        sim.schedule("ns/bad-pin", sim.pod_spec(vc="VC1", leaf_cells=1, pinned_cell_id="NOPE"))
    with pytest.raises(WebServerError):
        # opportunistic + pinned cell is rejected
        sim.schedule("ns/op-pin", sim.pod_spec(vc="VC1", priority=-1, leaf_cells=1,
                                               pinned_cell_id="VC1-PIN"))
    with pytest.raises(WebServerError):
        sim.schedule("ns/bad-type", sim.pod_spec(vc="VC1", leaf_cells=1, leaf_cell_type="H100"))
    with pytest.raises(WebServerError):
        # VC1 has no CT1 quota: guaranteed CT1 request is a user error
        sim.schedule("ns/vc1-ct1", sim.pod_spec(vc="VC1", leaf_cells=1, leaf_cell_type="CT1"))
    with pytest.raises(WebServerError):
        sim.schedule("ns/bad-prio", sim.pod_spec(vc="VC1", priority=2000, leaf_cells=1))
    with pytest.raises(WebServerError):
        sim.schedule("ns/bad-cells", sim.pod_spec(vc="VC1", leaf_cells=0))


def test_pinned_cell_scheduling(design_sim):
    sim = design_sim
    r = sim.schedule("ns/pin1", sim.pod_spec(vc="VC1", leaf_cells=8, pinned_cell_id="VC1-PIN"))
    assert r.kind == "bind"
    assert r.bind_info.node == "n4"  # the pinned node
    # pinned cell is full now; a second pinned pod waits
    r2 = sim.schedule("ns/pin2", sim.pod_spec(vc="VC1", leaf_cells=8, pinned_cell_id="VC1-PIN"))
    assert r2.kind == "wait"


def test_vc_isolation(design_sim):
    """VC2 guaranteed jobs cannot exceed VC2 quota even when idle capacity
    exists elsewhere (the whole point of VC safety)."""
    sim = design_sim
    # VC2 owns 2 MI355X nodes total (1 on MI355X-NODE chain + 1 on 2- chain)
    assert sim.schedule("ns/v2-1", sim.pod_spec(vc="VC2", leaf_cells=8)).kind == "bind"
    assert sim.schedule("ns/v2-2", sim.pod_spec(vc="VC2", leaf_cells=8)).kind == "bind"
    assert sim.schedule("ns/v2-3", sim.pod_spec(vc="VC2", leaf_cells=8)).kind == "wait"


def test_guaranteed_within_quota_always_schedulable(design_sim):
    """VC1 can always get its guaranteed cells regardless of other VCs' load."""
    sim = design_sim
    # VC2 fills its own quota
    assert sim.schedule("ns/v2-1", sim.pod_spec(vc="VC2", leaf_cells=8)).kind == "bind"
    assert sim.schedule("ns/v2-2", sim.pod_spec(vc="VC2", leaf_cells=8)).kind == "bind"
    # VC1 guaranteed quota: 2 nodes (rack3) + 1 node + 1 quad + pinned node
    for i, cells in enumerate((8, 8, 8, 4)):
        r = sim.schedule(f"ns/v1-{i}", sim.pod_spec(vc="VC1", leaf_cells=cells))
        assert r.kind == "bind", f"VC1 request {i} ({cells} cells) failed: {r}"
    r = sim.schedule("ns/v1-pin", sim.pod_spec(vc="VC1", leaf_cells=8, pinned_cell_id="VC1-PIN"))
    assert r.kind == "bind"


def test_placement_determinism():
    """Identical request sequences produce identical placements across two
    fresh schedulers (no randomness in the decision path — important for
    debugging and for the reference's golden-expectation test style)."""
    from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

    def run():
        sim = SimScheduler(mi355x_cluster_config(
            num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
        out = []
        for i, cells in enumerate([1, 2, 4, 1, 8, 2, 1]):
            r = sim.schedule(f"d/p{i}", sim.pod_spec(vc="VC1", leaf_cells=cells))
            out.append((r.kind, r.bind_info.node if r.kind == "bind" else "",
                        tuple(sorted(r.bind_info.leafCellIsolation)) if r.kind == "bind" else ()))
        return out

    assert run() == run()
