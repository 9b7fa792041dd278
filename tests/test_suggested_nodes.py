"""K8s suggested-node handling. Parity with reference testSuggestedNodes
(hived_algorithm_test.go:753-853): with ignoreK8sSuggestedNodes=false the
scheduler avoids binding cells on non-suggested nodes; Filtering phase never
creates preemption state."""
from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config


def sim2():
    return SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))


def test_respect_suggested_nodes():
    sim = sim2()
    spec = sim.pod_spec(leaf_cells=8, ignore_suggested=False)
    r = sim.schedule("ns/p1", spec, suggested=["node2"])
    assert r.kind == "bind"
    assert r.bind_info.node == "node2"
    # no suggested node usable -> wait
    r2 = sim.schedule("ns/p2", sim.pod_spec(leaf_cells=8, ignore_suggested=False), suggested=["node2"])
    assert r2.kind == "wait"


def test_ignore_suggested_nodes_default():
    sim = sim2()
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=8), suggested=["node2"])
    assert r.kind == "bind"  # default ignores the suggestion restriction


def test_filtering_phase_creates_no_preemption_state():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1, vcs={
        "VC1": [("MI355X-NODE.MI355X-QUAD", 1)], "VC2": [("MI355X-NODE.MI355X-QUAD", 1)]}))
    assert sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=8)).kind == "bind"
    r = sim.schedule("ns/g", sim.pod_spec(vc="VC1", priority=1, leaf_cells=4))
    assert r.kind == "preempt"
    # Filtering phase must not have registered the preemptor group
    assert {g["name"] for g in sim.alg.get_all_affinity_groups()} == {"ns/o1"}


def test_preempting_group_canceled_on_non_suggested(design_sim):
    """A Preempting group whose placement leaves the suggested set is canceled
    and rescheduled (only allocated groups insist)."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={
        "VC1": [("MI355X-NODE.MI355X-QUAD", 2)], "VC2": [("MI355X-NODE.MI355X-QUAD", 2)]}))
    # fill node1 with opportunistic
    assert sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=8)).kind == "bind"
    assert sim.schedule("ns/o2", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=8)).kind == "bind"
    spec = sim.pod_spec(vc="VC1", priority=1, leaf_cells=4, group="g", ignore_suggested=False)
    r = sim.preempt("ns/g", spec, suggested=["node1", "node2"])
    assert r.kind == "preempt"
    groups = {g["name"]: g for g in sim.alg.get_all_affinity_groups()}
    reserved_node = next(iter(groups["g"]["physicalPlacement"]))
    other = "node2" if reserved_node == "node1" else "node1"
    # now K8s only suggests the other node: the preemption is canceled and
    # re-created on the suggested node
    r2 = sim.preempt("ns/g", spec, suggested=[other])
    assert r2.kind == "preempt"
    groups = {g["name"]: g for g in sim.alg.get_all_affinity_groups()}
    assert next(iter(groups["g"]["physicalPlacement"])) == other
