"""Preemption scenarios: inter-VC (guaranteed over opportunistic), priority
preemption, preemption cancellation, lazy preemption. Parity with reference
testStatefulPreemption (hived_algorithm_test.go:855-907)."""
import pytest

from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config


def two_vc_sim():
    return SimScheduler(
        mi355x_cluster_config(num_nodes=1, vcs={"VC1": [("MI355X-NODE.MI355X-QUAD", 1)],
                                                 "VC2": [("MI355X-NODE.MI355X-QUAD", 1)]})
    )


def test_guaranteed_preempts_opportunistic():
    """BASELINE config 3: opportunistic 4-GPU job preempted by guaranteed
    4-GPU job on a shared node."""
    sim = two_vc_sim()
    # opportunistic jobs fill the whole node
    assert sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=4)).kind == "bind"
    assert sim.schedule("ns/o2", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=4)).kind == "bind"
    # guaranteed VC1 job must get its quad back via preemption
    spec = sim.pod_spec(vc="VC1", priority=1, leaf_cells=4)
    r = sim.schedule("ns/g1", spec)  # Filtering phase: reports victims, no state
    assert r.kind == "preempt"
    assert sim.alg.get_all_affinity_groups() and len(sim.alg.get_all_affinity_groups()) == 2
    final = sim.run_preemption_to_completion("ns/g1", spec)
    assert final.kind == "bind"
    assert len(final.bind_info.leafCellIsolation) == 4
    # exactly one opportunistic group was gang-victimized
    remaining = {g["name"] for g in sim.alg.get_all_affinity_groups()}
    assert "ns/g1" in remaining
    assert len(remaining) == 2


def test_gang_victim_semantics():
    """Preempting one cell of a group victimizes the whole group."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=2, vcs={
        "VC1": [("MI355X-NODE", 2)]}))
    # opportunistic group spanning both nodes (2 pods x 8 GPUs)
    spec_o = sim.pod_spec(vc="VC1", priority=-1, leaf_cells=8, group="og", members=[(2, 8)])
    assert sim.schedule("ns/og-0", spec_o).kind == "bind"
    assert sim.schedule("ns/og-1", spec_o).kind == "bind"
    # a guaranteed 8-GPU pod needs only one node, but victims = both pods
    spec_g = sim.pod_spec(vc="VC1", priority=0, leaf_cells=8)
    r = sim.preempt("ns/g", spec_g)
    assert r.kind == "preempt"
    all_victims = set()
    # victims are reported one node per round
    for _ in range(4):
        r = sim.preempt("ns/g", spec_g)
        if r.kind != "preempt":
            break
        all_victims.update(r.victim_pod_keys)
        for v in r.victim_pod_keys:
            sim.delete_pod(v)
    assert all_victims == {"ns/og-0", "ns/og-1"}
    assert r.kind == "bind"


def test_priority_preemption_and_cancellation():
    """A higher-priority preemptor cancels a lower-priority preempting group."""
    sim = two_vc_sim()
    assert sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=4)).kind == "bind"
    assert sim.schedule("ns/o2", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=4)).kind == "bind"

    # p1 (priority 1) starts preempting
    spec_p1 = sim.pod_spec(vc="VC1", priority=1, leaf_cells=4, group="p1")
    r = sim.preempt("ns/p1", spec_p1)
    assert r.kind == "preempt"
    groups = {g["name"]: g for g in sim.alg.get_all_affinity_groups()}
    assert groups["p1"]["state"] == "Preempting"

    # p2 (priority 2, same VC) needs VC1's quad — the very cells p1 reserved —
    # so p1's preemption is canceled by the strictly-higher priority
    spec_p2 = sim.pod_spec(vc="VC1", priority=2, leaf_cells=4, group="p2")
    r2 = sim.preempt("ns/p2", spec_p2)
    assert r2.kind == "preempt"
    groups = {g["name"]: g for g in sim.alg.get_all_affinity_groups()}
    assert "p1" not in groups, "lower-priority preemptor should be canceled"
    assert groups["p2"]["state"] == "Preempting"

    final = sim.run_preemption_to_completion("ns/p2", spec_p2)
    assert final.kind == "bind"


def test_preemption_cancel_when_pods_deleted():
    sim = two_vc_sim()
    assert sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=4)).kind == "bind"
    assert sim.schedule("ns/o2", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=4)).kind == "bind"
    spec = sim.pod_spec(vc="VC1", priority=1, leaf_cells=4, group="p1")
    assert sim.preempt("ns/p1", spec).kind == "preempt"
    assert any(g["name"] == "p1" for g in sim.alg.get_all_affinity_groups())
    # the preemptor pod itself is deleted while waiting
    sim.delete_unallocated("ns/p1", spec)
    assert not any(g["name"] == "p1" for g in sim.alg.get_all_affinity_groups())
    # the victims keep running
    assert {g["state"] for g in sim.alg.get_all_affinity_groups()} == {"Allocated"}


def test_equal_priority_no_preemption():
    sim = two_vc_sim()
    assert sim.schedule("ns/g1", sim.pod_spec(vc="VC1", priority=1, leaf_cells=4)).kind == "bind"
    assert sim.schedule("ns/g2", sim.pod_spec(vc="VC2", priority=1, leaf_cells=4)).kind == "bind"
    # same priority cannot preempt: waits (also out of quota)
    r = sim.preempt("ns/g3", sim.pod_spec(vc="VC1", priority=1, leaf_cells=4))
    assert r.kind == "wait"


def test_reserved_cells_not_stolen_by_equal_priority():
    """Anti-deadlock: a Preempting group's reserved cells are not contended by
    an equal-priority group."""
    sim = two_vc_sim()
    assert sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=8)).kind == "bind"
    spec_a = sim.pod_spec(vc="VC1", priority=1, leaf_cells=4, group="a")
    assert sim.preempt("ns/a", spec_a).kind == "preempt"
    # group b, same priority, also wants 4 GPUs: must not reuse a's reservation
    spec_b = sim.pod_spec(vc="VC2", priority=1, leaf_cells=4, group="b")
    rb = sim.preempt("ns/b", spec_b)
    assert rb.kind in ("preempt", "wait")
    # complete a's preemption; its placement must be intact
    final = sim.run_preemption_to_completion("ns/a", spec_a)
    assert final.kind == "bind"


def test_lazy_preemption():
    """A lazy-preemptable group is downgraded to opportunistic instead of
    being killed."""
    sim = two_vc_sim()
    # lazy-preemptable guaranteed job in VC1 occupying the whole node beyond
    # its quota? No: within quota (its quad); then VC2's guaranteed job needs
    # its own quad but VC1's job sits on it after VC1 expanded opportunistically.
    spec_lazy = sim.pod_spec(vc="VC1", priority=1, leaf_cells=4, group="lazy",
                             lazy_preemption=True)
    assert sim.schedule("ns/lazy", spec_lazy).kind == "bind"
    # fill rest with an opportunistic pod
    assert sim.schedule("ns/o1", sim.pod_spec(vc="VC2", priority=-1, leaf_cells=4)).kind == "bind"
    # VC2 guaranteed job: the free quad for VC2 is occupied by o1 -> preempt o1,
    # not the lazy group
    spec_g = sim.pod_spec(vc="VC2", priority=1, leaf_cells=4, group="g")
    final = sim.run_preemption_to_completion("ns/g", spec_g)
    assert final.kind == "bind"
    groups = {g["name"]: g for g in sim.alg.get_all_affinity_groups()}
    assert groups["lazy"]["state"] == "Allocated"
    assert groups["lazy"]["lazyPreemptionStatus"] is None


def test_lazy_preemption_downgrade():
    """A lazy-preemptable group whose virtual cells a higher-priority job in
    the SAME VC needs is downgraded to opportunistic (not killed)."""
    sim = two_vc_sim()
    spec_lazy = sim.pod_spec(vc="VC1", priority=1, leaf_cells=4, group="lazy",
                             lazy_preemption=True)
    assert sim.schedule("ns/lazy", spec_lazy).kind == "bind"
    # a priority-2 job in VC1 needs VC1's (only) quad
    spec_g = sim.pod_spec(vc="VC1", priority=2, leaf_cells=4, group="g")
    r = sim.preempt("ns/g", spec_g)
    groups = {g["name"]: g for g in sim.alg.get_all_affinity_groups()}
    assert groups["lazy"]["lazyPreemptionStatus"] is not None
    assert groups["lazy"]["lazyPreemptionStatus"]["preemptor"] == "g"
    # the lazy group keeps running (as opportunistic); g lands on the other quad
    assert groups["lazy"]["state"] in ("Allocated", "BeingPreempted")
    if r.kind == "preempt":
        final = sim.run_preemption_to_completion("ns/g", spec_g)
        assert final.kind == "bind"
    else:
        assert r.kind == "bind"
        assert "lazy" in {g["name"] for g in sim.alg.get_all_affinity_groups()}


def test_victim_node_spread():
    """Victim-node selection is randomized (seeded): repeated filter-phase
    preemption probes for a cross-node gang report victims from DIFFERENT
    nodes over rounds, spreading churn instead of herding on one node
    (reference utils.go:82-103 behavior)."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    for i in range(2):
        r = sim.schedule(f"ns/op{i}", sim.pod_spec(leaf_cells=8, priority=-1))
        assert r.kind == "bind", i
    spec = sim.pod_spec(leaf_cells=8, group="gang", members=[(2, 8)])
    seen = set()
    for _ in range(30):
        r = sim.schedule("ns/gang-0", spec, commit=False)
        assert r.kind == "preempt"
        seen.add(r.victim_node)
        if len(seen) == 2:
            break
    assert seen == {"node1", "node2"}, f"victim churn herded on {seen}"
