"""Randomized operation fuzzing with full invariant checking after every
mutation: schedule (all phases/priorities), delete, preemption protocol,
node/GPU health flapping. The C++ core's check_invariants() verifies the
correctness spec (docs/design.md): roll-ups, free-list/accounting consistency,
VC safety, binding symmetry."""
import random

import pytest

from hivedscheduler_amd.algorithm import PREEMPTING
from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

VCS = {
    "VC1": [("MI355X-NODE", 1), ("MI355X-NODE.MI355X-QUAD", 1)],
    "VC2": [("MI355X-NODE", 1), ("MI355X-NODE.MI355X-QUAD.MI355X-PAIR", 2)],
}


@pytest.mark.parametrize("seed", list(range(8)))
def test_fuzz_operations(seed):
    rng = random.Random(seed)
    sim = SimScheduler(mi355x_cluster_config(num_nodes=4, vcs=VCS))
    check = sim.alg._core.check_invariants
    check()
    live = {}  # key -> spec
    preempting = {}  # key -> spec
    counter = 0
    nodes = sim.alg.all_nodes()
    for step in range(220):
        op = rng.random()
        if op < 0.40:  # schedule a new pod
            counter += 1
            key = f"f/p{counter}"
            spec = sim.pod_spec(
                vc=rng.choice(list(VCS)),
                priority=rng.choice([-1, -1, 0, 1, 2]),
                leaf_cells=rng.choice([1, 2, 2, 4, 8]),
                lazy_preemption=rng.random() < 0.3,
            )
            phase = "Filtering" if rng.random() < 0.6 else PREEMPTING
            r = sim.schedule(key, spec, phase=phase)
            if r.kind == "bind":
                live[key] = spec
            elif r.kind == "preempt" and phase == PREEMPTING:
                preempting[key] = spec
        elif op < 0.60 and live:  # delete a bound pod
            key = rng.choice(list(live))
            del live[key]
            sim.delete_pod(key)
        elif op < 0.70 and preempting:  # advance or cancel a preemption
            key = rng.choice(list(preempting))
            spec = preempting[key]
            if rng.random() < 0.3:
                sim.delete_unallocated(key, spec)
                del preempting[key]
            else:
                r = sim.schedule(key, spec, phase=PREEMPTING)
                if r.kind == "preempt":
                    for v in r.victim_pod_keys:
                        if v in live:
                            del live[v]
                            sim.delete_pod(v)
                elif r.kind == "bind":
                    live[key] = spec
                    del preempting[key]
                else:
                    del preempting[key]
        elif op < 0.85:  # node health flap
            node = rng.choice(nodes)
            sim.alg.update_node(node, rng.random() < 0.7)
        else:  # GPU-level health flap
            node = rng.choice(nodes)
            sim.alg.set_leaf_cell_healthy(node, rng.randrange(8), rng.random() < 0.7)
        check()

    # drain everything; the cluster must return to a fully-free state
    for node in nodes:
        sim.alg.set_healthy_node(node)
        for i in range(8):
            sim.alg.set_leaf_cell_healthy(node, i, True)
    for key in list(preempting):
        sim.delete_unallocated(key, preempting[key])
    for key in list(live):
        sim.delete_pod(key)
    check()
    sim.assert_empty()
    # and the full quota is schedulable again (no leaked accounting)
    for vc, sizes in (("VC1", [8, 4]), ("VC2", [8, 2, 2])):
        for j, cells in enumerate(sizes):
            r = sim.schedule(f"f/final-{vc}-{j}", sim.pod_spec(vc=vc, priority=0,
                                                               leaf_cells=cells))
            assert r.kind == "bind", f"quota not restored for {vc} ({cells} cells): {r}"
    check()


VCS_VARIANTS = [
    VCS,
    {"A": [("MI355X-NODE", 2)], "B": [("MI355X-NODE.MI355X-QUAD", 2)]},
    {"A": [("MI355X-NODE.MI355X-QUAD.MI355X-PAIR", 3)],
     "B": [("MI355X-NODE", 1)], "C": [("MI355X-NODE.MI355X-QUAD", 1)]},
]


@pytest.mark.parametrize("seed", list(range(8)))
def test_fuzz_extended(seed):
    """Wider fuzz: gang groups, priorities up to 10, varied cluster shapes,
    preemption-protocol advancement/cancellation, health flapping — with the
    full invariant check after every operation. This configuration found five
    distinct core bugs during round 1 (see core/alloc.cpp, core/algorithm.cpp
    comments marked "Found by fuzzing")."""
    from hivedscheduler_amd.api.types import WebServerError

    rng = random.Random(1000 + seed)
    nnodes = rng.choice([3, 4])
    vcs = rng.choice(VCS_VARIANTS)
    sim = SimScheduler(mi355x_cluster_config(num_nodes=nnodes, vcs=vcs))
    check = sim.alg._core.check_invariants
    live, preempting, counter = {}, {}, 0
    nodes = sim.alg.all_nodes()
    for step in range(400):
        op = rng.random()
        try:
            if op < 0.38:
                counter += 1
                key = f"f/p{counter}"
                kw = dict(vc=rng.choice(list(vcs)), priority=rng.choice([-1, -1, 0, 1, 2, 10]),
                          leaf_cells=rng.choice([1, 2, 2, 4, 8]),
                          lazy_preemption=rng.random() < 0.3)
                if rng.random() < 0.25:
                    kw["group"] = key
                    kw["members"] = [(rng.choice([1, 2]), kw["leaf_cells"])]
                if rng.random() < 0.3:
                    kw["ignore_suggested"] = False
                spec = sim.pod_spec(**kw)
                phase = "Filtering" if rng.random() < 0.6 else PREEMPTING
                suggested = None
                if rng.random() < 0.3:
                    suggested = rng.sample(nodes, rng.randrange(1, len(nodes) + 1))
                r = sim.schedule(key, spec, phase=phase, suggested=suggested)
                if r.kind == "bind":
                    live[key] = spec
                elif r.kind == "preempt" and phase == PREEMPTING:
                    preempting[key] = spec
            elif op < 0.58 and live:
                key = rng.choice(list(live))
                del live[key]
                sim.delete_pod(key)
            elif op < 0.70 and preempting:
                key = rng.choice(list(preempting))
                spec = preempting[key]
                if rng.random() < 0.3:
                    sim.delete_unallocated(key, spec)
                    del preempting[key]
                else:
                    r = sim.schedule(key, spec, phase=PREEMPTING)
                    if r.kind == "preempt":
                        for v in r.victim_pod_keys:
                            if v in live:
                                del live[v]
                                sim.delete_pod(v)
                    elif r.kind == "bind":
                        live[key] = spec
                        del preempting[key]
                    else:
                        del preempting[key]
            elif op < 0.80:
                sim.alg.update_node(rng.choice(nodes), rng.random() < 0.7)
            elif op < 0.90:
                sim.alg.set_leaf_cell_healthy(rng.choice(nodes), rng.randrange(8),
                                              rng.random() < 0.7)
            elif op < 0.97:
                # xGMI link flapping: first-class link marks compose with
                # leaf/node health and placements must stay invariant-clean
                a, b = rng.sample(range(8), 2)
                sim.alg.set_xgmi_link_healthy(rng.choice(nodes), a, b,
                                              rng.random() < 0.6,
                                              rng.uniform(5.0, 160.0))
            else:
                # crash-recovery: fresh algorithm, replay live pods from bind
                # infos (pods are the database; Preempting state is volatile)
                new = SimScheduler(mi355x_cluster_config(num_nodes=nnodes, vcs=vcs))
                for k, (sp, info) in list(sim.pods.items()):
                    new.alg.add_allocated_pod(sp, info, k)
                    new.pods[k] = (sp, info)
                sim = new
                check = sim.alg._core.check_invariants
                preempting.clear()
            check()
        except WebServerError:
            continue  # 4xx user error (e.g. quota exceeded variants) is fine
    # drain: all healthy, delete everything, tree must be invariant-clean
    for node in nodes:
        sim.alg.set_healthy_node(node)
        for i in range(8):
            sim.alg.set_leaf_cell_healthy(node, i, True)
    for key in list(sim.pods):
        sim.delete_pod(key)
    check()


@pytest.mark.parametrize("seed", list(range(4)))
def test_fuzz_heterogeneous(seed):
    """Two-chain fuzz (MI355X + CT1 SKUs, typed and typeless requests):
    exercises chain iteration in scheduleNewAffinityGroup under churn."""
    from hivedscheduler_amd.api import config as apicfg
    from hivedscheduler_amd.api.types import (CellTypeSpec, Config, PhysicalCellSpec,
                                              PhysicalClusterSpec, VirtualCellSpec,
                                              VirtualClusterSpec, WebServerError)
    from hivedscheduler_amd.topo.mi355x import mi355x_cell_types, mi355x_node_cell

    ct = mi355x_cell_types()
    ct["CT1-NODE"] = CellTypeSpec(childCellType="CT1", childCellNumber=2, isNodeLevel=True)
    cfg = Config(
        physicalCluster=PhysicalClusterSpec(
            cellTypes=ct,
            physicalCells=[mi355x_node_cell("node1"), mi355x_node_cell("node2"),
                           PhysicalCellSpec(cellType="CT1-NODE", cellAddress="ct1"),
                           PhysicalCellSpec(cellType="CT1-NODE", cellAddress="ct2")]),
        virtualClusters={
            "X": VirtualClusterSpec(virtualCells=[
                VirtualCellSpec(cellType="MI355X-NODE", cellNumber=1),
                VirtualCellSpec(cellType="CT1-NODE", cellNumber=1)]),
            "Y": VirtualClusterSpec(virtualCells=[
                VirtualCellSpec(cellType="MI355X-NODE.MI355X-QUAD", cellNumber=2),
                VirtualCellSpec(cellType="CT1-NODE", cellNumber=1)]),
        })
    apicfg.infer_physical_cluster(cfg.physicalCluster)
    rng = random.Random(2000 + seed)
    sim = SimScheduler(cfg)
    check = sim.alg._core.check_invariants
    live, counter = {}, 0
    nodes = sim.alg.all_nodes()
    for step in range(300):
        op = rng.random()
        try:
            if op < 0.45:
                counter += 1
                key = f"h/p{counter}"
                kw = dict(vc=rng.choice(["X", "Y"]), priority=rng.choice([-1, 0, 1, 10]),
                          leaf_cells=rng.choice([1, 2, 2, 4]))
                if rng.random() < 0.5:
                    kw["leaf_cell_type"] = rng.choice(["MI355X", "CT1"])
                    if kw["leaf_cell_type"] == "CT1":
                        kw["leaf_cells"] = rng.choice([1, 2])
                r = sim.schedule(key, sim.pod_spec(**kw))
                if r.kind == "bind":
                    live[key] = kw
            elif op < 0.75 and live:
                key = rng.choice(list(live))
                del live[key]
                sim.delete_pod(key)
            else:
                sim.alg.set_leaf_cell_healthy(rng.choice(nodes),
                                              rng.randrange(2 if rng.random() < 0.5 else 8),
                                              rng.random() < 0.7)
            check()
        except WebServerError:
            continue
    for key in list(sim.pods):
        sim.delete_pod(key)
    check()


@pytest.mark.parametrize("seed", list(range(4)))
def test_fuzz_pinned_cells(seed):
    """Pinned-cell fuzz: requests with and without pinnedCellId against a
    config with a statically pinned node, under churn + health flapping."""
    from hivedscheduler_amd.api import config as apicfg
    from hivedscheduler_amd.api.types import (Config, PhysicalClusterSpec,
                                              PinnedCellSpec, VirtualCellSpec,
                                              VirtualClusterSpec, WebServerError)
    from hivedscheduler_amd.topo.mi355x import mi355x_cell_types, mi355x_node_cell

    n1 = mi355x_node_cell("node1")
    n1.pinnedCellId = "PIN1"
    cfg = Config(
        physicalCluster=PhysicalClusterSpec(
            cellTypes=mi355x_cell_types(),
            physicalCells=[n1, mi355x_node_cell("node2"), mi355x_node_cell("node3")]),
        virtualClusters={
            "P": VirtualClusterSpec(
                virtualCells=[VirtualCellSpec(cellType="MI355X-NODE", cellNumber=1)],
                pinnedCells=[PinnedCellSpec(pinnedCellId="PIN1")]),
            "Q": VirtualClusterSpec(
                virtualCells=[VirtualCellSpec(cellType="MI355X-NODE", cellNumber=1)]),
        })
    apicfg.infer_physical_cluster(cfg.physicalCluster)
    rng = random.Random(3000 + seed)
    sim = SimScheduler(cfg)
    check = sim.alg._core.check_invariants
    live, counter = {}, 0
    nodes = sim.alg.all_nodes()
    for step in range(300):
        op = rng.random()
        try:
            if op < 0.45:
                counter += 1
                key = f"pin/p{counter}"
                kw = dict(vc=rng.choice(["P", "P", "Q"]), priority=rng.choice([-1, 0, 1, 10]),
                          leaf_cells=rng.choice([1, 2, 4, 8]))
                if kw["vc"] == "P" and rng.random() < 0.5:
                    kw["pinned_cell_id"] = "PIN1"
                r = sim.schedule(key, sim.pod_spec(**kw))
                if r.kind == "bind":
                    live[key] = kw
                    if kw.get("pinned_cell_id"):
                        assert r.bind_info.node == "node1", r.bind_info.node
            elif op < 0.75 and live:
                key = rng.choice(list(live))
                del live[key]
                sim.delete_pod(key)
            elif op < 0.9:
                sim.alg.update_node(rng.choice(nodes), rng.random() < 0.7)
            else:
                sim.alg.set_leaf_cell_healthy(rng.choice(nodes), rng.randrange(8),
                                              rng.random() < 0.7)
            check()
        except WebServerError:
            continue
    for key in list(sim.pods):
        sim.delete_pod(key)
    check()


@pytest.mark.parametrize("seed", list(range(4)))
def test_fuzz_reconfiguration_mutation(seed):
    """Work-preserving reconfiguration fuzz: periodically restart into a
    MUTATED config (grown cluster / reshuffled quotas) and replay all live
    pods; mismatches must lazy-preempt or reject cleanly, never corrupt."""
    from hivedscheduler_amd.api.types import WebServerError

    rng = random.Random(4000 + seed)
    nnodes = 3
    vcs = {"VC1": [("MI355X-NODE", 1), ("MI355X-NODE.MI355X-QUAD", 1)],
           "VC2": [("MI355X-NODE", 1), ("MI355X-NODE.MI355X-QUAD.MI355X-PAIR", 2)]}
    sim = SimScheduler(mi355x_cluster_config(num_nodes=nnodes, vcs=vcs))
    check = sim.alg._core.check_invariants
    live, counter = {}, 0
    nodes = sim.alg.all_nodes()
    for step in range(300):
        op = rng.random()
        try:
            if op < 0.4:
                counter += 1
                key = f"rc/p{counter}"
                kw = dict(vc=rng.choice(list(vcs)), priority=rng.choice([-1, 0, 1, 10]),
                          leaf_cells=rng.choice([1, 2, 4, 8]))
                r = sim.schedule(key, sim.pod_spec(**kw))
                if r.kind == "bind":
                    live[key] = kw
            elif op < 0.65 and live:
                key = rng.choice(list(live))
                del live[key]
                sim.delete_pod(key)
            elif op < 0.9:
                sim.alg.update_node(rng.choice(nodes), rng.random() < 0.7)
            else:
                # mutate the cluster: grow, or shift a quad of quota
                mode = rng.randrange(3)
                if mode == 0:
                    nnodes += 1
                elif mode == 1:
                    vcs = {k: list(v) for k, v in vcs.items()}
                    vcs["VC1"] = vcs["VC1"] + [("MI355X-NODE.MI355X-QUAD", 1)]
                try:
                    new = SimScheduler(mi355x_cluster_config(num_nodes=nnodes, vcs=vcs))
                except WebServerError:
                    continue  # quota no longer fits; keep the old cluster
                for k, (sp, info) in list(sim.pods.items()):
                    try:
                        new.alg.add_allocated_pod(sp, info, k)
                        new.pods[k] = (sp, info)
                    except WebServerError:
                        pass
                sim = new
                check = sim.alg._core.check_invariants
                nodes = sim.alg.all_nodes()
                for n in nodes:
                    sim.alg.set_healthy_node(n)
                live = {k: live[k] for k in live if k in sim.pods}
            check()
        except WebServerError:
            continue
    for key in list(sim.pods):
        sim.delete_pod(key)
    check()


@pytest.mark.parametrize("seed", list(range(4)))
def test_fuzz_rack_chains(seed):
    """Five-level chain fuzz (leaf->pair->quad->node->rack): exercises buddy
    allocation and safety accounting at the pool level, plus crash-recovery
    replay on rack configs."""
    from hivedscheduler_amd.api import config as apicfg
    from hivedscheduler_amd.api.types import (Config, PhysicalCellSpec,
                                              PhysicalClusterSpec, VirtualCellSpec,
                                              VirtualClusterSpec, WebServerError)
    from hivedscheduler_amd.topo.mi355x import mi355x_cell_types, mi355x_node_cell

    rng = random.Random(5000 + seed)
    n_racks = rng.choice([2, 3])
    ct = mi355x_cell_types(pool_sizes=(2,))
    cells, idx = [], 0
    for _ in range(n_racks):
        kids = []
        for _ in range(2):
            idx += 1
            kids.append(mi355x_node_cell(f"node{idx}"))
        cells.append(PhysicalCellSpec(cellType="2-MI355X-NODE", cellChildren=kids))
    cfg = Config(
        physicalCluster=PhysicalClusterSpec(cellTypes=ct, physicalCells=cells),
        virtualClusters={
            "A": VirtualClusterSpec(virtualCells=[
                VirtualCellSpec(cellType="2-MI355X-NODE", cellNumber=n_racks - 1)]),
            "B": VirtualClusterSpec(virtualCells=[
                VirtualCellSpec(cellType="2-MI355X-NODE.MI355X-NODE", cellNumber=1),
                VirtualCellSpec(cellType="2-MI355X-NODE.MI355X-NODE.MI355X-QUAD", cellNumber=1)]),
        })
    apicfg.infer_physical_cluster(cfg.physicalCluster)
    sim = SimScheduler(cfg)
    check = sim.alg._core.check_invariants
    live, counter = {}, 0
    nodes = sim.alg.all_nodes()
    for step in range(300):
        op = rng.random()
        try:
            if op < 0.42:
                counter += 1
                key = f"rk/p{counter}"
                kw = dict(vc=rng.choice(["A", "B"]), priority=rng.choice([-1, 0, 1, 10]),
                          leaf_cells=rng.choice([1, 2, 4, 8]),
                          group=None)
                if rng.random() < 0.3:
                    kw["group"] = key
                    kw["members"] = [(2, kw["leaf_cells"])]
                else:
                    kw.pop("group")
                r = sim.schedule(key, sim.pod_spec(**kw))
                if r.kind == "bind":
                    live[key] = kw
            elif op < 0.7 and live:
                key = rng.choice(list(live))
                del live[key]
                sim.delete_pod(key)
            elif op < 0.9:
                sim.alg.update_node(rng.choice(nodes), rng.random() < 0.7)
            elif op < 0.97:
                sim.alg.set_leaf_cell_healthy(rng.choice(nodes), rng.randrange(8),
                                              rng.random() < 0.7)
            else:
                new = SimScheduler(cfg)
                for k, (sp, info) in list(sim.pods.items()):
                    try:
                        new.alg.add_allocated_pod(sp, info, k)
                        new.pods[k] = (sp, info)
                    except WebServerError:
                        pass
                sim = new
                check = sim.alg._core.check_invariants
                for n in nodes:
                    sim.alg.set_healthy_node(n)
                live = {k: live[k] for k in live if k in sim.pods}
            check()
        except WebServerError:
            continue
    for key in list(sim.pods):
        sim.delete_pod(key)
    check()
