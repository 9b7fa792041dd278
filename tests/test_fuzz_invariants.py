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
