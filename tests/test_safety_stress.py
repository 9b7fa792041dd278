"""Fragmentation stress (BASELINE config 4): 64 mixed 1/2/4-GPU requests on a
simulated 4-node x 8-MI355X cluster; VC-safety violations must be 0.

VC safety invariant (reference README.md:21-23, hived_algorithm.go:1378-1385):
a VC can always allocate its guaranteed cells, regardless of other VCs'
workloads — here asserted constructively by scheduling each VC's full quota
after arbitrary churn from other VCs.
"""
import random

import pytest

from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config


VCS = {
    "VC1": [("MI355X-NODE", 2)],                    # 2 whole nodes
    "VC2": [("MI355X-NODE", 1), ("MI355X-NODE.MI355X-QUAD", 1)],  # 1 node + 1 quad
    "VC3": [("MI355X-NODE.MI355X-QUAD", 1)],        # 1 quad
}  # total: 3 nodes + 2 quads over 4 physical nodes


def make_sim():
    return SimScheduler(mi355x_cluster_config(num_nodes=4, vcs=VCS))


def churn(sim, rng, rounds=64):
    """Random mixed 1/2/4-GPU requests (guaranteed within quota is not forced:
    opportunistic and guaranteed mixed) with random deletions."""
    live = []
    for i in range(rounds):
        vc = rng.choice(list(VCS))
        cells = rng.choice([1, 1, 2, 2, 4])
        prio = rng.choice([-1, -1, 0, 1])
        key = f"ns/churn-{i}"
        r = sim.schedule(key, sim.pod_spec(vc=vc, priority=prio, leaf_cells=cells))
        if r.kind == "bind":
            live.append(key)
        if live and rng.random() < 0.3:
            victim = live.pop(rng.randrange(len(live)))
            sim.delete_pod(victim)
    return live


@pytest.mark.parametrize("seed", [0, 1, 2, 3])
def test_vc_safety_under_fragmentation(seed):
    sim = make_sim()
    rng = random.Random(seed)
    live = churn(sim, rng)
    # Drain all opportunistic + churn pods, then every VC must be able to
    # allocate its FULL guaranteed quota (zero safety violations).
    for key in live:
        sim.delete_pod(key)
    sim.assert_empty()
    quota_requests = {
        "VC1": [8, 8],
        "VC2": [8, 4],
        "VC3": [4],
    }
    for vc, sizes in quota_requests.items():
        for j, cells in enumerate(sizes):
            r = sim.schedule(f"ns/{vc}-quota-{j}", sim.pod_spec(vc=vc, priority=0, leaf_cells=cells))
            assert r.kind == "bind", f"VC-safety violation: {vc} request {j} ({cells} GPUs): {r}"


@pytest.mark.parametrize("seed", [10, 11])
def test_vc_safety_with_live_guaranteed_load(seed):
    """Safety also holds while OTHER VCs keep their guaranteed load running:
    VC1 must always get its 2 nodes with only opportunistic jobs evictable."""
    sim = make_sim()
    rng = random.Random(seed)
    # VC2+VC3 run their full guaranteed quota
    for vc, sizes in (("VC2", [8, 4]), ("VC3", [4])):
        for j, cells in enumerate(sizes):
            assert sim.schedule(f"ns/{vc}-{j}", sim.pod_spec(vc=vc, priority=0, leaf_cells=cells)).kind == "bind"
    # random opportunistic load
    for i in range(16):
        sim.schedule(f"ns/ot-{i}", sim.pod_spec(vc=rng.choice(list(VCS)), priority=-1,
                                                leaf_cells=rng.choice([1, 2, 4])))
    # VC1's guaranteed quota must be obtainable via (at most) preemption of
    # opportunistic pods
    for j in range(2):
        spec = sim.pod_spec(vc="VC1", priority=0, leaf_cells=8)
        r = sim.run_preemption_to_completion(f"ns/vc1-{j}", spec)
        assert r.kind == "bind", f"VC-safety violation for VC1 node {j}: {r}"


def test_buddy_packing_prevents_fragmentation():
    """1-GPU jobs then a 4-GPU job: packing must keep a quad free."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    for i in range(4):
        assert sim.schedule(f"ns/s{i}", sim.pod_spec(leaf_cells=1)).kind == "bind"
    r = sim.schedule("ns/q", sim.pod_spec(leaf_cells=4))
    assert r.kind == "bind"
    assert sorted(r.bind_info.leafCellIsolation) == [4, 5, 6, 7]


def test_schedule_latency_scales_to_large_cluster():
    """Perf guard: p50 of a filter decision stays sub-millisecond on a
    128-node (1024-GPU) simulated cluster (generous CI bound; measured
    ~0.13 ms — see profiles/sched_scaling_r01.md)."""
    import random
    import time

    from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=128, vcs={"VC1": [("MI355X-NODE", 64)],
                            "VC2": [("MI355X-NODE", 64)]}))
    rng = random.Random(0)
    lat = []
    for i in range(256):
        spec = sim.pod_spec(vc=rng.choice(["VC1", "VC2"]), priority=0,
                            leaf_cells=rng.choice([1, 2, 4]))
        t0 = time.perf_counter_ns()
        sim.schedule(f"s/p{i}", spec)
        lat.append((time.perf_counter_ns() - t0) / 1e6)
    lat.sort()
    assert lat[len(lat) // 2] < 5.0, f"p50 regressed: {lat[len(lat)//2]:.3f} ms"


def test_trace_replay_small():
    """100-job OSDI'20-style trace replay: zero VC-safety violations, all
    invariant checks pass, every job eventually completes or is queued."""
    from bench_trace import TraceReplay

    rep = TraceReplay(nodes=4, seed=7, invariant_every=100).run(100)
    assert rep["vc_safety_violations"] == 0
    assert rep["invariant_checks"] > 0
    assert rep["binds"] > 0 and rep["completions"] > 0
