"""First-class xGMI link health (BASELINE north star).

A degraded link between two GPUs marks the LINK — multi-GPU placements avoid
co-placing its endpoints while both GPUs stay schedulable for 1-GPU work.
This is finer than the reference's healthiness model (cell.go:302-312), which
can only mark whole leaves/nodes bad. Also covers the hbmBytesPerCell request
extension (a GPU with measured HBM below the demand is avoided).
"""
import pytest

from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

GB = 1024 ** 3


def _mark_link(sim, node, a, b, healthy, gbps=0.0):
    sim.alg.set_xgmi_link_healthy(node, a, b, healthy, gbps)
    sim.alg._core.check_invariants()


def test_degraded_link_avoided_by_pair_request():
    """2-GPU guaranteed job avoids the 0<->1 pair while the link is bad."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False, 12.0)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=2))
    assert r.kind == "bind"
    cells = sorted(r.bind_info.leafCellIsolation)
    assert cells != [0, 1], "2-GPU job placed across the degraded 0<->1 link"
    # still a proper pair (LCA-minimal among clean pairs)
    assert cells in ([2, 3], [4, 5], [6, 7]), cells
    sim.alg._core.check_invariants()


def test_one_gpu_job_still_lands_on_degraded_pair_endpoints():
    """1-GPU work is unaffected by a degraded link — and is preferentially
    parked ON the degraded pair, keeping clean pairs free for gangs."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=1))
    assert r.kind == "bind"
    assert r.bind_info.leafCellIsolation[0] in (0, 1), (
        "1-GPU job should park on the degraded pair's endpoints")
    r2 = sim.schedule("ns/p2", sim.pod_spec(leaf_cells=1))
    assert r2.bind_info.leafCellIsolation[0] in (0, 1)
    sim.alg._core.check_invariants()


def test_degraded_link_avoided_by_opportunistic_request():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    r = sim.schedule("ns/op1", sim.pod_spec(leaf_cells=2, priority=-1))
    assert r.kind == "bind"
    assert sorted(r.bind_info.leafCellIsolation) != [0, 1]
    sim.alg._core.check_invariants()


def test_quad_request_uses_at_most_one_endpoint():
    """A 4-GPU job must not contain BOTH endpoints of the degraded link."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=4))
    assert r.kind == "bind"
    cells = set(r.bind_info.leafCellIsolation)
    assert not ({0, 1} <= cells), f"both endpoints co-placed: {sorted(cells)}"
    sim.alg._core.check_invariants()


def test_cross_quad_link_avoided():
    """Links are not limited to buddy pairs: a degraded 2<->5 link (LCA =
    node) forbids co-placing GPUs 2 and 5 in one gang."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 2, 5, False)
    # 6-GPU gang: clean capacity is 7 (drop one endpoint); must avoid {2,5}
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=6))
    assert r.kind == "bind"
    cells = set(r.bind_info.leafCellIsolation)
    assert not ({2, 5} <= cells), f"both endpoints co-placed: {sorted(cells)}"
    sim.alg._core.check_invariants()


def test_full_node_gang_still_schedules_dirty():
    """Capacity guarantees outrank link quality: an 8-GPU gang on a node with
    one degraded link still binds (clean capacity 7 < 8 -> dirty fallback)."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=8))
    assert r.kind == "bind"
    assert sorted(r.bind_info.leafCellIsolation) == list(range(8))
    sim.alg._core.check_invariants()


def test_gang_of_pods_respects_links_node_wide():
    """Gang-wide cleanliness: 3 pods x 2 GPUs on one node all communicate, so
    no pod may straddle the degraded link and no two pods may take one
    endpoint each... the endpoints simply can't both appear in the gang."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    spec = sim.pod_spec(leaf_cells=2, group="g1", members=[(3, 2)])
    cells = []
    for i in range(3):
        r = sim.schedule(f"ns/g1-{i}", spec)
        assert r.kind == "bind"
        cells.extend(r.bind_info.leafCellIsolation)
    assert not ({0, 1} <= set(cells)), f"gang co-placed endpoints: {sorted(cells)}"
    sim.alg._core.check_invariants()


def test_link_heal_restores_pair():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    _mark_link(sim, "node1", 0, 1, True, 153.0)
    # with the link healed, the packed pair [0,1] is the natural first pick
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=2))
    assert sorted(r.bind_info.leafCellIsolation) == [0, 1]
    links = sim.alg.get_xgmi_links("node1")
    assert links == [{"a": 0, "b": 1, "gbps": 153.0, "healthy": True}]
    sim.alg._core.check_invariants()


def test_multiple_degraded_links_max_independent_choice():
    """Bad links 0<->1 and 1<->2: dropping GPU 1 alone keeps 7 clean GPUs
    (max independent set), so a 6-GPU gang binds cleanly."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    _mark_link(sim, "node1", 1, 2, False)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=6))
    assert r.kind == "bind"
    cells = set(r.bind_info.leafCellIsolation)
    assert not ({0, 1} <= cells) and not ({1, 2} <= cells), sorted(cells)
    sim.alg._core.check_invariants()


def test_degraded_pair_taken_last_across_nodes():
    """2-node cluster, 16 GPUs, link 0<->1 on node1 degraded: the first 7
    pair requests all avoid the degraded pair; the 8th (no clean capacity
    left) takes it — capacity guarantees outrank link quality."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    _mark_link(sim, "node1", 0, 1, False)
    for i in range(7):
        r = sim.schedule(f"ns/p{i}", sim.pod_spec(leaf_cells=2))
        assert r.kind == "bind"
        if r.bind_info.node == "node1":
            assert sorted(r.bind_info.leafCellIsolation) != [0, 1], f"pod {i}"
    last = sim.schedule("ns/p7", sim.pod_spec(leaf_cells=2))
    assert last.kind == "bind"
    assert last.bind_info.node == "node1"
    assert sorted(last.bind_info.leafCellIsolation) == [0, 1]
    sim.alg._core.check_invariants()


def test_link_state_visible_in_inspect():
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False, 11.5)
    status = sim.alg.get_physical_cluster_status()

    def find_cells(c, out):
        out.append(c)
        for ch in c.get("cellChildren") or []:
            find_cells(ch, out)

    cells = []
    for c in status:
        find_cells(c, cells)
    flagged = [c for c in cells if c.get("badXgmiLinksUnder")]
    # the pair, its quad, and the node all report the degraded link under them
    assert len(flagged) == 3, [c.get("cellAddress") for c in flagged]
    # but none of them is "Bad" — the GPUs themselves are healthy
    assert all(c.get("cellHealthiness") == "Healthy" for c in flagged)
    assert sim.alg.get_xgmi_links("node1") == [
        {"a": 0, "b": 1, "gbps": 11.5, "healthy": False}]


def test_config_supplied_link_table():
    """Discovery-measured link tables in the physicalCells spec seed link
    state at construction (gbps recorded; unhealthy links degrade from the
    start)."""
    cfg = mi355x_cluster_config(num_nodes=1)
    node_cell = cfg.physicalCluster.physicalCells[0]
    node_cell.xgmiLinks = [
        {"a": 0, "b": 1, "gbps": 152.8, "healthy": True},
        {"a": 4, "b": 5, "gbps": 9.0, "healthy": False},
    ]
    sim = SimScheduler(cfg)
    links = {(l["a"], l["b"]): l for l in sim.alg.get_xgmi_links("node1")}
    assert links[(0, 1)]["healthy"] and links[(0, 1)]["gbps"] == 152.8
    assert not links[(4, 5)]["healthy"]
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=2))
    assert sorted(r.bind_info.leafCellIsolation) == [0, 1]
    r2 = sim.schedule("ns/p2", sim.pod_spec(leaf_cells=2))
    assert sorted(r2.bind_info.leafCellIsolation) == [2, 3]
    r3 = sim.schedule("ns/p3", sim.pod_spec(leaf_cells=2))
    assert sorted(r3.bind_info.leafCellIsolation) == [6, 7], (
        "third pair must skip the degraded 4<->5 pair")
    sim.alg._core.check_invariants()


def test_bad_leaf_and_bad_link_compose():
    """Leaf badness and link degradation are independent dimensions, and
    they COMPOSE on the guaranteed path: with GPU 7 bad and link 0<->1
    degraded, no single quad has 4 clean GPUs, so a clean placement must
    straddle quads. The physical clean-shape caps fed to the virtual
    descent (SchedulingRequest::physCleanCaps) make it pick that shape:
    the topology-preserving mapping then lands e.g. [0,4,5,6] — clean AND
    avoiding the bad GPU."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    sim.alg.set_leaf_cell_healthy("node1", 7, False)
    _mark_link(sim, "node1", 0, 1, False)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=4))
    assert r.kind == "bind"
    cells = set(r.bind_info.leafCellIsolation)
    assert 7 not in cells
    assert not ({0, 1} <= cells), sorted(cells)
    sim.alg._core.check_invariants()
    sim.delete_pod("ns/p1")
    # the opportunistic path (physical view, fully link-aware) places clean
    r2 = sim.schedule("ns/p2", sim.pod_spec(leaf_cells=4, priority=-1))
    assert r2.kind == "bind"
    cells2 = set(r2.bind_info.leafCellIsolation)
    assert 7 not in cells2
    assert not ({0, 1} <= cells2), sorted(cells2)
    sim.alg._core.check_invariants()


def test_clean_shape_caps_dont_break_capacity():
    """Caps only shape link-honoring attempts: with link 0<->1 degraded,
    two 4-GPU guaranteed jobs still fill the whole node (first clean quad,
    then the dirty remainder)."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    _mark_link(sim, "node1", 0, 1, False)
    r1 = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=4))
    assert sorted(r1.bind_info.leafCellIsolation) == [4, 5, 6, 7]
    r2 = sim.schedule("ns/p2", sim.pod_spec(leaf_cells=4))
    assert sorted(r2.bind_info.leafCellIsolation) == [0, 1, 2, 3]
    sim.alg._core.check_invariants()


# ---------------------------------------------------------------------------
# hbmBytesPerCell: measured HBM capacity as a scheduling constraint
# ---------------------------------------------------------------------------

def test_hbm_deficit_leaf_avoided():
    """A GPU with measured HBM below the request's demand is avoided."""
    cfg = mi355x_cluster_config(num_nodes=1)
    # GPU 0 reports only 280 GB (sick stack); the others the full 288
    leaf0 = cfg.physicalCluster.physicalCells[0].cellChildren[0].cellChildren[0].cellChildren[0]
    assert leaf0.cellAddress == "0"
    leaf0.hbmBytes = 280 * GB
    sim = SimScheduler(cfg)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=2, hbm_bytes_per_cell=288 * GB))
    assert r.kind == "bind"
    assert 0 not in r.bind_info.leafCellIsolation, r.bind_info.leafCellIsolation
    # a request without the demand still uses GPU 0
    rs = [sim.schedule(f"ns/q{i}", sim.pod_spec(leaf_cells=2)) for i in range(3)]
    used = [i for r2 in rs for i in r2.bind_info.leafCellIsolation]
    assert 0 in used
    sim.alg._core.check_invariants()


def test_hbm_demand_unsatisfiable_waits():
    cfg = mi355x_cluster_config(num_nodes=1)

    def leaves(spec):
        if not spec.cellChildren:
            return [spec]
        return [l for c in spec.cellChildren for l in leaves(c)]

    for leaf in leaves(cfg.physicalCluster.physicalCells[0]):
        leaf.hbmBytes = 280 * GB
    sim = SimScheduler(cfg)
    r = sim.schedule("ns/p1", sim.pod_spec(leaf_cells=1, hbm_bytes_per_cell=288 * GB))
    assert r.kind == "wait"
    sim.alg._core.check_invariants()


def test_hbm_negative_demand_rejected():
    from hivedscheduler_amd.api.types import WebServerError

    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    with pytest.raises(WebServerError):
        sim.schedule("ns/p1", sim.pod_spec(leaf_cells=1, hbm_bytes_per_cell=-1))


def test_clean_caps_honor_suggested_nodes():
    """Regression: the clean-shape caps must be computed over SUGGESTED
    nodes only — capacity on non-suggested nodes cannot satisfy the request,
    and counting it made the virtual descent pick unmappable-clean shapes
    that fell through to dirty placements (found via the degraded-link
    example scenario on the rack-chain config)."""
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    _mark_link(sim, "node1", 0, 1, False)
    sug = ["node1"]
    r1 = sim.schedule("ns/pair", sim.pod_spec(leaf_cells=2, ignore_suggested=False),
                      suggested=sug)
    assert r1.bind_info.node == "node1"
    assert sorted(r1.bind_info.leafCellIsolation) != [0, 1]
    # 4 GPUs with pairs [4,5] (wait, the pair landed on a clean pair) used:
    # no single quad of node1 has 4 clean free GPUs once its clean quad is
    # partially used -> the clean shape must straddle quads ON NODE1 (node2
    # capacity must not fool the caps)
    r2 = sim.schedule("ns/quad", sim.pod_spec(leaf_cells=4, ignore_suggested=False),
                      suggested=sug)
    assert r2.kind == "bind" and r2.bind_info.node == "node1"
    cells = set(r2.bind_info.leafCellIsolation)
    assert not ({0, 1} <= cells), sorted(cells)
    sim.alg._core.check_invariants()


def _random_link_case(rng, n_links):
    """Random 1-node cluster state for the property oracle: random 1-GPU
    occupancy, n_links random degraded links, one guaranteed request."""
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    for i in range(rng.randrange(0, 6)):
        r = sim.schedule(f"occ/p{i}", sim.pod_spec(leaf_cells=1))
        assert r.kind == "bind"
    used = {i for k, (sp, info) in sim.pods.items() for i in info.leafCellIsolation}
    free = set(range(8)) - used
    links = set()
    for _ in range(n_links):
        a, b = rng.sample(range(8), 2)
        links.add((min(a, b), max(a, b)))
    for a, b in links:
        sim.alg.set_xgmi_link_healthy("node1", a, b, False)
    q = rng.choice([2, 3, 4, 5, 6, 8])
    return sim, free, links, q


def _assert_dirty_only_when_forced(sim, free, links, q):
    import itertools

    r = sim.schedule("req/q", sim.pod_spec(leaf_cells=q))
    sim.alg._core.check_invariants()
    if r.kind != "bind":
        assert len(free) < q, f"wait with {len(free)} free >= q={q}: {r.wait_reason}"
        return
    cells = set(r.bind_info.leafCellIsolation)
    assert cells <= free and len(cells) == q
    if any({a, b} <= cells for a, b in links):
        # oracle: no clean subset of size q must exist among the free leaves
        for subset in itertools.combinations(sorted(free), q):
            ss = set(subset)
            if not any({a, b} <= ss for a, b in links):
                raise AssertionError(
                    f"dirty placement {sorted(cells)} but clean subset "
                    f"{subset} existed (links {sorted(links)}, free {sorted(free)})")


@pytest.mark.parametrize("seed", range(40))
def test_property_single_link_dirty_only_when_forced(seed):
    """Property (brute-force oracle), EXACT for one degraded link — the
    realistic failure mode: a guaranteed request binds DIRTY (contains both
    endpoints) only when NO clean free subset of its size exists. Verified
    over 4000 random cases offline (0 violations); 40 seeds run in CI."""
    import random

    rng = random.Random(9000 + seed)
    sim, free, links, q = _random_link_case(rng, 1)
    _assert_dirty_only_when_forced(sim, free, links, q)


@pytest.mark.parametrize("seed", range(30))
def test_property_multi_link_dirty_only_when_forced(seed):
    """Property (brute-force oracle) with SEVERAL simultaneous degraded
    links: clean-shape world enumeration (every max independent set of the
    bad-link graph) plus the physical-mirroring world descent make this
    exact too — verified over 6000 random 1-4-link cases offline with ZERO
    dirty-when-clean placements and zero spurious waits."""
    import random

    rng = random.Random(7000 + seed)
    sim, free, links, q = _random_link_case(rng, rng.choice([2, 3, 4]))
    _assert_dirty_only_when_forced(sim, free, links, q)


@pytest.mark.parametrize("seed", range(15))
def test_property_multi_node_dirty_only_when_forced(seed):
    """Multi-node oracle: on a 2-node cluster with random occupancy and
    random degraded links on either node, a single-node placement that
    straddles a degraded link is only allowed when NO node offers a clean
    same-size free subset (verified over 2000 random cases offline with
    zero violations)."""
    import itertools
    import random

    rng = random.Random(50000 + seed)
    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]}))
    for i in range(rng.randrange(0, 8)):
        sim.schedule(f"occ/p{i}", sim.pod_spec(leaf_cells=rng.choice([1, 1, 2])))
    free = {"node1": set(range(8)), "node2": set(range(8))}
    for k, (sp, info) in sim.pods.items():
        free[info.node] -= set(info.leafCellIsolation)
    links = {"node1": set(), "node2": set()}
    for _ in range(rng.randrange(1, 4)):
        node = rng.choice(["node1", "node2"])
        a, b = rng.sample(range(8), 2)
        links[node].add((min(a, b), max(a, b)))
        sim.alg.set_xgmi_link_healthy(node, a, b, False)
    q = rng.choice([2, 3, 4, 6, 8])
    r = sim.schedule("req/q", sim.pod_spec(leaf_cells=q))
    sim.alg._core.check_invariants()
    if r.kind != "bind":
        return
    node = r.bind_info.node
    cells = set(r.bind_info.leafCellIsolation)
    if any({a, b} <= cells for a, b in links[node]):
        for n2 in ("node1", "node2"):
            if len(free[n2]) < q:
                continue
            for ss in itertools.combinations(sorted(free[n2]), q):
                if not any({a, b} <= set(ss) for a, b in links[n2]):
                    raise AssertionError(
                        f"dirty {sorted(cells)} on {node} but clean {ss} on {n2}")


def test_pinned_cell_respects_degraded_link():
    """Link avoidance inside a PINNED cell: a 2-GPU pinned request avoids
    the degraded pair of its pinned node while the full 8-GPU pinned gang
    still binds (capacity over quality)."""
    from hivedscheduler_amd.api.types import PinnedCellSpec

    cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 1)]})
    cfg.physicalCluster.physicalCells[1].pinnedCellId = "VC1-PIN"
    cfg.virtualClusters["VC1"].pinnedCells = [PinnedCellSpec(pinnedCellId="VC1-PIN")]
    sim = SimScheduler(cfg)
    _mark_link(sim, "node2", 0, 1, False)
    r = sim.schedule("ns/pin2", sim.pod_spec(leaf_cells=2, pinned_cell_id="VC1-PIN"))
    assert r.kind == "bind" and r.bind_info.node == "node2"
    assert sorted(r.bind_info.leafCellIsolation) != [0, 1], r.bind_info.leafCellIsolation
    sim.delete_pod("ns/pin2")
    r8 = sim.schedule("ns/pin8", sim.pod_spec(leaf_cells=8, pinned_cell_id="VC1-PIN"))
    assert r8.kind == "bind" and sorted(r8.bind_info.leafCellIsolation) == list(range(8))
    sim.alg._core.check_invariants()


def test_opportunistic_hbm_demand():
    """hbmBytesPerCell applies on the opportunistic (physical-view) path."""
    cfg = mi355x_cluster_config(num_nodes=1)
    leaf0 = cfg.physicalCluster.physicalCells[0].cellChildren[0].cellChildren[0].cellChildren[0]
    leaf0.hbmBytes = 200 * GB
    sim = SimScheduler(cfg)
    r = sim.schedule("ns/ot", sim.pod_spec(leaf_cells=4, priority=-1,
                                           hbm_bytes_per_cell=288 * GB))
    assert r.kind == "bind"
    assert 0 not in r.bind_info.leafCellIsolation
    sim.alg._core.check_invariants()


@pytest.mark.parametrize("seed", range(20))
def test_property_gang_union_dirty_only_when_forced(seed):
    """Gang oracle: ALL pods of an affinity group communicate (the group's
    collective), so the UNION of their placements must avoid degraded links
    unless no clean union of that size exists — exact over 3000 random
    multi-pod gang cases offline (found and fixed two session-state bugs:
    non-recursive hint availability and cross-pod hint collisions)."""
    import itertools
    import random

    rng = random.Random(300000 + seed)
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    for i in range(rng.randrange(0, 3)):
        sim.schedule(f"occ/p{i}", sim.pod_spec(leaf_cells=1))
    used = {i for k, (sp, info) in sim.pods.items() for i in info.leafCellIsolation}
    free = set(range(8)) - used
    links = set()
    for _ in range(rng.randrange(1, 4)):
        a, b = rng.sample(range(8), 2)
        links.add((min(a, b), max(a, b)))
    for a, b in links:
        sim.alg.set_xgmi_link_healthy("node1", a, b, False)
    pods, size = rng.choice([(2, 2), (2, 3), (3, 2), (2, 4), (4, 2)])
    spec = sim.pod_spec(leaf_cells=size, group="g", members=[(pods, size)])
    cells = []
    for i in range(pods):
        r = sim.schedule(f"g/p{i}", spec)
        if r.kind != "bind":
            sim.alg._core.check_invariants()
            return
        cells.extend(r.bind_info.leafCellIsolation)
    sim.alg._core.check_invariants()
    cset = set(cells)
    if any({a, b} <= cset for a, b in links):
        q = pods * size
        for ss in itertools.combinations(sorted(free), q):
            if not any({a, b} <= set(ss) for a, b in links):
                raise AssertionError(
                    f"gang union {sorted(cells)} dirty but clean {ss} existed "
                    f"(links {sorted(links)})")


@pytest.mark.parametrize("seed", range(15))
def test_property_preemptive_dirty_only_when_forced(seed):
    """Preemptive-tier oracle: a guaranteed request that must preempt
    opportunistic squatters still places link-clean whenever ANY clean
    same-size subset exists among free + preemptible GPUs (exact over 2000
    random cases offline; link-clean placements outrank sparing an
    opportunistic pod)."""
    import itertools
    import random

    rng = random.Random(900000 + seed)
    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    for i in range(rng.randrange(2, 7)):
        sim.schedule(f"ot/p{i}", sim.pod_spec(leaf_cells=1, priority=-1))
    links = set()
    for _ in range(rng.randrange(1, 4)):
        a, b = rng.sample(range(8), 2)
        links.add((min(a, b), max(a, b)))
    for a, b in links:
        sim.alg.set_xgmi_link_healthy("node1", a, b, False)
    q = rng.choice([2, 3, 4, 6])
    r = sim.run_preemption_to_completion("req/q", sim.pod_spec(leaf_cells=q, priority=0))
    sim.alg._core.check_invariants()
    if r.kind != "bind":
        return
    cells = set(r.bind_info.leafCellIsolation)
    if any({a, b} <= cells for a, b in links):
        for ss in itertools.combinations(range(8), q):
            if not any({a, b} <= set(ss) for a, b in links):
                raise AssertionError(
                    f"preemptive placement {sorted(cells)} dirty but clean "
                    f"{ss} existed (links {sorted(links)})")
