"""Example configs parse and construct a working scheduler."""
import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_example_cluster_config():
    from hivedscheduler_amd.api import config as apicfg
    from hivedscheduler_amd.sim import SimScheduler

    cfg = apicfg.init_raw_config(os.path.join(REPO, "examples/config/mi355x-cluster.yaml"))
    sim = SimScheduler(cfg)
    r = sim.schedule("ns/j1", sim.pod_spec(vc="prod", leaf_cells=8))
    assert r.kind == "bind"
    r2 = sim.schedule("ns/j2", sim.pod_spec(vc="research", leaf_cells=4))
    assert r2.kind == "bind"


def test_standalone_entry_config_parses():
    # the deploy ConfigMap's embedded config also parses
    import yaml
    from hivedscheduler_amd.api.config import new_config
    from hivedscheduler_amd.algorithm import HivedAlgorithm

    docs = list(yaml.safe_load_all(open(os.path.join(REPO, "examples/deploy/hivedscheduler.yaml"))))
    cm = [d for d in docs if d and d.get("kind") == "ConfigMap"
          and d["metadata"]["name"] == "hivedscheduler-config"][0]
    cfg = new_config(yaml.safe_load(cm["data"]["hivedscheduler.yaml"]))
    alg = HivedAlgorithm(cfg)
    assert alg.all_nodes() == ["mi355x-node-1"]


def test_example_request_pods_parse_and_validate():
    """Every example request pod YAML parses into a valid scheduling spec —
    including the unmodified legacy-HiveD pod."""
    import glob

    import yaml

    from hivedscheduler_amd.internal import pod as podmod

    files = sorted(glob.glob(os.path.join(REPO, "examples/request/*.yaml")))
    assert len(files) >= 4
    for f in files:
        # multi-document files (e.g. the distributed-training gang) contain
        # several pods; every one must validate
        for p in yaml.safe_load_all(open(f)):
            assert podmod.is_hived_enabled(p), f
            spec = podmod.extract_pod_scheduling_spec(p)
            assert spec.virtualCluster and spec.leafCellNumber > 0, f
            if "gang" in f:
                assert spec.affinityGroup and spec.affinityGroup.members[0].podNumber == 2
        if "legacy" in f:
            assert spec.leafCellType == "MI355X" and spec.leafCellNumber == 4
        if "pinned" in f:
            assert spec.pinnedCellId
        if "opportunistic" in f:
            assert spec.priority == -1


def test_example_basic_and_design_configs():
    """The basic and heterogeneous design example configs parse into working
    schedulers; the design config exercises pinned cells + multi-chain VCs."""
    from hivedscheduler_amd.api import config as apicfg
    from hivedscheduler_amd.sim import SimScheduler

    basic = apicfg.init_raw_config(os.path.join(REPO, "examples/config/basic.yaml"))
    sim = SimScheduler(basic)
    assert sim.schedule("ns/b", sim.pod_spec(vc="default", leaf_cells=8)).kind == "bind"

    design = apicfg.init_raw_config(
        os.path.join(REPO, "examples/config/design-heterogeneous.yaml"))
    sim2 = SimScheduler(design)
    r = sim2.schedule("ns/pin", sim2.pod_spec(vc="VC1", leaf_cells=8,
                                              pinned_cell_id="VC1-PIN"))
    assert r.kind == "bind" and r.bind_info.node == "n4"
    r2 = sim2.schedule("ns/ct1", sim2.pod_spec(vc="VC2", leaf_cells=2,
                                               leaf_cell_type="CT1"))
    assert r2.kind == "bind" and r2.bind_info.node in ("c1", "c2")


def test_scenario_cli():
    """The scenario replay CLI runs the example preemption scenario with all
    expectations met (exit 0)."""
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "-m", "hivedscheduler_amd.sim",
         os.path.join(REPO, "examples/scenario-preemption.yaml")],
        capture_output=True, text=True, timeout=120, cwd=REPO)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "0 expectation failure(s)" in out.stdout


def test_scenario_degraded_link_cli():
    """The degraded-link scenario: gangs avoid the bad 0<->1 link under a
    suggested-node restriction, 1-GPU work still uses the endpoints, the
    healed link serves gangs again (exit 0)."""
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "-m", "hivedscheduler_amd.sim",
         os.path.join(REPO, "examples/scenario-degraded-link.yaml")],
        capture_output=True, text=True, timeout=120, cwd=REPO)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "0 expectation failure(s)" in out.stdout
