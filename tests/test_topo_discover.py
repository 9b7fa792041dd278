"""Measured-topology discovery consumed end-to-end: rocm-topo-discover YAML
-> Config -> live scheduler with per-link gbps and per-leaf hbmBytes."""
import textwrap

from hivedscheduler_amd.sim import SimScheduler
from hivedscheduler_amd.topo.discover import (
    cluster_config_from_discovery,
    parse_discovery_output,
)

GB = 1024 ** 3

# canned output in the exact shape native/rocm_topo_discover.cpp emits for an
# 8-GPU MI355X node (abridged link table; GPU 5 with a VRAM deficit and the
# 4<->5 link measured degraded)
CANNED = textwrap.dedent("""\
    # rocm-topo-discover: node gpunode1, 8 GPUs, xGMI links 56, fullMesh=true, symmetric=true
    nodeName: gpunode1
    numGpus: 8
    gpus:
    - index: 0
      vramBytes: 309237645312
      xgmiPeers: [1, 2, 3, 4, 5, 6, 7]
    physicalCells:
    - cellType: MI355X-NODE
      cellAddress: gpunode1
      xgmiLinks:
      - {a: 0, b: 1, gbps: 152.8, healthy: true}
      - {a: 2, b: 3, gbps: 151.9, healthy: true}
      - {a: 4, b: 5, gbps: 17.2, healthy: false}
      cellChildren:
      - cellChildren:  # MI355X-QUAD
        - cellChildren:  # MI355X-PAIR
          - cellAddress: 0
            hbmBytes: 309237645312
          - cellAddress: 1
            hbmBytes: 309237645312
        - cellChildren:  # MI355X-PAIR
          - cellAddress: 2
            hbmBytes: 309237645312
          - cellAddress: 3
            hbmBytes: 309237645312
      - cellChildren:  # MI355X-QUAD
        - cellChildren:  # MI355X-PAIR
          - cellAddress: 4
            hbmBytes: 309237645312
          - cellAddress: 5
            hbmBytes: 300647710720
        - cellChildren:  # MI355X-PAIR
          - cellAddress: 6
            hbmBytes: 309237645312
          - cellAddress: 7
            hbmBytes: 309237645312
""")


def test_discovery_yaml_end_to_end():
    doc = parse_discovery_output(CANNED)
    assert doc["nodeName"] == "gpunode1"
    cfg = cluster_config_from_discovery([doc])
    sim = SimScheduler(cfg)
    assert sim.alg.all_nodes() == ["gpunode1"]
    # the measured link table seeded core state: 4<->5 degraded with gbps
    links = {(l["a"], l["b"]): l for l in sim.alg.get_xgmi_links("gpunode1")}
    assert links[(0, 1)] == {"a": 0, "b": 1, "gbps": 152.8, "healthy": True}
    assert not links[(4, 5)]["healthy"] and links[(4, 5)]["gbps"] == 17.2
    # placements honor the measured facts: a pair request avoids 4<->5
    for i in range(3):
        r = sim.schedule(f"ns/p{i}", sim.pod_spec(leaf_cells=2))
        assert r.kind == "bind"
        assert sorted(r.bind_info.leafCellIsolation) != [4, 5]
    # measured per-leaf HBM flows into scheduling: GPU 5 (280 GB) avoided by
    # a full-capacity demand
    sim2 = SimScheduler(cluster_config_from_discovery([parse_discovery_output(CANNED)]))
    r = sim2.schedule("ns/h1", sim2.pod_spec(leaf_cells=4, hbm_bytes_per_cell=288 * GB))
    assert r.kind == "bind"
    assert 5 not in r.bind_info.leafCellIsolation
    sim.alg._core.check_invariants()


def test_discovery_multi_node_merge():
    doc1 = parse_discovery_output(CANNED)
    doc2 = parse_discovery_output(CANNED.replace("gpunode1", "gpunode2"))
    cfg = cluster_config_from_discovery([doc1, doc2], vcs={"VC1": [("MI355X-NODE", 2)]})
    sim = SimScheduler(cfg)
    assert sorted(sim.alg.all_nodes()) == ["gpunode1", "gpunode2"]
    r = sim.schedule("ns/g0", sim.pod_spec(leaf_cells=8))
    assert r.kind == "bind"
