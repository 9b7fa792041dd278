"""Scenario replay CLI: evaluate a cluster config against a scripted
workload without Kubernetes.

    python -m hivedscheduler_amd.sim scenario.yaml [-v]

Scenario YAML:

    config: examples/config/mi355x-cluster.yaml   # or inline `cluster: {...}`
    events:
    - submit: {pod: a/train-0, vc: prod, priority: 100, leafCellNumber: 4,
               group: a/train, members: [{podNumber: 2, leafCellNumber: 4}]}
      expect: bind            # optional: bind | wait | preempt
      expectNode: node1       # optional
    - delete: {pod: a/train-0}
    - nodeHealth: {node: node2, healthy: false}
    - leafHealth: {node: node1, leafIndex: 3, healthy: true}
    - linkHealth: {node: node1, a: 0, b: 1, healthy: false, gbps: 12.5}
    - preemptToCompletion: {pod: b/prod-0, vc: prod, priority: 200,
                            leafCellNumber: 8}

Submit events may also assert `expectCellsNot: [0, 1]` (the placement must
not contain ALL of the listed GPU indices — e.g. both endpoints of a
degraded link) and request `hbmBytesPerCell`.

Exit code 0 when every `expect` matched, 1 otherwise. This is the same
in-memory path the test suite and bench.py drive (SURVEY.md §4's K8s-free
test strategy), packaged for operators evaluating VC configs.
"""
from __future__ import annotations

import argparse
import sys

import yaml

from ..api import config as apicfg
from ..api.types import AffinityGroupMemberSpec, AffinityGroupSpec, PodSchedulingSpec, WebServerError
from .harness import SimScheduler


def build_spec(ev: dict) -> PodSchedulingSpec:
    spec = PodSchedulingSpec(
        virtualCluster=ev.get("vc", ""),
        priority=int(ev.get("priority", 0)),
        pinnedCellId=ev.get("pinnedCellId", ""),
        leafCellType=ev.get("leafCellType", ""),
        leafCellNumber=int(ev.get("leafCellNumber", 1)),
        lazyPreemptionEnable=bool(ev.get("lazyPreemptionEnable", False)),
        ignoreK8sSuggestedNodes=bool(ev.get("ignoreK8sSuggestedNodes", True)),
        hbmBytesPerCell=int(ev.get("hbmBytesPerCell", 0)),
    )
    if ev.get("group"):
        spec.affinityGroup = AffinityGroupSpec(
            name=ev["group"],
            members=[AffinityGroupMemberSpec(podNumber=int(m["podNumber"]),
                                             leafCellNumber=int(m["leafCellNumber"]))
                     for m in ev.get("members", [{"podNumber": 1,
                                                  "leafCellNumber": spec.leafCellNumber}])],
        )
    return spec


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="hivedscheduler-amd-sim")
    ap.add_argument("scenario", help="scenario YAML file")
    ap.add_argument("-v", "--verbose", action="store_true")
    args = ap.parse_args(argv)

    with open(args.scenario) as f:
        sc = yaml.safe_load(f)

    if "config" in sc:
        cfg = apicfg.init_raw_config(sc["config"])
    elif "cluster" in sc:
        cfg = apicfg.new_config(sc["cluster"])
    else:
        print("scenario needs 'config: <path>' or inline 'cluster:'", file=sys.stderr)
        return 2

    sim = SimScheduler(cfg)
    failures = 0
    for i, event in enumerate(sc.get("events", [])):
        try:
            if "submit" in event or "preemptToCompletion" in event:
                full_protocol = "preemptToCompletion" in event
                ev = event.get("submit") or event["preemptToCompletion"]
                key = ev["pod"]
                spec = build_spec(ev)
                if full_protocol:
                    r = sim.run_preemption_to_completion(key, spec,
                                                         suggested=ev.get("suggestedNodes"))
                else:
                    r = sim.schedule(key, spec, suggested=ev.get("suggestedNodes"))
                out = f"[{i}] {key}: {r.kind}"
                if r.kind == "bind":
                    out += f" {r.bind_info.node} GPUs {r.bind_info.leafCellIsolation}"
                elif r.kind == "preempt":
                    out += f" victims {sorted(r.victim_pod_keys)}"
                print(out)
                want = event.get("expect")
                if want and r.kind != want:
                    print(f"    EXPECT FAILED: wanted {want}", file=sys.stderr)
                    failures += 1
                want_node = event.get("expectNode")
                if want_node and (r.kind != "bind" or r.bind_info.node != want_node):
                    print(f"    EXPECT FAILED: wanted node {want_node}", file=sys.stderr)
                    failures += 1
                cells_not = event.get("expectCellsNot")
                if cells_not and r.kind == "bind" and set(
                        int(c) for c in cells_not) <= set(r.bind_info.leafCellIsolation):
                    print(f"    EXPECT FAILED: placement contains all of {cells_not} "
                          "(e.g. both endpoints of a degraded link)", file=sys.stderr)
                    failures += 1
            elif "delete" in event:
                sim.delete_pod(event["delete"]["pod"])
                print(f"[{i}] deleted {event['delete']['pod']}")
            elif "nodeHealth" in event:
                ev = event["nodeHealth"]
                sim.alg.update_node(ev["node"], bool(ev["healthy"]))
                print(f"[{i}] node {ev['node']} healthy={ev['healthy']}")
            elif "leafHealth" in event:
                ev = event["leafHealth"]
                sim.alg.set_leaf_cell_healthy(ev["node"], int(ev["leafIndex"]),
                                              bool(ev["healthy"]))
                print(f"[{i}] leaf {ev['node']}/{ev['leafIndex']} healthy={ev['healthy']}")
            elif "linkHealth" in event:
                ev = event["linkHealth"]
                sim.alg.set_xgmi_link_healthy(ev["node"], int(ev["a"]), int(ev["b"]),
                                              bool(ev["healthy"]),
                                              float(ev.get("gbps", 0.0)))
                print(f"[{i}] link {ev['node']}/{ev['a']}<->{ev['b']} "
                      f"healthy={ev['healthy']}")
            else:
                print(f"[{i}] unknown event {sorted(event)}", file=sys.stderr)
                failures += 1
            sim.alg._core.check_invariants()
        except WebServerError as e:
            print(f"[{i}] rejected: {e}")
            if event.get("expect") not in (None, "reject"):
                failures += 1
            elif event.get("expect") is None:
                failures += 1
        if args.verbose:
            for g in sim.alg.get_all_affinity_groups():
                print(f"      group {g['name']}: {g['state']} {g.get('physicalPlacement')}")
    print(f"done: {failures} expectation failure(s); "
          f"{len(sim.alg.get_all_affinity_groups())} group(s) allocated")
    return 1 if failures else 0


if __name__ == "__main__":
    raise SystemExit(main())
