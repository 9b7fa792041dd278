#!/usr/bin/env python3
"""Placement-quality oracle soak: brute-force verification that link-clean
placement is EXACT.

For each random case (random 1-GPU occupancy, N degraded xGMI links, one
guaranteed request) the oracle enumerates EVERY same-size subset of the free
GPUs and asserts:
  - a placement straddling a degraded link ("dirty") is only produced when
    NO clean subset of that size exists at all;
  - the request never waits while enough free GPUs exist;
  - the core's full invariant check passes after every case.

This reproduces the round-2 claims in BENCHMARKS.md (zero dirty-when-clean
over 19,000+ random cases with 1-8 simultaneous degraded links).

Usage: python bench_oracle.py [--cases 2000] [--links-min 1] [--links-max 4]
                              [--nodes 1] [--seed-base 0]
"""
import argparse
import itertools
import random
import sys


def run(cases, links_min, links_max, nodes, seed_base):
    from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

    dirty_when_clean = bad_waits = dirty_forced = clean = waits = 0
    for case in range(cases):
        rng = random.Random(seed_base + case)
        vcs = {"VC1": [("MI355X-NODE", nodes)]}
        sim = SimScheduler(mi355x_cluster_config(num_nodes=nodes, vcs=vcs))
        for i in range(rng.randrange(0, 6)):
            sim.schedule(f"occ/p{i}", sim.pod_spec(leaf_cells=rng.choice([1, 1, 2])))
        free = {f"node{n + 1}": set(range(8)) for n in range(nodes)}
        for k, (sp, info) in sim.pods.items():
            free[info.node] -= set(info.leafCellIsolation)
        links = {f"node{n + 1}": set() for n in range(nodes)}
        for _ in range(rng.randrange(links_min, links_max + 1)):
            node = f"node{rng.randrange(nodes) + 1}"
            a, b = rng.sample(range(8), 2)
            links[node].add((min(a, b), max(a, b)))
            sim.alg.set_xgmi_link_healthy(node, a, b, False)
        q = rng.choice([2, 3, 4, 5, 6, 8])
        r = sim.schedule("req/q", sim.pod_spec(leaf_cells=q))
        sim.alg._core.check_invariants()
        if r.kind != "bind":
            waits += 1
            if any(len(f) >= q for f in free.values()):
                bad_waits += 1
                print(f"BAD WAIT case={seed_base + case} q={q}")
            continue
        node = r.bind_info.node
        cells = set(r.bind_info.leafCellIsolation)
        if any({a, b} <= cells for a, b in links[node]):
            clean_exists = any(
                not any({a, b} <= set(ss) for a, b in links[n2])
                for n2 in free
                if len(free[n2]) >= q
                for ss in itertools.combinations(sorted(free[n2]), q))
            if clean_exists:
                dirty_when_clean += 1
                print(f"DIRTY-WHEN-CLEAN case={seed_base + case} "
                      f"placed={sorted(cells)}@{node} links={links}")
            else:
                dirty_forced += 1
        else:
            clean += 1
    print(f"cases={cases} nodes={nodes} links={links_min}-{links_max}: "
          f"clean={clean} dirty_forced={dirty_forced} waits={waits} | "
          f"dirty_when_clean={dirty_when_clean} bad_waits={bad_waits}")
    return dirty_when_clean + bad_waits


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cases", type=int, default=2000)
    ap.add_argument("--links-min", type=int, default=1)
    ap.add_argument("--links-max", type=int, default=4)
    ap.add_argument("--nodes", type=int, default=1)
    ap.add_argument("--seed-base", type=int, default=0)
    args = ap.parse_args()
    sys.exit(1 if run(args.cases, args.links_min, args.links_max,
                      args.nodes, args.seed_base) else 0)


if __name__ == "__main__":
    main()
