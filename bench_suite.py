#!/usr/bin/env python3
"""Full benchmark suite: every BASELINE.json config, one JSON report.

bench.py remains the driver-contract flagship (config 4); this suite measures
all five named configs so the round report covers the whole baseline:

  1 single VC, one 2-GPU cell request, simulated 1-node/8-GPU cluster
  2 8-GPU gang (affinity group size 8) guaranteed bind [+ RCCL probe on GPU]
  3 two VCs on one node: guaranteed 4-GPU preempts opportunistic 4-GPU
  4 fragmentation stress (64 mixed requests, 4 nodes) — same as bench.py
  5 one MI355X marked unhealthy: placement avoids it + replay keeps jobs

All scheduling paths run the C++ core (in-memory, no K8s), exactly like the
real extender does. Latencies are per-decision, in milliseconds.

Usage: python bench_suite.py [--iters 200] [--out report.json]
"""
import argparse
import json
import statistics
import sys
import time


def pcts(samples):
    s = sorted(samples)
    return {
        "p50_ms": round(s[len(s) // 2], 6),
        "p99_ms": round(s[min(len(s) - 1, int(len(s) * 0.99))], 6),
        "mean_ms": round(statistics.fmean(s), 6),
        "n": len(s),
    }


def timed(fn):
    t0 = time.perf_counter_ns()
    r = fn()
    return r, (time.perf_counter_ns() - t0) / 1e6


def config1(iters):
    """Single VC, one 2-GPU cell request on a 1-node/8-GPU simulated cluster."""
    from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    lat = []
    for i in range(-5, iters):  # 5 untimed warmup iterations
        spec = sim.pod_spec(vc="VC1", priority=0, leaf_cells=2)
        r, ms = timed(lambda: sim.schedule(f"c1/p{i}", spec))
        assert r.kind == "bind" and len(r.bind_info.leafCellIsolation) == 2
        # pair-aligned placement (xGMI pair = indices {2k, 2k+1})
        a, b = sorted(r.bind_info.leafCellIsolation)
        assert b == a + 1 and a % 2 == 0, r.bind_info.leafCellIsolation
        if i >= 0:
            lat.append(ms)
        sim.delete_pod(f"c1/p{i}")
    return {"desc": "single VC, one 2-GPU cell, 1 node", "latency": pcts(lat),
            "placement": "xGMI-pair aligned"}


def config2(iters, gpu):
    """8-GPU gang job, guaranteed priority; on a GPU box, validate the placed
    cell with the RCCL probe (sizes limited to visible GPUs)."""
    from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

    sim = SimScheduler(mi355x_cluster_config(num_nodes=1))
    lat = []
    for i in range(iters):
        spec = sim.pod_spec(vc="VC1", priority=100, leaf_cells=1,
                            group=f"gang8-{i}", members=[(8, 1)])
        keys = [f"c2/g{i}p{j}" for j in range(8)]
        t0 = time.perf_counter_ns()
        results = [sim.schedule(k, spec) for k in keys]
        lat.append((time.perf_counter_ns() - t0) / 1e6)
        assert all(r.kind == "bind" for r in results)
        nodes = {r.bind_info.node for r in results}
        assert len(nodes) == 1, f"gang split across {nodes}"
        for k in keys:
            sim.delete_pod(k)
    out = {"desc": "8-GPU gang bind (8 filter decisions, one node)",
           "gang_latency": pcts(lat)}
    if gpu:
        import torch

        from hivedscheduler_amd.probe import CellProbeRunner

        runner = CellProbeRunner()
        if runner.available():
            n = torch.cuda.device_count()
            curve = {}
            for size in (1, 2, 4, 8):
                if size <= n:
                    p = runner.probe_cell(list(range(size)), size_mb=64, iters=10)
                    if p.get("ok"):
                        if size == 1:
                            # a 1-device probe never leaves the GPU: it is an
                            # HBM copy floor, not an xGMI number
                            out["hbm_copy_gbps"] = round(p["busbw_gbps"], 2)
                        else:
                            curve[str(size)] = round(p["busbw_gbps"], 2)
            out["rccl_busbw_gbps"] = curve
    return out


def config3(iters):
    """Inter-VC preemption: guaranteed 4-GPU evicts opportunistic 4-GPU.
    Latency = full protocol (preempt decision + victim delete + re-filter)."""
    from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

    sim = SimScheduler(mi355x_cluster_config(
        num_nodes=1, vcs={"VC1": [("MI355X-NODE.MI355X-QUAD", 1)],
                          "VC2": [("MI355X-NODE.MI355X-QUAD", 1)]}))
    lat = []
    for i in range(iters):
        assert sim.schedule(f"c3/o{i}", sim.pod_spec(vc="VC2", priority=-1,
                                                     leaf_cells=8)).kind == "bind"
        spec = sim.pod_spec(vc="VC1", priority=10, leaf_cells=4)
        r, ms = timed(lambda: sim.run_preemption_to_completion(f"c3/g{i}", spec))
        assert r.kind == "bind"
        lat.append(ms)
        for k in list(sim.pods):
            sim.delete_pod(k)
    return {"desc": "guaranteed 4-GPU preempts opportunistic (full protocol)",
            "preemption_latency": pcts(lat)}


def config4(iters):
    """Fragmentation stress — identical workload to bench.py."""
    import bench

    sim = bench.make_sim()
    requests = bench.build_requests(seed=0, count=64)
    lat = []
    for _ in range(max(1, iters // 10)):
        bench.run_round(sim, requests, lat)
    violations = 0
    for key in list(sim.pods):
        sim.delete_pod(key)
    for vc, sizes in (("VC1", [8, 8]), ("VC2", [8, 4]), ("VC3", [4])):
        for j, cells in enumerate(sizes):
            r = sim.schedule(f"c4/safety-{vc}-{j}",
                             sim.pod_spec(vc=vc, priority=0, leaf_cells=cells))
            violations += r.kind != "bind"
    return {"desc": "64 mixed 1/2/4-GPU requests, 4-node x 8-GPU",
            "latency": pcts(lat), "vc_safety_violations": violations}


def config5(iters, gpu):
    """Bad-hardware awareness: one GPU marked unhealthy (as the health agent
    does from rocm-smi/HIP-kernel evidence) -> placement avoids its pair cell;
    restart replay keeps running jobs (work-preserving reconfiguration)."""
    from hivedscheduler_amd.sim import SimScheduler, mi355x_cluster_config

    lat = []
    for i in range(max(1, iters // 10)):
        sim = SimScheduler(mi355x_cluster_config(num_nodes=2,
                                                 vcs={"VC1": [("MI355X-NODE", 2)]}))
        # a running 4-GPU job on node1's first quad
        r0 = sim.schedule("c5/run", sim.pod_spec(vc="VC1", leaf_cells=4))
        assert r0.kind == "bind"
        # GPU 6 on node1 goes bad (leaf-cell health intake path)
        t0 = time.perf_counter_ns()
        sim.alg.set_leaf_cell_healthy("node1", 6, False)
        spec = sim.pod_spec(vc="VC1", leaf_cells=2)
        r = sim.schedule("c5/new", spec)
        lat.append((time.perf_counter_ns() - t0) / 1e6)
        assert r.kind == "bind"
        placed = set(r.bind_info.leafCellIsolation)
        assert not (r.bind_info.node == "node1" and 6 in placed), "placed on bad GPU"
        assert not (r.bind_info.node == "node1" and placed == {6, 7}), "bad pair used"
        # restart: replay both pods into a fresh algorithm -> placements kept
        sim2 = SimScheduler(mi355x_cluster_config(num_nodes=2,
                                                  vcs={"VC1": [("MI355X-NODE", 2)]}))
        for key in ("c5/run", "c5/new"):
            s, info = sim.pods[key]
            sim2.alg.add_allocated_pod(s, info, key)
        groups = sim2.alg.get_all_affinity_groups()
        assert len(groups) == 2 and all(g["state"] == "Allocated" for g in groups)
    out = {"desc": "1 bad MI355X: avoid + work-preserving replay",
           "mark_and_reschedule_latency": pcts(lat)}
    if gpu:
        from hivedscheduler_amd.ops import gpu_health_report

        rep = gpu_health_report(0, quick=True)
        out["live_health_probe"] = {"hbm_gbps": round(rep["hbm_gbps"], 1),
                                    "mfma_ok": rep["mfma_ok"], "healthy": rep["healthy"]}
    return out


def config_http(iters):
    """End-to-end extender HTTP latency: JSON ExtenderArgs over TCP to a real
    uvicorn server with a keep-alive connection — the path kube-scheduler
    exercises (reference deploy sets httpTimeout: 5s)."""
    import http.client
    import json as jsonlib
    import socket
    import threading

    import uvicorn
    import yaml

    from hivedscheduler_amd.api import constants
    from hivedscheduler_amd.scheduler import HivedScheduler
    from hivedscheduler_amd.sim import mi355x_cluster_config
    from hivedscheduler_amd.webserver import create_app

    sched = HivedScheduler(mi355x_cluster_config(num_nodes=1))
    sched.on_node_add({
        "metadata": {"name": "node1", "uid": "node-node1"},
        "spec": {},
        "status": {"conditions": [{"type": "Ready", "status": "True"}]},
    })
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    server = uvicorn.Server(uvicorn.Config(create_app(sched), host="127.0.0.1",
                                           port=port, log_level="error"))
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    for _ in range(100):
        if server.started:
            break
        time.sleep(0.05)
    assert server.started, "uvicorn did not start"

    conn = http.client.HTTPConnection("127.0.0.1", port)
    spec = yaml.safe_dump({"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    lat = []
    try:
        for i in range(-10, iters):
            pod = {
                "metadata": {"name": f"hp{i}", "namespace": "b", "uid": f"uid-b-hp{i}",
                             "annotations": {constants.AnnotationKeyPodSchedulingSpec: spec}},
                "spec": {"containers": [{"resources": {
                    "limits": {constants.ResourceNamePodSchedulingEnable: 1}}}]},
                "status": {"phase": "Pending"},
            }
            body = jsonlib.dumps({"Pod": pod, "NodeNames": ["node1"]})

            def post():
                conn.request("POST", constants.FilterPath, body,
                             {"Content-Type": "application/json"})
                resp = conn.getresponse()
                return resp.status, resp.read()

            (status, data), ms = timed(post)
            assert status == 200 and jsonlib.loads(data).get("NodeNames") == ["node1"], data
            if i >= 0:
                lat.append(ms)
            # release so the 8-GPU node never fills (full pod: delete path
            # needs the spec annotation + opt-in resource limit)
            sched.on_pod_delete(pod)
    finally:
        conn.close()
        server.should_exit = True
        th.join(timeout=5)
    return {"desc": "extender filter over HTTP (uvicorn, keep-alive, JSON in/out)",
            "http_latency": pcts(lat)}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=200)
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    gpu = False
    try:
        import torch

        gpu = torch.cuda.is_available()
    except ImportError:
        pass

    report = {"suite": "BASELINE.json configs 1-5", "gpu": gpu}
    for name, fn in (("config1", lambda: config1(args.iters)),
                     ("config2", lambda: config2(args.iters, gpu)),
                     ("config3", lambda: config3(args.iters)),
                     ("config4", lambda: config4(args.iters)),
                     ("config5", lambda: config5(args.iters, gpu)),
                     ("extender_http", lambda: config_http(args.iters))):
        t0 = time.perf_counter()
        report[name] = fn()
        report[name]["wall_s"] = round(time.perf_counter() - t0, 3)
        print(f"{name}: {report[name]}", file=sys.stderr)

    js = json.dumps(report, indent=1)
    print(js)
    if args.out:
        with open(args.out, "w") as f:
            f.write(js + "\n")


if __name__ == "__main__":
    main()
