#!/usr/bin/env python3
"""Production-style serving soak: the REAL extender stack under sustained
HTTP load on a live MI355X node.

Runs uvicorn with the full scheduler, then drives it with concurrent HTTP
clients executing complete pod lifecycles (filter -> bind -> delete) across
two VCs with mixed sizes/priorities, while:
  - a health-agent loop runs REAL HIP kernels (bf16+fp8+fp4 MFMA, HBM triad,
    LDS) every ~30 s and posts reports;
  - xGMI link flaps and heals are injected through the health API;
  - /healthz and inspect endpoints are polled concurrently.

Reports sustained decisions/s, HTTP latency percentiles, kernel-sweep
counts, link events applied, and a final full invariant check.

Usage: python bench_serving.py [--seconds 300] [--clients 8] [--out f.json]
"""
import argparse
import json
import random
import socket
import threading
import time


def pcts(samples):
    if not samples:
        return {}
    s = sorted(samples)
    return {"p50_ms": round(s[len(s) // 2] * 1e3, 3),
            "p95_ms": round(s[min(len(s) - 1, int(len(s) * 0.95))] * 1e3, 3),
            "p99_ms": round(s[min(len(s) - 1, int(len(s) * 0.99))] * 1e3, 3),
            "n": len(s)}


def make_pod(name, spec_yaml):
    import yaml

    from hivedscheduler_amd.api import constants

    ns, podname = name.split("/")
    return {"metadata": {"name": podname, "namespace": ns, "uid": f"uid-{name}",
                         "annotations": {
                             constants.AnnotationKeyPodSchedulingSpec: yaml.safe_dump(spec_yaml)}},
            "spec": {"containers": [{
                "name": "main",
                "resources": {"limits": {constants.ResourceNamePodSchedulingEnable: 1}}}]}}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=300.0)
    ap.add_argument("--clients", type=int, default=8)
    ap.add_argument("--nodes", type=int, default=4)
    ap.add_argument("--out", default=None)
    args = ap.parse_args()

    import requests
    import uvicorn

    from hivedscheduler_amd.scheduler import HivedScheduler
    from hivedscheduler_amd.sim import mi355x_cluster_config
    from hivedscheduler_amd.webserver import create_app

    vcs = {"prod": [("MI355X-NODE", args.nodes // 2)],
           "research": [("MI355X-NODE", args.nodes - args.nodes // 2)]}
    sched = HivedScheduler(mi355x_cluster_config(num_nodes=args.nodes, vcs=vcs))
    for n in sched.algorithm.all_nodes():
        sched.on_node_add({"metadata": {"name": n, "uid": f"n-{n}"}, "spec": {},
                           "status": {"conditions": [{"type": "Ready", "status": "True"}]}})

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    server = uvicorn.Server(uvicorn.Config(create_app(sched), host="127.0.0.1", port=port,
                                           log_level="error"))
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    base = f"http://127.0.0.1:{port}"
    for _ in range(200):
        if server.started:
            break
        time.sleep(0.05)

    stop = threading.Event()
    stats_lock = threading.Lock()
    filter_lat, bind_lat, healthz_lat = [], [], []
    counters = {"binds": 0, "waits": 0, "errors": 0, "deletes": 0,
                "health_sweeps": 0, "link_events": 0}

    def client_loop(cid):
        rng = random.Random(cid)
        sess = requests.Session()
        i = 0
        while not stop.is_set():
            i += 1
            vc = rng.choice(list(vcs))
            size = rng.choice([1, 1, 2, 2, 4])
            prio = rng.choice([-1, 0, 0, 1])
            name = f"load/c{cid}-{i}"
            pod = make_pod(name, {"virtualCluster": vc, "priority": prio,
                                  "leafCellNumber": size})
            t0 = time.perf_counter()
            r = sess.post(f"{base}/v1/extender/filter",
                          json={"Pod": pod, "NodeNames": sched.algorithm.all_nodes()},
                          timeout=30)
            dt = time.perf_counter() - t0
            with stats_lock:
                filter_lat.append(dt)
            if r.status_code != 200:
                with stats_lock:
                    counters["errors"] += 1
                continue
            body = r.json()
            if body.get("NodeNames"):
                node = body["NodeNames"][0]
                t0 = time.perf_counter()
                rb = sess.post(f"{base}/v1/extender/bind",
                               json={"PodName": pod["metadata"]["name"],
                                     "PodNamespace": pod["metadata"]["namespace"],
                                     "PodUID": pod["metadata"]["uid"], "Node": node},
                               timeout=30)
                dtb = time.perf_counter() - t0
                with stats_lock:
                    bind_lat.append(dtb)
                    if rb.status_code == 200 and not rb.json().get("Error"):
                        counters["binds"] += 1
                    else:
                        counters["errors"] += 1
                # hold the pod briefly, then release (lifecycle churn)
                time.sleep(rng.uniform(0.005, 0.05))
                st = sched.pod_statuses.get(pod["metadata"]["uid"])
                if st is not None and st.pod_scheduling_spec is not None:
                    try:
                        sched.on_pod_delete(st.pod)
                        with stats_lock:
                            counters["deletes"] += 1
                    except Exception:
                        with stats_lock:
                            counters["errors"] += 1
            else:
                with stats_lock:
                    counters["waits"] += 1
                time.sleep(0.02)

    def health_loop():
        """Real HIP kernels on cuda:0 every ~30 s, posted as the agent does."""
        try:
            import torch

            has_gpu = torch.cuda.is_available()
        except Exception:
            has_gpu = False
        node = sched.algorithm.all_nodes()[0]
        while not stop.is_set():
            report = {"gpus": {}}
            if has_gpu:
                from hivedscheduler_amd.ops import gpu_health_report

                rep = gpu_health_report(0, quick=True)
                report["gpus"]["0"] = {"healthy": bool(rep["healthy"]),
                                       "hbm_gbps": round(rep["hbm_gbps"], 1),
                                       "mfma_ok": bool(rep["mfma_ok"]),
                                       "mfma_lowprec_ok": bool(rep["mfma_lowprec_ok"])}
            try:
                requests.post(f"{base}/v1/health/nodes/{node}", json=report, timeout=30)
                with stats_lock:
                    counters["health_sweeps"] += 1
            except Exception:
                pass
            stop.wait(30.0)

    def weather_loop():
        """Random link flaps through the health API, healed a few seconds later."""
        rng = random.Random(999)
        nodes = sched.algorithm.all_nodes()
        while not stop.is_set():
            stop.wait(rng.uniform(5.0, 15.0))
            if stop.is_set():
                break
            node = rng.choice(nodes)
            a, b = rng.sample(range(8), 2)
            try:
                requests.post(f"{base}/v1/health/nodes/{node}",
                              json={"gpus": {}, "links": [
                                  {"a": a, "b": b, "healthy": False,
                                   "gbps": rng.uniform(5, 40)}]}, timeout=30)
                with stats_lock:
                    counters["link_events"] += 1
                stop.wait(rng.uniform(3.0, 10.0))
                requests.post(f"{base}/v1/health/nodes/{node}",
                              json={"gpus": {}, "links": [
                                  {"a": a, "b": b, "healthy": True, "gbps": 153.0}]},
                              timeout=30)
                with stats_lock:
                    counters["link_events"] += 1
            except Exception:
                pass

    def monitor_loop():
        sess = requests.Session()
        while not stop.is_set():
            t0 = time.perf_counter()
            try:
                sess.get(f"{base}/healthz", timeout=10)
                with stats_lock:
                    healthz_lat.append(time.perf_counter() - t0)
                sess.get(f"{base}/v1/inspect/clusterstatus", timeout=10)
            except Exception:
                pass
            stop.wait(1.0)

    threads = [threading.Thread(target=client_loop, args=(c,), daemon=True)
               for c in range(args.clients)]
    threads += [threading.Thread(target=health_loop, daemon=True),
                threading.Thread(target=weather_loop, daemon=True),
                threading.Thread(target=monitor_loop, daemon=True)]
    t_start = time.perf_counter()
    for t in threads:
        t.start()
    time.sleep(args.seconds)
    stop.set()
    for t in threads:
        t.join(timeout=10)
    elapsed = time.perf_counter() - t_start

    # final consistency: invariants + full-quota drain on a healed cluster
    for n in sched.algorithm.all_nodes():
        for l in sched.get_xgmi_links(n):
            if not l["healthy"]:
                sched.algorithm.set_xgmi_link_healthy(n, l["a"], l["b"], True, 153.0)
    sched.algorithm._core.check_invariants()

    out = {
        "seconds": round(elapsed, 1),
        "clients": args.clients,
        "nodes": args.nodes,
        "http_filter": pcts(filter_lat),
        "http_bind": pcts(bind_lat),
        "http_healthz_under_load": pcts(healthz_lat),
        "lifecycles_per_s": round(counters["binds"] / elapsed, 1),
        "decisions_per_s": round((counters["binds"] + counters["waits"]) / elapsed, 1),
        **counters,
        "invariants_ok": True,
        "server_decisions": sched.algorithm.schedule_count(),
    }
    print(json.dumps(out, indent=1))
    if args.out:
        with open(args.out, "w") as f:
            json.dump(out, f, indent=1)
    server.should_exit = True


if __name__ == "__main__":
    main()
