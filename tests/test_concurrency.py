"""Concurrency stress: the reference relies on coarse locks + `go test -race`
(SURVEY.md §5). Here N threads hammer the scheduler framework concurrently —
filter, delete, node health flaps, inspect reads — and the C++ core's full
invariant checker must hold at the end (and no exception may escape a lock).
"""
import random
import threading

import yaml

from hivedscheduler_amd.api import constants
from hivedscheduler_amd.scheduler import HivedScheduler
from hivedscheduler_amd.sim import mi355x_cluster_config

from test_scheduler_framework import make_node, make_pod


def test_concurrent_filter_delete_health_inspect():
    cfg = mi355x_cluster_config(num_nodes=4, vcs={"VC1": [("MI355X-NODE", 2)],
                                                  "VC2": [("MI355X-NODE", 2)]})
    sched = HivedScheduler(cfg)
    for i in range(4):
        sched.on_node_add(make_node(f"node{i + 1}"))
    nodes = [f"node{i + 1}" for i in range(4)]
    errors = []
    stop = threading.Event()

    def worker(tid):
        rng = random.Random(tid)
        my_pods = []
        try:
            for i in range(200):
                spec = {"virtualCluster": rng.choice(["VC1", "VC2"]),
                        "priority": rng.choice([-1, 0, 1]),
                        "leafCellNumber": rng.choice([1, 2, 4])}
                pod = make_pod(f"t{tid}p{i}", spec, ns=f"w{tid}")
                res = sched.filter({"Pod": pod, "NodeNames": list(nodes)})
                if res.get("NodeNames"):
                    my_pods.append(pod)
                if my_pods and rng.random() < 0.6:
                    sched.on_pod_delete(my_pods.pop(rng.randrange(len(my_pods))))
            for pod in my_pods:
                sched.on_pod_delete(pod)
        except Exception as e:  # pragma: no cover
            errors.append((tid, repr(e)))

    def flapper():
        rng = random.Random(99)
        try:
            while not stop.is_set():
                n = rng.choice(nodes[2:])  # flap only VC2's nodes
                sched.on_node_update(make_node(n, ready=False), make_node(n, ready=True))
        except Exception as e:  # pragma: no cover
            errors.append(("flapper", repr(e)))

    def inspector():
        try:
            while not stop.is_set():
                sched.get_cluster_status()
                sched.get_all_affinity_groups()
        except Exception as e:  # pragma: no cover
            errors.append(("inspector", repr(e)))

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(6)]
    aux = [threading.Thread(target=flapper), threading.Thread(target=inspector)]
    for t in aux + threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    stop.set()
    for t in aux:
        t.join(timeout=10)
    assert not errors, errors[:5]
    # every pod was deleted; the tree must be pristine and invariant-clean
    sched.algorithm._core.check_invariants()
    assert sched.get_all_affinity_groups() == []
