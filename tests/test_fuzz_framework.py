"""Framework-layer fuzz: random K8s-shaped traffic through HivedScheduler's
filter/bind/preempt/informer callbacks — including malformed annotations,
out-of-order binds, unknown pods, and node flaps. Nothing but WebServerError
may escape, the pod state machine must stay consistent, and the core's
invariants must hold throughout."""
import random

import pytest
import yaml

from hivedscheduler_amd.api import constants
from hivedscheduler_amd.api.types import WebServerError
from hivedscheduler_amd.scheduler import HivedScheduler
from hivedscheduler_amd.sim import mi355x_cluster_config

from test_scheduler_framework import make_node, make_pod


@pytest.mark.parametrize("seed", list(range(6)))
def test_fuzz_framework_traffic(seed):
    rng = random.Random(7000 + seed)
    cfg = mi355x_cluster_config(num_nodes=3, vcs={"VC1": [("MI355X-NODE", 2)],
                                                  "VC2": [("MI355X-NODE", 1)]})
    sched = HivedScheduler(cfg)
    nodes = [f"node{i + 1}" for i in range(3)]
    for n in nodes:
        sched.on_node_add(make_node(n))
    pods = {}  # name -> (pod, chosen_node or None)
    counter = 0
    for step in range(400):
        op = rng.random()
        try:
            if op < 0.35:
                counter += 1
                name = f"fp{counter}"
                if rng.random() < 0.1:  # malformed spec variants
                    bad = rng.choice([
                        "not: [valid", "virtualCluster: ''",
                        "virtualCluster: NOPE\nleafCellNumber: 2",
                        "virtualCluster: VC1\nleafCellNumber: -3",
                        "virtualCluster: VC1\npriority: 99999\nleafCellNumber: 1",
                    ])
                    pod = make_pod(name, {})
                    pod["metadata"]["annotations"][
                        constants.AnnotationKeyPodSchedulingSpec] = bad
                else:
                    spec = {"virtualCluster": rng.choice(["VC1", "VC2"]),
                            "priority": rng.choice([-1, 0, 10]),
                            "leafCellNumber": rng.choice([1, 2, 4, 8])}
                    pod = make_pod(name, spec)
                r = sched.filter({"Pod": pod,
                                  "NodeNames": rng.sample(nodes, rng.randrange(1, 4))})
                chosen = (r.get("NodeNames") or [None])[0]
                pods[name] = (pod, chosen)
            elif op < 0.5 and pods:
                # bind: sometimes the right node, sometimes wrong/unknown pod
                name = rng.choice(list(pods))
                pod, chosen = pods[name]
                node = chosen if (chosen and rng.random() < 0.8) else rng.choice(nodes)
                sched.bind({"PodName": name, "PodNamespace": "ns",
                            "PodUID": pod["metadata"]["uid"], "Node": node})
            elif op < 0.6 and pods:
                name = rng.choice(list(pods))
                pod, _ = pods.pop(name)
                sched.on_pod_delete(pod)
            elif op < 0.72 and pods:
                name = rng.choice(list(pods))
                pod, _ = pods[name]
                sched.preempt({"Pod": pod, "NodeNameToMetaVictims":
                               {n: {} for n in rng.sample(nodes, rng.randrange(1, 4))}})
            elif op < 0.85:
                n = rng.choice(nodes)
                sched.on_node_update(make_node(n, ready=rng.random() < 0.7), make_node(n))
            elif op < 0.95:
                sched.on_health_report(rng.choice(nodes), {"gpus": {
                    str(rng.randrange(8)): {"healthy": rng.random() < 0.7}}})
            else:
                sched.get_cluster_status()
                sched.get_all_affinity_groups()
            sched.algorithm._core.check_invariants()
        except WebServerError:
            continue
    # cleanup must fully drain
    for name, (pod, _) in list(pods.items()):
        try:
            sched.on_pod_delete(pod)
        except WebServerError:
            pass
    sched.algorithm._core.check_invariants()
