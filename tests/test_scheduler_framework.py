"""Scheduler framework (L4): filter/bind/preempt routines, pod state machine,
optimistic commit, force-bind, recovery via informer callbacks. Drives
HivedScheduler with K8s-shaped dicts, no cluster."""
import time

import pytest
import yaml

from hivedscheduler_amd.api import constants
from hivedscheduler_amd.api.types import WebServerError
from hivedscheduler_amd.scheduler import HivedScheduler
from hivedscheduler_amd.sim import mi355x_cluster_config


def make_pod(name, spec: dict, uid=None, ns="ns", node=None, bind_info=None):
    pod = {
        "metadata": {
            "name": name,
            "namespace": ns,
            "uid": uid or f"uid-{ns}-{name}",
            "annotations": {
                constants.AnnotationKeyPodSchedulingSpec: yaml.safe_dump(spec),
            },
        },
        "spec": {
            "containers": [{
                "name": "main",
                "resources": {"limits": {constants.ResourceNamePodSchedulingEnable: 1}},
            }],
        },
        "status": {"phase": "Pending"},
    }
    if node:
        pod["spec"]["nodeName"] = node
    if bind_info is not None:
        pod["metadata"]["annotations"][constants.AnnotationKeyPodBindInfo] = yaml.safe_dump(
            bind_info)
    return pod


def make_node(name, ready=True, unschedulable=False):
    return {
        "metadata": {"name": name, "uid": f"node-{name}"},
        "spec": {"unschedulable": unschedulable},
        "status": {"conditions": [{"type": "Ready", "status": "True" if ready else "False"}]},
    }


@pytest.fixture()
def sched():
    cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]})
    s = HivedScheduler(cfg)
    for n in ("node1", "node2"):
        s.on_node_add(make_node(n))
    return s


SPEC2 = {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2}


def test_filter_bind_flow(sched):
    pod = make_pod("p1", SPEC2)
    result = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    assert result["NodeNames"] and result["NodeNames"][0] in ("node1", "node2")
    node = result["NodeNames"][0]
    uid = pod["metadata"]["uid"]
    st = sched.pod_statuses[uid]
    assert st.state == "Binding"
    # binding pod carries the decision annotations
    ann = st.pod["metadata"]["annotations"]
    assert ann[constants.AnnotationKeyPodLeafCellIsolation]
    assert ann[constants.AnnotationKeyPodBindInfo]
    # bind verb
    r = sched.bind({"PodName": "p1", "PodNamespace": "ns", "PodUID": uid, "Node": node})
    assert r == {}
    assert sched.pod_statuses[uid].state == "Bound"
    # second filter for a bound pod is a client error
    with pytest.raises(WebServerError):
        sched.filter({"Pod": pod, "NodeNames": ["node1"]})


def test_filter_insists_binding_decision(sched):
    pod = make_pod("p1", SPEC2)
    r1 = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    node = r1["NodeNames"][0]
    # repeated filter (K8s retry) returns the same node and bumps attempts
    r2 = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    assert r2["NodeNames"] == [node]
    assert sched.pod_statuses[pod["metadata"]["uid"]].pod_bind_attempts == 1


def test_bind_node_mismatch_rejected(sched):
    pod = make_pod("p1", SPEC2)
    r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    wrong = "node2" if r["NodeNames"][0] == "node1" else "node1"
    with pytest.raises(WebServerError):
        sched.bind({"PodName": "p1", "PodNamespace": "ns",
                    "PodUID": pod["metadata"]["uid"], "Node": wrong})


def test_wait_returns_failed_nodes(sched):
    pod = make_pod("big", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 8,
                           "affinityGroup": {"name": "big", "members": [
                               {"podNumber": 3, "leafCellNumber": 8}]}})
    r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    assert constants.ComponentName in r["FailedNodes"]


def test_preempt_flow(sched):
    # fill both nodes with opportunistic pods
    for i, name in enumerate(("o1", "o2")):
        pod = make_pod(name, {"virtualCluster": "VC1", "priority": -1, "leafCellNumber": 8})
        r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
        node = r["NodeNames"][0]
        sched.bind({"PodName": name, "PodNamespace": "ns",
                    "PodUID": pod["metadata"]["uid"], "Node": node})
    g = make_pod("g", {"virtualCluster": "VC1", "priority": 5, "leafCellNumber": 8})
    # filter reports preemption potential via FailedNodes
    r = sched.filter({"Pod": g, "NodeNames": ["node1", "node2"]})
    assert r.get("FailedNodes"), r
    # preempt verb commits the preemption and returns victims
    r = sched.preempt({"Pod": g, "NodeNameToMetaVictims": {"node1": {}, "node2": {}}})
    victims = r["NodeNameToMetaVictims"]
    assert len(victims) == 1
    node, meta = next(iter(victims.items()))
    assert meta["Pods"], r
    assert sched.pod_statuses[g["metadata"]["uid"]].state == "Preempting"
    # K8s deletes the victims -> informer delete events
    victim_uids = {p["UID"] for p in meta["Pods"]}
    for name in ("o1", "o2"):
        key = f"ns/{name}"
        if key in victim_uids:
            st = [s for s in sched.pod_statuses.values()
                  if s.pod["metadata"]["name"] == name][0]
            sched.on_pod_delete(st.pod)
    # next preempt round converges to bind-ready (empty result), then filter binds
    for _ in range(4):
        r = sched.preempt({"Pod": g, "NodeNameToMetaVictims": {"node1": {}, "node2": {}}})
        if r == {}:
            break
        for node, meta in r["NodeNameToMetaVictims"].items():
            for p in meta["Pods"]:
                name = p["UID"].split("/", 1)[1]
                st = [s for s in sched.pod_statuses.values()
                      if s.pod["metadata"]["name"] == name][0]
                sched.on_pod_delete(st.pod)
    assert r == {}
    r = sched.filter({"Pod": g, "NodeNames": ["node1", "node2"]})
    assert r.get("NodeNames"), r


def test_recovery_via_informer(sched):
    pod = make_pod("p1", SPEC2)
    r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    node = r["NodeNames"][0]
    st = sched.pod_statuses[pod["metadata"]["uid"]]
    bound_pod = st.pod  # carries bind-info annotations + nodeName

    # "restart": fresh scheduler, informer replays node + bound pod
    cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]})
    s2 = HivedScheduler(cfg)
    for n in ("node1", "node2"):
        s2.on_node_add(make_node(n))
    s2.on_pod_add(bound_pod)
    groups = s2.get_all_affinity_groups()
    assert [g["name"] for g in groups] == ["ns/p1"]
    assert node in groups[0]["physicalPlacement"]
    # state survives: a full-node pod still fits only once
    p2 = make_pod("p2", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 8})
    assert s2.filter({"Pod": p2, "NodeNames": ["node1", "node2"]}).get("NodeNames")
    p3 = make_pod("p3", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 8})
    assert constants.ComponentName in s2.filter(
        {"Pod": p3, "NodeNames": ["node1", "node2"]})["FailedNodes"]


def test_node_events_drive_health(sched):
    pod = make_pod("p1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 8})
    sched.on_node_update(make_node("node1"), make_node("node1", ready=False))
    sched.on_node_delete(make_node("node2"))
    r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    assert constants.ComponentName in r.get("FailedNodes", {})
    sched.on_node_update(make_node("node1", ready=False), make_node("node1"))
    r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    assert r.get("NodeNames") == ["node1"]


def test_force_bind_on_threshold(sched):
    sched.force_bind_threshold = 2
    pod = make_pod("p1", SPEC2)
    r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    node = r["NodeNames"][0]
    uid = pod["metadata"]["uid"]
    # two more filter retries hit the threshold -> force bind fires async
    sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    deadline = time.time() + 5
    while time.time() < deadline and sched.pod_statuses[uid].state != "Bound":
        time.sleep(0.05)
    assert sched.pod_statuses[uid].state == "Bound"
    assert sched.pod_statuses[uid].node == node


def test_health_report_marks_leaf_cells(sched):
    sched.on_health_report("node1", {"gpus": {str(i): {"healthy": False} for i in range(8)}})
    pod = make_pod("p8", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 8})
    r = sched.filter({"Pod": pod, "NodeNames": ["node1", "node2"]})
    assert r["NodeNames"] == ["node2"]
    # recovery
    sched.on_health_report("node1", {"gpus": {str(i): {"healthy": True} for i in range(8)}})
    pod2 = make_pod("p8b", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 8})
    assert sched.filter({"Pod": pod2, "NodeNames": ["node1", "node2"]})["NodeNames"] == ["node1"]
