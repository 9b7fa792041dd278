"""Webserver (L5): extender HTTP protocol + inspect API + metrics, via the
Starlette test client (no real server)."""
import pytest
import yaml

from hivedscheduler_amd.api import constants
from hivedscheduler_amd.scheduler import HivedScheduler
from hivedscheduler_amd.sim import mi355x_cluster_config
from hivedscheduler_amd.webserver import create_app

from test_scheduler_framework import make_node, make_pod


@pytest.fixture()
def client():
    from starlette.testclient import TestClient

    cfg = mi355x_cluster_config(num_nodes=1)
    sched = HivedScheduler(cfg)
    sched.on_node_add(make_node("node1"))
    return TestClient(create_app(sched), raise_server_exceptions=False)


def test_root_lists_paths(client):
    r = client.get("/")
    assert r.status_code == 200
    assert constants.FilterPath in r.json()["paths"]


def test_filter_bind_over_http(client):
    pod = make_pod("p1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    r = client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    assert r.status_code == 200
    assert r.json()["NodeNames"] == ["node1"]
    r = client.post(constants.BindPath, json={
        "PodName": "p1", "PodNamespace": "ns", "PodUID": pod["metadata"]["uid"],
        "Node": "node1"})
    assert r.status_code == 200
    assert not r.json().get("Error")


def test_bind_error_in_band(client):
    r = client.post(constants.BindPath, json={
        "PodName": "nope", "PodNamespace": "ns", "PodUID": "u-nope", "Node": "node1"})
    assert r.status_code == 200
    assert r.json()["Error"]


def test_bad_request_translation(client):
    pod = make_pod("p1", {"virtualCluster": "NOPE", "priority": 0, "leafCellNumber": 2})
    r = client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    assert r.status_code == 400
    assert "NOPE" in r.json()["error"]
    r = client.post(constants.FilterPath, json={"NodeNames": ["node1"]})
    assert r.status_code == 400


def test_inspect_api(client):
    pod = make_pod("p1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    r = client.get(constants.AffinityGroupsPath)
    assert r.status_code == 200
    assert [g["name"] for g in r.json()] == ["ns/p1"]
    r = client.get(constants.AffinityGroupsPath + "ns/p1")
    assert r.status_code == 200
    assert r.json()["vc"] == "VC1"
    assert client.get(constants.AffinityGroupsPath + "missing").status_code == 400
    r = client.get(constants.ClusterStatusPath)
    assert r.status_code == 200
    assert "physicalCluster" in r.json() and "virtualClusters" in r.json()
    r = client.get(constants.PhysicalClusterPath)
    assert r.status_code == 200
    assert r.json()[0]["cellType"] == "MI355X-NODE"
    r = client.get(constants.VirtualClustersPath + "VC1")
    assert r.status_code == 200
    assert client.get(constants.VirtualClustersPath + "NOPE").status_code == 404


def test_metrics_endpoint(client):
    pod = make_pod("p1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    r = client.get(constants.MetricsPath)
    assert r.status_code == 200
    assert "hived_schedule_seconds" in r.text


def test_healthz(client):
    assert client.get("/healthz").text == "ok"


def test_health_report_endpoint(client):
    # one bad GPU reported by the node agent
    r = client.post("/v1/health/nodes/node1", json={
        "gpus": {"0": {"healthy": False, "hbm_gbps": 900.0},
                 "1": {"healthy": True}}})
    assert r.status_code == 200
    assert r.json()["applied"] == {"0": False, "1": True}
    # placement now avoids GPU 0
    pod = make_pod("p1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    rr = client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    assert rr.status_code == 200 and rr.json().get("NodeNames")
    reports = client.get("/v1/inspect/health").json()
    assert "node1" in reports


def test_probe_workflow(client):
    # binding a pod enqueues a placement-probe task for its node
    pod = make_pod("p1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    tasks = client.get("/v1/health/probes/node1").json()
    assert len(tasks) == 1
    assert tasks[0]["leafCellIndices"] == [0, 1]
    group = tasks[0]["group"]
    # polling drains the queue
    assert client.get("/v1/health/probes/node1").json() == []
    # a healthy probe result is recorded
    r = client.post("/v1/health/probes", json={
        "group": group, "node": "node1", "leafCellIndices": [0, 1],
        "ok": True, "busbw_gbps": 142.0, "algbw_gbps": 142.0})
    assert r.json()["healthy"] is True
    results = client.get(f"/v1/inspect/probes/{group}").json()
    assert results and results[0]["busbw_gbps"] == 142.0


def test_probe_degraded_link_marks_cells_bad(client):
    pod = make_pod("p1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    task = client.get("/v1/health/probes/node1").json()[0]
    # degraded xGMI: far below the ~153 GB/s link expectation
    r = client.post("/v1/health/probes", json={
        "group": task["group"], "node": "node1", "leafCellIndices": [0, 1],
        "ok": True, "busbw_gbps": 9.0})
    assert r.json()["healthy"] is False
    # the pair is now bad: next pair request avoids GPUs 0/1
    pod2 = make_pod("p2", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    rr = client.post(constants.FilterPath, json={"Pod": pod2, "NodeNames": ["node1"]})
    assert rr.json().get("NodeNames")
    st = client.get(constants.AffinityGroupsPath + "ns/p2").json()
    assert set(st["physicalPlacement"]["node1"]).isdisjoint({0, 1})


def test_health_intake_applies_links(client):
    """A degraded-link report marks the LINK first-class: the endpoint GPUs
    stay schedulable (cells stay Healthy) while the link shows in inspect."""
    r = client.post("/v1/health/nodes/node1", json={
        "gpus": {}, "links": [{"a": 0, "b": 1, "healthy": False, "gbps": 12.5}]})
    assert r.status_code == 200
    assert r.json()["appliedLinks"] == {"0-1": False}
    links = client.get("/v1/inspect/links/node1").json()
    assert links == [{"a": 0, "b": 1, "gbps": 12.5, "healthy": False}]
    # no cell went Bad: link degradation is not leaf badness
    status = client.get(constants.PhysicalClusterPath).json()

    def walk(c):
        yield c
        for ch in c.get("cellChildren") or []:
            yield from walk(ch)

    cells = [c for top in status for c in walk(top)]
    assert all(c.get("cellHealthiness") == "Healthy" for c in cells)
    assert any(c.get("badXgmiLinksUnder") for c in cells)
    # a 2-GPU filter avoids the degraded pair
    pod = make_pod("p-l1", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 2})
    fr = client.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
    assert fr.status_code == 200 and fr.json()["NodeNames"] == ["node1"]


def test_probe_result_pair_marks_link_not_leaves(client):
    """A failed 2-GPU placement probe localizes to ONE link; the endpoint
    GPUs stay schedulable for 1-GPU work."""
    r = client.post("/v1/health/probes", json={
        "ok": True, "healthy": False, "node": "node1", "group": "g1",
        "leafCellIndices": [2, 3], "busbw_gbps": 8.0})
    assert r.status_code == 200
    links = {(l["a"], l["b"]): l for l in client.get("/v1/inspect/links/node1").json()}
    assert not links[(2, 3)]["healthy"]
    assert links[(2, 3)]["gbps"] == 8.0
    status = client.get(constants.PhysicalClusterPath).json()

    def walk(c):
        yield c
        for ch in c.get("cellChildren") or []:
            yield from walk(ch)

    leaves = [c for top in status for c in walk(top) if c.get("leafCellIndex") is not None]
    assert all(c["cellHealthiness"] == "Healthy" for c in leaves)


def test_event_loop_not_blocked_by_fifo_wait():
    """A waiting pod's FIFO block (50 ms) must not stall concurrent requests:
    scheduler calls run in the threadpool and the sleep happens outside the
    scheduler lock (reference: goroutine-per-request)."""
    import threading
    import time as _time

    from hivedscheduler_amd.api.config import Config as _C  # noqa: F401

    cfg = mi355x_cluster_config(num_nodes=1)
    cfg.waitingPodSchedulingBlockMilliSec = 200
    sched = HivedScheduler(cfg)
    sched.on_node_add(make_node("node1"))
    from starlette.testclient import TestClient

    c = TestClient(create_app(sched), raise_server_exceptions=False)
    # an unsatisfiable request -> wait path with the 200 ms FIFO block
    pod = make_pod("p-wait", {"virtualCluster": "VC1", "priority": 0, "leafCellNumber": 8,
                              "affinityGroup": {"name": "gw", "members": [
                                  {"podNumber": 2, "leafCellNumber": 8}]}})
    timings = {}

    def waiter():
        t0 = _time.perf_counter()
        c.post(constants.FilterPath, json={"Pod": pod, "NodeNames": ["node1"]})
        timings["wait"] = _time.perf_counter() - t0

    th = threading.Thread(target=waiter)
    th.start()
    _time.sleep(0.05)  # let the filter reach its FIFO sleep
    t0 = _time.perf_counter()
    r = c.get("/healthz")
    healthz_t = _time.perf_counter() - t0
    assert r.status_code == 200
    t0 = _time.perf_counter()
    r2 = c.post("/v1/health/nodes/node1", json={"gpus": {"0": {"healthy": True}}})
    health_post_t = _time.perf_counter() - t0
    assert r2.status_code == 200
    th.join(timeout=5)
    assert timings["wait"] >= 0.2, f"FIFO block missing: {timings}"
    # concurrent requests must not have waited for the 200 ms block
    assert healthz_t < 0.15, f"/healthz stalled {healthz_t * 1e3:.0f} ms behind FIFO block"
    assert health_post_t < 0.15, f"health POST stalled {health_post_t * 1e3:.0f} ms"


def test_probe_result_suspect_links_localize(client):
    """A >2-GPU probe with a p2p matrix marks the localized links, not the
    probed leaves."""
    r = client.post("/v1/health/probes", json={
        "ok": True, "healthy": False, "node": "node1", "group": "g2",
        "leafCellIndices": [4, 5, 6, 7], "busbw_gbps": 20.0,
        "suspect_links": [[6, 7, 9.5]]})
    assert r.status_code == 200
    links = {(l["a"], l["b"]): l for l in client.get("/v1/inspect/links/node1").json()}
    assert not links[(6, 7)]["healthy"] and links[(6, 7)]["gbps"] == 9.5
    status = client.get(constants.PhysicalClusterPath).json()

    def walk(c):
        yield c
        for ch in c.get("cellChildren") or []:
            yield from walk(ch)

    leaves = [c for top in status for c in walk(top) if c.get("leafCellIndex") is not None]
    assert all(c["cellHealthiness"] == "Healthy" for c in leaves)
