"""Full control-loop integration test, no real Kubernetes:

fake API server (path-aware list+watch+bind) -> KubeClient + Informers ->
HivedScheduler -> uvicorn extender HTTP -> kube-scheduler-style client calls
(filter, bind) -> Bind subresource lands on the fake API server -> the bound
pod (with the annotations the binding wrote) is fed back -> a SECOND
scheduler instance recovers the full allocation from the pod alone.

This is the reference's whole deployment loop executed
in-process.
"""
import json
import queue
import socket
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest
import yaml

from hivedscheduler_amd.api import constants
from hivedscheduler_amd.k8s import Informer, KubeClient
from hivedscheduler_amd.scheduler import HivedScheduler
from hivedscheduler_amd.sim import mi355x_cluster_config
from hivedscheduler_amd.webserver import create_app

from test_k8s_informer import node, wait_until


class PathFakeAPIServer:
    """Path-aware fake K8s API server: /api/v1/nodes and /api/v1/pods with
    list + watch, plus the pod binding subresource."""

    def __init__(self):
        self.stores = {"/api/v1/nodes": {}, "/api/v1/pods": {}}
        self.rv = 1
        self.queues = {"/api/v1/nodes": [], "/api/v1/pods": []}
        self.bindings = []
        fake = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_GET(self):
                path, _, qs = self.path.partition("?")
                params = dict(p.split("=", 1) for p in qs.split("&") if "=" in p)
                store = fake.stores.get(path)
                if store is None:
                    self.send_response(404)
                    self.end_headers()
                    return
                if params.get("watch") == "true":
                    q = queue.Queue()
                    fake.queues[path].append(q)
                    self.send_response(200)
                    self.end_headers()
                    deadline = time.time() + 5
                    try:
                        while time.time() < deadline:
                            try:
                                ev = q.get(timeout=0.1)
                            except queue.Empty:
                                continue
                            self.wfile.write((json.dumps(ev) + "\n").encode())
                            self.wfile.flush()
                    except BrokenPipeError:
                        pass
                    finally:
                        fake.queues[path].remove(q)
                    return
                body = json.dumps({
                    "metadata": {"resourceVersion": str(fake.rv)},
                    "items": list(store.values()),
                }).encode()
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_POST(self):
                length = int(self.headers.get("Content-Length", 0))
                data = json.loads(self.rfile.read(length) or b"{}")
                if self.path.endswith("/binding"):
                    fake.bindings.append((self.path, data))
                    self.send_response(201)
                    self.send_header("Content-Length", "2")
                    self.end_headers()
                    self.wfile.write(b"{}")
                else:
                    self.send_response(404)
                    self.end_headers()

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        threading.Thread(target=self.server.serve_forever, daemon=True).start()

    @property
    def url(self):
        return f"http://127.0.0.1:{self.server.server_port}"

    def add(self, path, obj):
        self.rv += 1
        obj["metadata"]["resourceVersion"] = str(self.rv)
        self.stores[path][obj["metadata"]["uid"]] = obj
        for q in list(self.queues[path]):
            q.put({"type": "ADDED", "object": obj})

    def stop(self):
        self.server.shutdown()


def make_request_pod(name, spec, ns="e2e"):
    return {
        "metadata": {"name": name, "namespace": ns, "uid": f"uid-{ns}-{name}",
                     "annotations": {
                         constants.AnnotationKeyPodSchedulingSpec: yaml.safe_dump(spec)}},
        "spec": {"containers": [{"resources": {"limits": {
            constants.ResourceNamePodSchedulingEnable: 1}}}]},
        "status": {"phase": "Pending"},
    }


@pytest.mark.timeout(120)
def test_full_control_loop_and_recovery():
    fake = PathFakeAPIServer()
    try:
        # -- scheduler #1 wired exactly like __main__ does -------------------
        cfg = mi355x_cluster_config(num_nodes=2, vcs={"VC1": [("MI355X-NODE", 2)]})
        k8s = KubeClient(api_server=fake.url)
        sched = HivedScheduler(cfg, k8s_client=k8s)
        fake.add("/api/v1/nodes", node("node1"))
        fake.add("/api/v1/nodes", node("node2"))
        ni = Informer(k8s, "/api/v1/nodes", on_add=sched.on_node_add,
                      on_update=sched.on_node_update, on_delete=sched.on_node_delete).start()
        pi = Informer(k8s, "/api/v1/pods", on_add=sched.on_pod_add,
                      on_update=sched.on_pod_update, on_delete=sched.on_pod_delete).start()
        assert ni.wait_for_cache_sync(30) and pi.wait_for_cache_sync(30)
        sched.synced.set()
        assert wait_until(lambda: not sched.algorithm.bad_nodes())

        # -- real extender HTTP server --------------------------------------
        import uvicorn

        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        server = uvicorn.Server(uvicorn.Config(create_app(sched), host="127.0.0.1",
                                               port=port, log_level="error"))
        th = threading.Thread(target=server.run, daemon=True)
        th.start()
        assert wait_until(lambda: server.started, 30)

        import requests

        base = f"http://127.0.0.1:{port}"
        pod = make_request_pod("train-0", {"virtualCluster": "VC1", "priority": 10,
                                           "leafCellNumber": 4})
        # kube-scheduler: filter
        r = requests.post(base + constants.FilterPath,
                          json={"Pod": pod, "NodeNames": ["node1", "node2"]}, timeout=30)
        assert r.status_code == 200 and len(r.json().get("NodeNames", [])) == 1, r.text
        chosen = r.json()["NodeNames"][0]
        # kube-scheduler: bind -> Bind subresource on the fake API server
        r = requests.post(base + constants.BindPath,
                          json={"PodName": "train-0", "PodNamespace": "e2e",
                                "PodUID": pod["metadata"]["uid"], "Node": chosen}, timeout=30)
        assert r.status_code == 200 and not r.json().get("Error"), r.text
        assert wait_until(lambda: fake.bindings)
        bpath, binding = fake.bindings[0]
        assert "e2e/pods/train-0/binding" in bpath
        assert binding["target"]["name"] == chosen
        ann = binding["metadata"]["annotations"]
        iso = ann[constants.AnnotationKeyPodLeafCellIsolation]
        assert len(iso.split(",")) == 4
        assert constants.AnnotationKeyPodBindInfo in ann

        server.should_exit = True
        th.join(timeout=10)
        ni.stop()
        pi.stop()

        # -- crash + recovery: scheduler #2 rebuilds from the bound pod -----
        bound_pod = dict(pod)
        bound_pod["metadata"] = {**pod["metadata"],
                                 "annotations": {**pod["metadata"]["annotations"], **ann}}
        bound_pod["spec"] = {**pod["spec"], "nodeName": chosen}
        fake.add("/api/v1/pods", bound_pod)

        sched2 = HivedScheduler(cfg, k8s_client=KubeClient(api_server=fake.url))
        ni2 = Informer(KubeClient(api_server=fake.url), "/api/v1/nodes",
                       on_add=sched2.on_node_add, on_update=sched2.on_node_update,
                       on_delete=sched2.on_node_delete).start()
        pi2 = Informer(KubeClient(api_server=fake.url), "/api/v1/pods",
                       on_add=sched2.on_pod_add, on_update=sched2.on_pod_update,
                       on_delete=sched2.on_pod_delete).start()
        assert ni2.wait_for_cache_sync(30) and pi2.wait_for_cache_sync(30)
        sched2.synced.set()
        groups = sched2.get_all_affinity_groups()
        assert len(groups) == 1 and groups[0]["state"] == "Allocated"
        assert groups[0]["physicalPlacement"] == {chosen: [int(i) for i in iso.split(",")]}
        sched2.algorithm._core.check_invariants()
        ni2.stop()
        pi2.stop()
    finally:
        fake.stop()
