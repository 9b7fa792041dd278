"""Minimal Kubernetes REST client (no official client dependency).

Covers exactly what the scheduler needs: list/watch of nodes and pods, and
the pod Bind subresource (atomic, at-most-once — reference
pkg/internal/utils.go:291-314). Supports in-cluster service-account auth and
plain API-server addresses (kubeApiServerAddress in the config).
"""
from __future__ import annotations

import json
import os
from typing import Any, Dict, Iterator, Optional

import requests

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubeClient:
    def __init__(
        self,
        api_server: Optional[str] = None,
        token: Optional[str] = None,
        ca_cert: Optional[str] = None,
        timeout_s: float = 30.0,
    ):
        if api_server is None:
            host = os.environ.get("KUBERNETES_SERVICE_HOST")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            if host:
                api_server = f"https://{host}:{port}"
        if api_server is None:
            raise ValueError("no API server address (set kubeApiServerAddress or run in-cluster)")
        self.api_server = api_server.rstrip("/")
        self.timeout_s = timeout_s
        self.session = requests.Session()
        if token is None and os.path.exists(os.path.join(SA_DIR, "token")):
            with open(os.path.join(SA_DIR, "token")) as f:
                token = f.read().strip()
        if token:
            self.session.headers["Authorization"] = f"Bearer {token}"
        if ca_cert is None and os.path.exists(os.path.join(SA_DIR, "ca.crt")):
            ca_cert = os.path.join(SA_DIR, "ca.crt")
        self.session.verify = ca_cert if ca_cert else True

    def _url(self, path: str) -> str:
        return self.api_server + path

    def list(self, path: str, params: Optional[Dict[str, Any]] = None) -> dict:
        r = self.session.get(self._url(path), params=params or {}, timeout=self.timeout_s)
        r.raise_for_status()
        return r.json()

    def watch(self, path: str, resource_version: str,
              timeout_s: int = 300) -> Iterator[dict]:
        """Stream watch events ({'type': 'ADDED'|'MODIFIED'|'DELETED'|...,
        'object': {...}}) until the server closes the connection."""
        params = {
            "watch": "true",
            "resourceVersion": resource_version,
            "timeoutSeconds": timeout_s,
            "allowWatchBookmarks": "true",
        }
        with self.session.get(self._url(path), params=params, stream=True,
                              timeout=timeout_s + 30) as r:
            r.raise_for_status()
            for line in r.iter_lines():
                if line:
                    yield json.loads(line)

    def bind_pod(self, namespace: str, name: str, uid: str, node: str,
                 annotations: Dict[str, str]) -> None:
        binding = {
            "apiVersion": "v1",
            "kind": "Binding",
            "metadata": {
                "namespace": namespace,
                "name": name,
                "uid": uid,
                "annotations": annotations,
            },
            "target": {"kind": "Node", "apiVersion": "v1", "name": node},
        }
        r = self.session.post(
            self._url(f"/api/v1/namespaces/{namespace}/pods/{name}/binding"),
            json=binding, timeout=self.timeout_s)
        r.raise_for_status()
