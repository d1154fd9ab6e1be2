"""Thin Kubernetes API client abstraction.

No kubernetes client package exists in this offline image, and the reference's
client-go usage is confined to list/patch/bind helpers
(/root/reference/pkg/util/util.go:273-319, pkg/k8sutil/client.go), so the
whole stack talks to this small interface instead.  Two implementations:

- ``FakeKubeClient``: in-memory, used by every unit test (the reference's
  testing strategy keeps k8s behind thin helpers for the same reason,
  SURVEY.md §4).
- ``RestKubeClient``: straight REST against the API server using ``requests``
  with in-cluster service-account credentials or a kubeconfig token.
"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List, Optional

from .types import NodeInfo, PodInfo


class KubeError(RuntimeError):
    pass


class ConflictError(KubeError):
    """409 from the API server (annotation CAS races)."""


class KubeClient:
    """Interface: the minimal verbs the stack needs."""

    def list_pods(self, namespace: str = "") -> List[PodInfo]:
        raise NotImplementedError

    def list_nodes(self) -> List[NodeInfo]:
        raise NotImplementedError

    def get_node(self, name: str) -> NodeInfo:
        raise NotImplementedError

    def get_pod(self, name: str, namespace: str = "default") -> PodInfo:
        raise NotImplementedError

    def patch_node_annotations(self, name: str, annotations: Dict[str, Optional[str]]) -> None:
        raise NotImplementedError

    def patch_pod_annotations(
        self, name: str, namespace: str, annotations: Dict[str, Optional[str]]
    ) -> None:
        raise NotImplementedError

    def bind_pod(self, name: str, namespace: str, node: str) -> None:
        raise NotImplementedError


class FakeKubeClient(KubeClient):
    """In-memory cluster for tests: plain dicts behind a lock."""

    def __init__(self):
        self._lock = threading.RLock()
        self.nodes: Dict[str, NodeInfo] = {}
        self.pods: Dict[str, PodInfo] = {}  # key: ns/name
        self.bindings: List[tuple] = []

    # -- test helpers -----------------------------------------------------
    def add_node(self, node: NodeInfo) -> None:
        with self._lock:
            self.nodes[node.name] = node

    def add_pod(self, pod: PodInfo) -> None:
        with self._lock:
            self.pods[f"{pod.namespace}/{pod.name}"] = pod

    def delete_pod(self, name: str, namespace: str = "default") -> None:
        with self._lock:
            self.pods.pop(f"{namespace}/{name}", None)

    # -- KubeClient -------------------------------------------------------
    def list_pods(self, namespace: str = "") -> List[PodInfo]:
        with self._lock:
            if namespace:
                return [p for p in self.pods.values() if p.namespace == namespace]
            return list(self.pods.values())

    def list_nodes(self) -> List[NodeInfo]:
        with self._lock:
            return list(self.nodes.values())

    def get_node(self, name: str) -> NodeInfo:
        with self._lock:
            if name not in self.nodes:
                raise KubeError(f"node {name} not found")
            return self.nodes[name]

    def get_pod(self, name: str, namespace: str = "default") -> PodInfo:
        with self._lock:
            key = f"{namespace}/{name}"
            if key not in self.pods:
                raise KubeError(f"pod {key} not found")
            return self.pods[key]

    def patch_node_annotations(self, name, annotations):
        with self._lock:
            node = self.get_node(name)
            _apply_annotation_patch(node.annotations, annotations)

    def patch_pod_annotations(self, name, namespace, annotations):
        with self._lock:
            pod = self.get_pod(name, namespace)
            _apply_annotation_patch(pod.annotations, annotations)

    def bind_pod(self, name, namespace, node):
        with self._lock:
            pod = self.get_pod(name, namespace)
            pod.node_name = node
            self.bindings.append((namespace, name, node))


def _apply_annotation_patch(dst: Dict[str, str], patch: Dict[str, Optional[str]]) -> None:
    # Strategic-merge semantics for map fields: None deletes the key.
    for k, v in patch.items():
        if v is None:
            dst.pop(k, None)
        else:
            dst[k] = str(v)


class RestKubeClient(KubeClient):
    """Direct REST client (in-cluster or kubeconfig-token auth)."""

    SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"

    def __init__(self, server: str = "", token: str = "", verify=None):
        import requests  # installed in the image

        self._requests = requests
        if not server:
            host = os.environ.get("KUBERNETES_SERVICE_HOST", "")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            if host:
                server = f"https://{host}:{port}"
        if not server:
            raise KubeError("no API server address (KUBERNETES_SERVICE_HOST unset)")
        self.server = server.rstrip("/")
        if not token and os.path.exists(f"{self.SA_DIR}/token"):
            with open(f"{self.SA_DIR}/token") as f:
                token = f.read().strip()
        self.session = requests.Session()
        if token:
            self.session.headers["Authorization"] = f"Bearer {token}"
        if verify is None:
            ca = f"{self.SA_DIR}/ca.crt"
            verify = ca if os.path.exists(ca) else False
        self.session.verify = verify

    def _url(self, path: str) -> str:
        return f"{self.server}{path}"

    def _check(self, r):
        if r.status_code == 409:
            raise ConflictError(r.text)
        if r.status_code >= 300:
            raise KubeError(f"{r.status_code}: {r.text[:500]}")
        return r

    def list_pods(self, namespace: str = "") -> List[PodInfo]:
        path = f"/api/v1/namespaces/{namespace}/pods" if namespace else "/api/v1/pods"
        r = self._check(self.session.get(self._url(path), timeout=30))
        return [PodInfo.from_k8s(item) for item in r.json().get("items", [])]

    def list_nodes(self) -> List[NodeInfo]:
        r = self._check(self.session.get(self._url("/api/v1/nodes"), timeout=30))
        return [NodeInfo.from_k8s(item) for item in r.json().get("items", [])]

    def get_node(self, name: str) -> NodeInfo:
        r = self._check(self.session.get(self._url(f"/api/v1/nodes/{name}"), timeout=30))
        return NodeInfo.from_k8s(r.json())

    def get_pod(self, name: str, namespace: str = "default") -> PodInfo:
        r = self._check(
            self.session.get(self._url(f"/api/v1/namespaces/{namespace}/pods/{name}"), timeout=30)
        )
        return PodInfo.from_k8s(r.json())

    def _patch_annotations(self, path: str, annotations) -> None:
        body = json.dumps({"metadata": {"annotations": annotations}})
        r = self.session.patch(
            self._url(path),
            data=body,
            headers={"Content-Type": "application/strategic-merge-patch+json"},
            timeout=30,
        )
        self._check(r)

    def patch_node_annotations(self, name, annotations):
        self._patch_annotations(f"/api/v1/nodes/{name}", annotations)

    def patch_pod_annotations(self, name, namespace, annotations):
        self._patch_annotations(f"/api/v1/namespaces/{namespace}/pods/{name}", annotations)

    def bind_pod(self, name, namespace, node):
        body = {
            "apiVersion": "v1",
            "kind": "Binding",
            "metadata": {"name": name, "namespace": namespace},
            "target": {"apiVersion": "v1", "kind": "Node", "name": node},
        }
        r = self.session.post(
            self._url(f"/api/v1/namespaces/{namespace}/pods/{name}/binding"),
            json=body,
            timeout=30,
        )
        self._check(r)
