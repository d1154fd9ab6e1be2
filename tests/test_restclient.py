"""RestKubeClient against a mock Kubernetes API server (HTTP).

The production client's URL shapes, auth header, strategic-merge patches
and Binding POST — the reference's thin patch/list helpers
(pkg/util/util.go:273-319, k8sutil/client.go)."""
import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from k8s_device_plugin_amd.utils.kubeclient import (
    ConflictError,
    KubeError,
    RestKubeClient,
)


class MockAPI(BaseHTTPRequestHandler):
    store = {}

    def log_message(self, *a):
        pass

    def _send(self, code, obj):
        body = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_GET(self):
        s = type(self).store
        s.setdefault("auth", []).append(self.headers.get("Authorization"))
        if self.path == "/api/v1/nodes":
            self._send(200, {"items": list(s["nodes"].values())})
        elif self.path.startswith("/api/v1/nodes/"):
            name = self.path.rsplit("/", 1)[1]
            if name in s["nodes"]:
                self._send(200, s["nodes"][name])
            else:
                self._send(404, {"message": "not found"})
        elif self.path == "/api/v1/pods":
            self._send(200, {"items": list(s["pods"].values())})
        elif "/pods/" in self.path:
            name = self.path.rsplit("/", 1)[1]
            if name in s["pods"]:
                self._send(200, s["pods"][name])
            else:
                self._send(404, {"message": "not found"})
        else:
            self._send(404, {})

    def do_PATCH(self):
        s = type(self).store
        assert self.headers["Content-Type"] == \
            "application/strategic-merge-patch+json"
        n = int(self.headers["Content-Length"])
        patch = json.loads(self.rfile.read(n))
        name = self.path.rsplit("/", 1)[1]
        coll = s["nodes"] if "/nodes/" in self.path else s["pods"]
        if name == "conflicted":
            self._send(409, {"message": "conflict"})
            return
        if name not in coll:
            self._send(404, {"message": "not found"})
            return
        annos = coll[name]["metadata"].setdefault("annotations", {})
        for k, v in patch["metadata"]["annotations"].items():
            if v is None:
                annos.pop(k, None)
            else:
                annos[k] = v
        self._send(200, coll[name])

    def do_POST(self):
        s = type(self).store
        if self.path.endswith("/binding"):
            n = int(self.headers["Content-Length"])
            s["bindings"].append(json.loads(self.rfile.read(n)))
            self._send(201, {"kind": "Status", "status": "Success"})
        else:
            self._send(404, {})


@pytest.fixture
def api():
    MockAPI.store = {
        "nodes": {"n1": {"metadata": {"name": "n1", "annotations": {}}}},
        "pods": {
            "p1": {"kind": "Pod",
                   "metadata": {"name": "p1", "namespace": "default",
                                "uid": "u1", "annotations": {}},
                   "spec": {"containers": []}},
            "conflicted": {"kind": "Pod",
                           "metadata": {"name": "conflicted",
                                        "namespace": "default", "uid": "u2"},
                           "spec": {"containers": []}},
        },
        "bindings": [],
    }
    srv = ThreadingHTTPServer(("127.0.0.1", 0), MockAPI)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_address[1]}"
    srv.shutdown()
    srv.server_close()


def test_list_get_patch_bind(api):
    c = RestKubeClient(server=api, token="tok-123")
    nodes = c.list_nodes()
    assert [n.name for n in nodes] == ["n1"]
    assert MockAPI.store["auth"][0] == "Bearer tok-123"

    pods = c.list_pods()
    assert {p.name for p in pods} == {"p1", "conflicted"}
    assert c.get_pod("p1").uid == "u1"

    c.patch_node_annotations("n1", {"amd.io/node-handshake": "Reported x"})
    assert MockAPI.store["nodes"]["n1"]["metadata"]["annotations"][
        "amd.io/node-handshake"] == "Reported x"
    # None deletes (lock release semantics)
    c.patch_node_annotations("n1", {"amd.io/node-handshake": None})
    assert "amd.io/node-handshake" not in \
        MockAPI.store["nodes"]["n1"]["metadata"]["annotations"]

    c.bind_pod("p1", "default", "n1")
    assert MockAPI.store["bindings"][0]["target"]["name"] == "n1"


def test_error_mapping(api):
    c = RestKubeClient(server=api, token="t")
    with pytest.raises(KubeError):
        c.get_pod("missing")
    with pytest.raises(ConflictError):
        c.patch_pod_annotations("conflicted", "default", {"a": "b"})


def test_requires_server_address(monkeypatch):
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    with pytest.raises(KubeError):
        RestKubeClient()
