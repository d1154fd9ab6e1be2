"""HTTP routes for the scheduler extender.

Reference: /root/reference/pkg/scheduler/routes/route.go:41-134 — POST
``/filter`` (ExtenderArgs -> ExtenderFilterResult), POST ``/bind``
(ExtenderBindingArgs -> ExtenderBindingResult), POST ``/webhook``
(AdmissionReview).  Implemented on the stdlib HTTP server (threaded) so the
extender has zero web-framework dependencies; handlers are small shims over
pure functions, which is where the tests bite.
"""
from __future__ import annotations

import json
import logging
import ssl
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

from ..utils.types import PodInfo
from .core import Scheduler
from .webhook import handle_admission_review

log = logging.getLogger(__name__)


def handle_filter(scheduler: Scheduler, args: dict) -> dict:
    """ExtenderArgs JSON -> ExtenderFilterResult JSON."""
    pod_obj = args.get("Pod") or args.get("pod") or {}
    pod = PodInfo.from_k8s(pod_obj)
    node_names = args.get("NodeNames") or args.get("nodenames") or []
    res = scheduler.filter(pod, list(node_names))
    out = {
        "NodeNames": res.node_names or None,
        "FailedNodes": res.failed_nodes or None,
        "Error": res.error,
    }
    return out


def handle_bind(scheduler: Scheduler, args: dict) -> dict:
    """ExtenderBindingArgs JSON -> ExtenderBindingResult JSON."""
    pod_name = args.get("PodName") or args.get("podName") or ""
    pod_ns = args.get("PodNamespace") or args.get("podNamespace") or "default"
    node = args.get("Node") or args.get("node") or ""
    res = scheduler.bind(pod_name, pod_ns, node)
    return {"Error": res.error}


class _Handler(BaseHTTPRequestHandler):
    scheduler: Scheduler = None
    scheduler_name: str = "vgpu-scheduler"
    metrics_fn = None

    def log_message(self, fmt, *a):  # route to logging, not stderr
        log.debug("http: " + fmt, *a)

    def _send_json(self, code: int, obj) -> None:
        body = json.dumps(obj).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def _read_json(self) -> Optional[dict]:
        try:
            n = int(self.headers.get("Content-Length", "0"))
            return json.loads(self.rfile.read(n) or b"{}")
        except (ValueError, json.JSONDecodeError):
            return None

    def do_POST(self):
        body = self._read_json()
        if body is None:
            self._send_json(400, {"Error": "bad request body"})
            return
        try:
            if self.path.startswith("/filter"):
                self._send_json(200, handle_filter(self.scheduler, body))
            elif self.path.startswith("/bind"):
                self._send_json(200, handle_bind(self.scheduler, body))
            elif self.path.startswith("/webhook"):
                self._send_json(
                    200, handle_admission_review(body, self.scheduler_name)
                )
            else:
                self._send_json(404, {"Error": f"no route {self.path}"})
        except Exception as e:
            log.exception("handler error")
            self._send_json(500, {"Error": str(e)})

    def do_GET(self):
        if self.path.startswith("/healthz"):
            self._send_json(200, {"ok": True})
        elif self.path.startswith("/metrics") and self.metrics_fn is not None:
            body = self.metrics_fn()
            self.send_response(200)
            self.send_header("Content-Type", "text/plain; version=0.0.4")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)
        else:
            self._send_json(404, {"Error": f"no route {self.path}"})


class ExtenderServer:
    """Threaded HTTP(S) server hosting the extender verbs."""

    def __init__(
        self,
        scheduler: Scheduler,
        host: str = "127.0.0.1",
        port: int = 0,
        cert_file: str = "",
        key_file: str = "",
        scheduler_name: str = "vgpu-scheduler",
        metrics_fn=None,
    ):
        handler = type(
            "BoundHandler",
            (_Handler,),
            {
                "scheduler": scheduler,
                "scheduler_name": scheduler_name,
                "metrics_fn": staticmethod(metrics_fn) if metrics_fn else None,
            },
        )
        self.httpd = ThreadingHTTPServer((host, port), handler)
        if cert_file and key_file:
            ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
            ctx.load_cert_chain(cert_file, key_file)
            self.httpd.socket = ctx.wrap_socket(self.httpd.socket, server_side=True)
        self._thread: Optional[threading.Thread] = None

    @property
    def port(self) -> int:
        return self.httpd.server_address[1]

    def start(self) -> None:
        self._thread = threading.Thread(target=self.httpd.serve_forever, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self.httpd.shutdown()
        self.httpd.server_close()
        if self._thread:
            self._thread.join(timeout=5)
