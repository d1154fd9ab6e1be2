"""Scheduler-extender daemon entry point.

Reference behavior: cmd/scheduler/main.go:48-94 — flags ``--http_bind``,
``--cert_file``/``--key_file`` (the webhook endpoint must be HTTPS),
``--scheduler-name``, ``--default-mem``, ``--default-cores``,
``--metrics-bind-address``; starts the node-registry sync loop, the pod-cache
rebuild, and the HTTP(S) server hosting /filter, /bind, /webhook, /metrics,
/healthz.  Run as ``python -m k8s_device_plugin_amd.scheduler.main``.
"""
from __future__ import annotations

import argparse
import logging
import os
import signal
import sys
import threading

from ..device import init_devices
from ..utils.kubeclient import KubeError, RestKubeClient
from .core import Scheduler
from .metrics import metrics_text
from .routes import ExtenderServer

log = logging.getLogger(__name__)


def parse_args(argv=None):
    p = argparse.ArgumentParser("amd-vgpu-scheduler")
    p.add_argument("--http_bind", default="0.0.0.0:443")
    p.add_argument("--cert_file", default="")
    p.add_argument("--key_file", default="")
    p.add_argument("--scheduler-name", default="vgpu-scheduler")
    p.add_argument("--default-mem", type=int, default=0,
                   help="MiB assumed when a pod requests amd.com/gpu without gpumem")
    p.add_argument("--default-cores", type=int, default=0,
                   help="core %% assumed when a pod omits amd.com/gpucores")
    p.add_argument("--resource-name", default="amd.com/gpu")
    p.add_argument("--resource-mem", default="amd.com/gpumem")
    p.add_argument("--resource-mem-percentage", default="amd.com/gpumem-percentage")
    p.add_argument("--resource-cores", default="amd.com/gpucores")
    p.add_argument("--resource-priority", default="amd.com/priority")
    p.add_argument("--metrics-bind-address", default=":9395")
    return p.parse_args(argv)


def main(argv=None) -> int:
    logging.basicConfig(
        level=os.environ.get("LOG_LEVEL", "INFO"),
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )
    args = parse_args(argv)
    init_devices(
        resource_name=args.resource_name,
        resource_mem=args.resource_mem,
        resource_mem_percentage=args.resource_mem_percentage,
        resource_cores=args.resource_cores,
        resource_priority=args.resource_priority,
        default_mem=args.default_mem,
        default_cores=args.default_cores,
    )
    host, _, port = args.http_bind.rpartition(":")
    try:
        client = RestKubeClient()
    except KubeError as e:
        log.error("kubernetes API unreachable: %s", e)
        return 2
    sched = Scheduler(client)
    sched.rebuild_pod_cache()

    reg_thread = threading.Thread(target=sched.register_loop, daemon=True)
    reg_thread.start()

    server = ExtenderServer(
        sched,
        host=host or "0.0.0.0",
        port=int(port or 443),
        cert_file=args.cert_file,
        key_file=args.key_file,
        scheduler_name=args.scheduler_name,
        metrics_fn=lambda: metrics_text(sched),
    )
    server.start()
    log.info("extender serving on %s:%d (tls=%s)", host or "0.0.0.0",
             server.port, bool(args.cert_file))

    # Separate plaintext metrics listener (reference serves Prometheus on
    # :9395 next to the TLS extender, cmd/scheduler/main.go:58).
    metrics_server = None
    if args.metrics_bind_address:
        mhost, _, mport = args.metrics_bind_address.rpartition(":")
        if int(mport or 0) != server.port:
            metrics_server = ExtenderServer(
                sched, host=mhost or "0.0.0.0", port=int(mport or 9395),
                scheduler_name=args.scheduler_name,
                metrics_fn=lambda: metrics_text(sched),
            )
            metrics_server.start()
            log.info("metrics serving on %s:%d", mhost or "0.0.0.0", metrics_server.port)

    stop = threading.Event()

    def _sig(*_):
        stop.set()

    signal.signal(signal.SIGTERM, _sig)
    signal.signal(signal.SIGINT, _sig)
    stop.wait()
    server.stop()
    if metrics_server:
        metrics_server.stop()
    sched.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
