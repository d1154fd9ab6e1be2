"""Device-plugin daemon entry point.

Reference behavior: cmd/device-plugin/nvidia/main.go:38-238 — parse flags
(+ per-node JSON ConfigMap override), build the resource managers, start the
gRPC plugin, and run a restart loop that tears the plugin down and brings it
back up when

- the kubelet re-creates its socket (kubelet restart: the plugin must
  re-Register, watched via inotify in the reference, polled mtime/inode
  here), or
- SIGHUP is received (config reload).

SIGTERM/SIGINT exit the loop cleanly.  Run as
``python -m k8s_device_plugin_amd.plugin.main`` (DaemonSet container).
"""
from __future__ import annotations

import logging
import os
import signal
import sys
import threading
import time
from typing import Optional

from ..utils.kubeclient import KubeClient, KubeError, RestKubeClient
from .config import PluginConfig, parse_args
from .health import HealthChecker
from .kfd import enumerate_gpus, kfd_healthy
from .register import RegisterLoop
from .rm import ResourceManager
from .server import VGPUDevicePlugin

log = logging.getLogger(__name__)


def build_resource_manager(cfg: PluginConfig) -> ResourceManager:
    if cfg.compute_partition.upper() != "KEEP":
        from . import partition

        if not partition.apply_mode(cfg.compute_partition):
            log.error(
                "could not apply compute partition %s (GPU busy or "
                "unsupported); advertising current partitioning",
                cfg.compute_partition)
    gpus = enumerate_gpus()
    if not gpus:
        log.warning("no AMD GPUs found in KFD topology")
    for g in gpus:
        log.info(
            "GPU %d: %s %s cu=%d mem=%d MiB numa=%d bdf=%s",
            g.index, g.uuid, g.gfx_target, g.cu_count,
            g.mem_bytes >> 20, g.numa_node, g.pci_bdf,
        )
    return ResourceManager(
        gpus,
        split_count=cfg.device_split_count,
        memory_scaling=cfg.device_memory_scaling,
        cores_scaling=cfg.device_cores_scaling,
        replica_overrides=cfg.replica_overrides,
    )


class _SocketWatch:
    """Poll the kubelet socket identity; True once it has been re-created.

    The reference uses fsnotify on /var/lib/kubelet/device-plugins
    (main.go:199-231); a 1 s inode/mtime poll has the same restart semantics
    without requiring inotify in minimal containers.
    """

    def __init__(self, path: str):
        self.path = path
        self.ident = self._ident()

    def _ident(self):
        try:
            st = os.stat(self.path)
            return (st.st_ino, st.st_mtime_ns)
        except OSError:
            return None

    def changed(self) -> bool:
        now = self._ident()
        if now != self.ident:
            # only a *new* socket should trigger re-registration; a vanished
            # socket means kubelet is down — wait for it to come back
            if now is not None:
                self.ident = now
                return True
            self.ident = now
        return False


class PluginDaemon:
    def __init__(self, cfg: PluginConfig, client: Optional[KubeClient] = None):
        self.cfg = cfg
        self.client = client or RestKubeClient()
        self._stop = threading.Event()
        self._hup = threading.Event()

    def request_stop(self, *_):
        self._stop.set()

    def request_reload(self, *_):
        self._hup.set()

    def run_session(self) -> str:
        """One plugin lifetime; returns why it ended
        ('stop' | 'reload' | 'kubelet-restart')."""
        cfg = self.cfg
        rm = build_resource_manager(cfg)
        try:
            from . import cdi as cdimod

            if rm.gpus:
                cdimod.write_spec(rm.gpus, cfg.cdi_spec_dir, cfg.hook_path)
        except OSError as e:
            log.warning("CDI spec write failed: %s", e)
        plugin = VGPUDevicePlugin(cfg, rm, self.client)
        plugin.serve()
        try:
            plugin.register_with_kubelet()
        except Exception as e:
            log.error("kubelet registration failed: %s (will keep serving)", e)
        reg = RegisterLoop(self.client, cfg.node_name, rm,
                           interval_s=cfg.register_interval_s)
        reg.start()
        health = HealthChecker(rm, on_change=plugin.notify_update,
                               interval_s=cfg.health_interval_s)
        health.start()
        watch = _SocketWatch(cfg.kubelet_socket)

        def reconcile_once():
            try:
                live = {p.uid for p in self.client.list_pods()}
                plugin.reconcile(live)
            except Exception as e:
                log.warning("reconcile failed: %s", e)

        reconcile_once()  # adopt pre-restart allocations before new Allocates
        reason = "stop"
        ticks = 0
        try:
            while not self._stop.is_set():
                if self._hup.is_set():
                    self._hup.clear()
                    reason = "reload"
                    break
                if watch.changed():
                    log.info("kubelet socket re-created; restarting plugin")
                    reason = "kubelet-restart"
                    break
                ticks += 1
                if ticks % 30 == 0:
                    reconcile_once()
                time.sleep(1.0)
        finally:
            health.stop()
            reg.stop()
            plugin.stop()
        return reason

    def run(self) -> None:
        backoff = 1.0
        while not self._stop.is_set():
            try:
                reason = self.run_session()
                log.info("plugin session ended: %s", reason)
                backoff = 1.0
            except Exception:
                log.exception("plugin session crashed; retrying in %.0fs", backoff)
                self._stop.wait(backoff)
                backoff = min(backoff * 2, 30.0)


def main(argv=None) -> int:
    logging.basicConfig(
        level=os.environ.get("LOG_LEVEL", "INFO"),
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )
    cfg = parse_args(argv)
    if not cfg.node_name:
        log.error("node name required (--node-name or $NODE_NAME)")
        return 2
    if not kfd_healthy():
        log.warning("/dev/kfd not openable — running without enforcement-capable GPUs")
    try:
        daemon = PluginDaemon(cfg)
    except KubeError as e:
        log.error("kubernetes API unreachable: %s", e)
        return 2
    signal.signal(signal.SIGTERM, daemon.request_stop)
    signal.signal(signal.SIGINT, daemon.request_stop)
    signal.signal(signal.SIGHUP, daemon.request_reload)
    daemon.run()
    return 0


if __name__ == "__main__":
    sys.exit(main())
