"""Node registration loop: advertise the inventory via node annotations.

Reference: plugin/register.go:164-200 — every 30 s (5 s after an error)
patch ``amd.io/node-handshake = Reported <time>`` +
``amd.io/node-amd-register = <encoded devices>``.
"""
from __future__ import annotations

import logging
import threading
import time

from ..device.amd import HANDSHAKE_ANNO, REGISTER_ANNO
from ..utils.codec import encode_node_devices, encode_node_xgmi
from ..utils.kubeclient import KubeClient
from ..utils.types import XGMI_ANNO
from .rm import ResourceManager

XGMI_LINK_TYPE = 11  # KFD io_link type for xGMI (topology.py)


def xgmi_adjacency(gpus):
    """{uuid: [xGMI peer uuids]} from the KFD io_links (7 p2p links/GPU on
    an 8-GPU MI355X node)."""
    node_to_uuid = {g.node_id: g.uuid for g in gpus}
    adj = {}
    for g in gpus:
        peers = [node_to_uuid[n] for n, t in g.io_links.items()
                 if t == XGMI_LINK_TYPE and n in node_to_uuid]
        if peers:
            adj[g.uuid] = peers
    return adj

log = logging.getLogger(__name__)


def register_once(client: KubeClient, node_name: str, rm: ResourceManager) -> None:
    devices = rm.api_devices()
    annos = {
        HANDSHAKE_ANNO: "Reported " + time.strftime("%Y-%m-%d %H:%M:%S"),
        REGISTER_ANNO: encode_node_devices(devices),
    }
    adj = xgmi_adjacency(rm.gpus)
    if adj:
        annos[XGMI_ANNO] = encode_node_xgmi(adj)
    client.patch_node_annotations(node_name, annos)


class RegisterLoop:
    def __init__(self, client: KubeClient, node_name: str, rm: ResourceManager,
                 interval_s: float = 30.0, error_interval_s: float = 5.0):
        self.client = client
        self.node_name = node_name
        self.rm = rm
        self.interval_s = interval_s
        self.error_interval_s = error_interval_s
        self._stop = threading.Event()
        self._thread = None

    def run(self) -> None:
        while not self._stop.is_set():
            try:
                register_once(self.client, self.node_name, self.rm)
                wait = self.interval_s
            except Exception as e:
                log.error("node registration failed: %s", e)
                wait = self.error_interval_s
            self._stop.wait(wait)

    def start(self) -> None:
        self._thread = threading.Thread(target=self.run, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
