"""CDI (Container Device Interface) spec generation for MI355X.

Reference behavior: pkg/device-plugin/nvidiadevice/nvinternal/cdi/cdi.go —
at plugin start a CDI spec is written for every physical GPU so CDI-aware
runtimes can inject the device nodes, and Allocate can alternatively hand
the kubelet ``cdi.k8s.io/*`` annotations instead of raw DeviceSpecs
(plugin/server.go:446-498).

MI355X devices are simpler than NVIDIA's (no driver-library discovery hooks
needed): a GPU is `/dev/kfd` (shared compute node) + its
`/dev/dri/card<N>` + `/dev/dri/renderD<minor>` render nodes, and the
enforcement artifacts ride in as mounts exactly as in the non-CDI path.
"""
from __future__ import annotations

import json
import logging
import os
from typing import Dict, List

from .kfd import PhysicalGPU

log = logging.getLogger(__name__)

CDI_VERSION = "0.5.0"
CDI_VENDOR = "amd.com"
CDI_CLASS = "gpu"
CDI_KIND = f"{CDI_VENDOR}/{CDI_CLASS}"
DEFAULT_SPEC_DIR = "/var/run/cdi"
ANNOTATION_PREFIX = "cdi.k8s.io/"


def _device_nodes(gpu: PhysicalGPU) -> List[Dict]:
    nodes = [{"path": "/dev/kfd", "permissions": "rw"}]
    for path in gpu.device_paths:
        if path != "/dev/kfd":
            nodes.append({"path": path, "permissions": "rw"})
    return nodes


def generate_spec(gpus: List[PhysicalGPU], hook_path: str = "") -> Dict:
    """One spec, one CDI device per physical GPU named by UUID."""
    devices = []
    for g in gpus:
        edits = {"deviceNodes": _device_nodes(g)}
        if hook_path:
            lib = os.path.join(hook_path, "vgpu", "libvgpu-hip.so")
            edits["mounts"] = [
                {
                    "hostPath": lib,
                    "containerPath": "/usr/local/vgpu/libvgpu-hip.so",
                    "options": ["ro", "nosuid", "nodev", "bind"],
                },
                {
                    "hostPath": os.path.join(hook_path, "vgpu", "ld.so.preload"),
                    "containerPath": "/etc/ld.so.preload",
                    "options": ["ro", "nosuid", "nodev", "bind"],
                },
            ]
        devices.append({"name": g.uuid, "containerEdits": edits})
    if gpus:
        # composite "all" device (the reference's nvcdi generates one,
        # cdi/cdi.go): every GPU's nodes behind a single CDI name
        seen = set()
        union = []
        for g in gpus:
            for node in _device_nodes(g):
                if node["path"] not in seen:
                    seen.add(node["path"])
                    union.append(node)
        devices.append({"name": "all",
                        "containerEdits": {"deviceNodes": union}})
    return {
        "cdiVersion": CDI_VERSION,
        "kind": CDI_KIND,
        "devices": devices,
    }


def write_spec(gpus: List[PhysicalGPU], spec_dir: str = DEFAULT_SPEC_DIR,
               hook_path: str = "") -> str:
    """Write (atomically) the node's CDI spec; returns the path."""
    spec = generate_spec(gpus, hook_path)
    os.makedirs(spec_dir, exist_ok=True)
    path = os.path.join(spec_dir, f"{CDI_VENDOR}-{CDI_CLASS}.json")
    tmp = path + ".tmp"
    with open(tmp, "w") as f:
        json.dump(spec, f, indent=2)
    os.replace(tmp, path)
    log.info("CDI spec for %d GPUs -> %s", len(gpus), path)
    return path


def annotations(uuids: List[str], prefix: str = "vgpu-amd") -> Dict[str, str]:
    """kubelet Allocate-response annotations naming the CDI devices
    (reference cdiAnnotations, server.go:475-498)."""
    if not uuids:
        return {}
    names = ",".join(f"{CDI_KIND}={u}" for u in uuids)
    return {ANNOTATION_PREFIX + prefix: names}


def parse_annotation(value: str) -> List[str]:
    """'amd.com/gpu=UUID1,amd.com/gpu=UUID2' -> [UUID1, UUID2]."""
    out = []
    for part in value.split(","):
        part = part.strip()
        if not part:
            continue
        kind, _, name = part.partition("=")
        if kind == CDI_KIND and name:
            out.append(name)
    return out
